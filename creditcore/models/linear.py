"""Linear / logistic scorer — the "linear predict" model family.

BASELINE.json's north star names "gradient-boosted-tree traversal / linear
predict" as the tabular scoring hot path. The forest kernel covers tree
traversal; this family covers dense linear scoring, whose GPU hot path is an
MFMA-tiled GEMM on gfx950 (csrc/kernels/linear.hip): out = sigmoid(X @ W + b)
over the encoded (one-hot + imputed-numeric) feature matrix.

The CPU reference here is plain numpy fp64; training uses
sklearn.linear_model.LogisticRegression on the same encoded matrix the forest
pipeline produces.
"""

from __future__ import annotations

import numpy as np


class LinearScorer:
    """Dense logistic scorer over an encoded feature matrix."""

    def __init__(self, weight: np.ndarray, bias: float | np.ndarray):
        self.weight = np.asarray(weight, dtype=np.float64).reshape(-1)
        self.bias = float(np.asarray(bias).reshape(-1)[0])

    def predict_proba1(self, x_encoded: np.ndarray) -> np.ndarray:
        """P(class 1) per row; x_encoded is the dense B x F matrix."""
        z = np.asarray(x_encoded, dtype=np.float64) @ self.weight + self.bias
        return 1.0 / (1.0 + np.exp(-z))

    @classmethod
    def fit(cls, x_encoded: np.ndarray, y: np.ndarray, **kwargs) -> "LinearScorer":
        from sklearn.linear_model import LogisticRegression

        kwargs.setdefault("max_iter", 200)
        clf = LogisticRegression(**kwargs)
        clf.fit(x_encoded, y)
        return cls(clf.coef_[0], clf.intercept_[0])
