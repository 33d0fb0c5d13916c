"""Linear / logistic scorer — the "linear predict" model family.

BASELINE.json's north star names "gradient-boosted-tree traversal / linear
predict" as the tabular scoring hot path. The forest kernel covers RF and
GBT traversal (creditcore.pack); dense linear scoring at scale lives in
creditcore.dense (fused dense_score_kernel: impute + w·x + sigmoid per
wavefront — the op is GEMV-shaped, so MFMA does not apply; batched
multi-output models would route through rocBLAS GEMM instead).

This module is the small CPU-side LinearScorer used for training-side
experiments and as the numpy reference for the dense family's logistic
head; training uses sklearn.linear_model.LogisticRegression on the encoded
matrix the forest pipeline produces.
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
