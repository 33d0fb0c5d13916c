"""Isolation-forest outlier detector with alibi-detect ``IForest`` semantics.

The reference packages ``alibi_detect.od.IForest(threshold=0.95)`` fitted on
the 14 numeric features (reference 02-register-model.ipynb cell-6) and reports
``is_outlier`` per row (cell-9). alibi-detect's IForest is a thin wrapper over
sklearn's IsolationForest:

    instance_score = -decision_function(X)         # = anomaly_score - 0.5
    is_outlier     = (instance_score > threshold)

(alibi-detect 0.12.0 semantics, pinned at reference app/requirements.txt:7.)

This module re-implements that wrapper on top of sklearn so the artifact can
be built and loaded without alibi-detect. The GPU path scores the same trees
via the packed representation (creditcore.pack.pack_isolation_forest).
"""

from __future__ import annotations

import numpy as np
from sklearn.ensemble import IsolationForest


class IForestDetector:
    """alibi-detect-compatible isolation-forest outlier detector."""

    def __init__(
        self,
        threshold: float = 0.95,
        n_estimators: int = 100,
        random_state: int | None = 2024,
    ):
        self.threshold = float(threshold)
        self.isolationforest = IsolationForest(
            n_estimators=n_estimators, random_state=random_state
        )

    def fit(self, x: np.ndarray) -> None:
        self.isolationforest.fit(np.asarray(x, dtype=np.float64))

    def score(self, x: np.ndarray) -> np.ndarray:
        """instance_score = -decision_function (alibi-detect convention)."""
        return -self.isolationforest.decision_function(np.asarray(x, dtype=np.float64))

    def predict(self, x: np.ndarray) -> dict:
        """Return the alibi-detect response envelope subset the reference uses
        (02-register cell-9 reads ``["data"]["is_outlier"]``)."""
        iscore = self.score(x)
        return {
            "data": {
                "is_outlier": (iscore > self.threshold).astype(int),
                "instance_score": iscore,
            },
            "meta": {"name": "IForestDetector", "detector_type": "offline"},
        }
