"""Tabular drift detector with alibi-detect ``TabularDrift`` semantics.

The reference fits ``TabularDrift(ref.values, p_val=0.05,
categories_per_feature={0..8: None})`` on the full training feature matrix
(reference 02-register-model.ipynb cell-6) and per request batch reports
``1 - p_val`` per feature (cell-9). alibi-detect 0.12 semantics:

- categorical feature: chi-square test on the 2 x k contingency table of
  (reference counts, batch counts) — scipy.stats.chi2_contingency;
- numeric feature: two-sample Kolmogorov-Smirnov test, asymptotic p-value —
  scipy.stats.ks_2samp(..., method="asymp");
- per-feature p-values are returned uncorrected; the batch-level ``is_drift``
  decision applies a Bonferroni correction (threshold = p_val / n_features).

Categories are inferred from the reference data; batch values outside the
reference vocabulary are added as extra contingency columns (the natural
generalisation — the reference never exercises unseen categories because its
classifier uses handle_unknown="ignore").

The GPU path (csrc/kernels/drift.hip) computes the K-S D statistic and the
categorical histograms on-device; this module's ``chi2_from_counts`` /
``ks_asymp_pvalue`` convert those statistics to p-values identically on host,
so CPU and GPU paths share the final numerics.
"""

from __future__ import annotations

import numpy as np
from scipy import stats


def ks_asymp_pvalue(d: float, n_ref: int, n_batch: int) -> float:
    """Asymptotic two-sided two-sample K-S p-value from the D statistic.

    Matches scipy.stats.ks_2samp(method="asymp") (verified by test):
    en = n*m/(n+m); p = kstwo.sf(d, round(en))  (Smirnov's asymptotic form,
    scipy/stats/_stats_py.py ks_2samp 'asymp' branch).
    """
    en = float(n_ref) * float(n_batch) / (float(n_ref) + float(n_batch))
    return float(np.clip(stats.kstwo.sf(float(d), int(np.round(en))), 0.0, 1.0))


def ks_2samp_d(ref_sorted: np.ndarray, batch: np.ndarray) -> float:
    """Exact two-sample K-S D statistic, reference pre-sorted.

    Evaluates sup|F_ref - F_batch| at every batch jump point from both sides;
    this equals the sup over the combined sample (both CDFs are step
    functions, and between batch jumps F_batch is constant while F_ref is
    monotone, so extrema occur at batch points' left/right limits).
    This is the same algorithm the HIP kernel uses.
    """
    n = len(ref_sorted)
    b = np.sort(np.asarray(batch, dtype=np.float64))
    m = len(b)
    sl = np.searchsorted(ref_sorted, b, side="left") / n
    sr = np.searchsorted(ref_sorted, b, side="right") / n
    # With ties inside the batch, F_batch's one-sided limits at value v are
    # the run bounds (first index / last index + 1), not the per-element rank.
    bl = np.searchsorted(b, b, side="left") / m
    br = np.searchsorted(b, b, side="right") / m
    d = np.maximum(np.abs(sl - bl), np.abs(sr - br)).max()
    return float(d)


def chi2_from_counts(ref_counts: np.ndarray, batch_counts: np.ndarray) -> float:
    """Chi-square p-value from a 2 x k contingency table of counts.

    Drops categories absent from both samples (scipy raises on zero
    marginals). Uses scipy.stats.chi2_contingency, which applies the Yates
    continuity correction for 2 x 2 tables — same as alibi-detect's
    ChiSquareDrift calling convention.
    """
    rc = np.asarray(ref_counts, dtype=np.int64)
    bc = np.asarray(batch_counts, dtype=np.int64)
    keep = (rc + bc) > 0
    rc, bc = rc[keep], bc[keep]
    if len(rc) < 2 or rc.sum() == 0 or bc.sum() == 0:
        return 1.0
    table = np.stack([rc, bc])
    res = stats.chi2_contingency(table)
    return float(res.pvalue if hasattr(res, "pvalue") else res[1])


class TabularDriftDetector:
    """alibi-detect-TabularDrift-compatible detector.

    ``categorical_idx`` are the column indices (into the feature matrix)
    holding categorical values; all other columns are numeric. The reference
    passes indices 0..8 (02-register cell-6).
    """

    def __init__(
        self,
        x_ref: np.ndarray,
        p_val: float = 0.05,
        categorical_idx: tuple[int, ...] = tuple(range(9)),
    ):
        x_ref = np.asarray(x_ref, dtype=object)
        self.p_val = float(p_val)
        self.n_features = x_ref.shape[1]
        self.categorical_idx = tuple(categorical_idx)
        self.numeric_idx = tuple(
            i for i in range(self.n_features) if i not in self.categorical_idx
        )
        self.n_ref = x_ref.shape[0]
        # Per categorical feature: inferred vocabulary (sorted) + ref counts.
        self.categories: dict[int, list] = {}
        self.ref_counts: dict[int, np.ndarray] = {}
        for i in self.categorical_idx:
            vals, counts = np.unique(x_ref[:, i].astype(str), return_counts=True)
            self.categories[i] = list(vals)
            self.ref_counts[i] = counts.astype(np.int64)
        # Per numeric feature: sorted reference values (the exact-ECDF basis).
        self.ref_sorted: dict[int, np.ndarray] = {}
        for i in self.numeric_idx:
            self.ref_sorted[i] = np.sort(x_ref[:, i].astype(np.float64))

    def feature_pvals(self, x: np.ndarray) -> np.ndarray:
        x = np.asarray(x, dtype=object)
        pvals = np.ones(self.n_features, dtype=np.float64)
        for i in self.categorical_idx:
            col = x[:, i].astype(str)
            cats = list(self.categories[i])
            extra = sorted(set(col) - set(cats))
            all_cats = cats + extra
            idx = {c: j for j, c in enumerate(all_cats)}
            bc = np.zeros(len(all_cats), dtype=np.int64)
            for v in col:
                bc[idx[v]] += 1
            rc = np.zeros(len(all_cats), dtype=np.int64)
            rc[: len(cats)] = self.ref_counts[i]
            pvals[i] = chi2_from_counts(rc, bc)
        for i in self.numeric_idx:
            col = x[:, i].astype(np.float64)
            d = ks_2samp_d(self.ref_sorted[i], col)
            pvals[i] = ks_asymp_pvalue(d, self.n_ref, len(col))
        return pvals.astype(np.float32)

    def predict(self, x: np.ndarray) -> dict:
        """alibi-detect-shaped response (the reference reads
        ``["data"]["p_val"]``, 02-register cell-9)."""
        pvals = self.feature_pvals(x)
        threshold = self.p_val / self.n_features  # Bonferroni
        return {
            "data": {
                "is_drift": int((pvals < threshold).any()),
                "p_val": pvals,
                "threshold": threshold,
            },
            "meta": {"name": "TabularDriftDetector", "detector_type": "offline"},
        }
