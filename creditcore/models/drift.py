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

The GPU path (csrc/creditcore_kernels.hip (ks_kernel_t / cat_hist_kernel)) computes the K-S D statistic and the
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


def ks_asymp_pvalue_many(ds: np.ndarray, n_ref: int, n_batch: int) -> np.ndarray:
    """Vectorized ks_asymp_pvalue for equal sample sizes across features.

    scipy's kstwo.sf costs ~0.1 ms *per value* (exact Pomeranz/Durbin path) —
    ~1.5 ms for the 14 numeric features, which would dominate request
    latency. For en >= 300 the Pelz-Good series agrees with the exact value
    to < 3e-7 absolute (verified in tests at en = 974: < 1e-8), so the hot
    path uses a vectorized Pelz-Good; smaller en falls back to exact scipy.
    """
    en = int(np.round(float(n_ref) * float(n_batch) / (float(n_ref) + float(n_batch))))
    ds = np.asarray(ds, dtype=np.float64)
    if en >= 300:
        return _pelz_good_sf(ds, en)
    return np.clip(stats.kstwo.sf(ds, en), 0.0, 1.0)


_PI2 = np.pi**2
_PI4 = np.pi**4
_PI6 = np.pi**6
_SQRT2PI = np.sqrt(2 * np.pi)
_SQRT3 = np.sqrt(3.0)


def _pelz_good_sf(xs: np.ndarray, n: int) -> np.ndarray:
    """Vectorized Pelz-Good approximation of the one-sample K-S survival
    function P(D_n > x) (Pelz & Good 1976; series form as in Simard &
    L'Ecuyer 2011): Prob(Dn <= x) ~ K0(z) + K1(z)/sqrt(n) + K2(z)/n +
    K3(z)/n^1.5 with z = x*sqrt(n), each K_i expressed through Jacobi theta
    series that converge fast for the z values drift testing produces."""
    xs = np.atleast_1d(xs).astype(np.float64)
    out = np.empty_like(xs)
    out[xs <= 0.0] = 1.0
    out[xs >= 1.0] = 0.0
    sel = (xs > 0.0) & (xs < 1.0)
    if not sel.any():
        return np.clip(out, 0.0, 1.0)
    z = np.sqrt(n) * xs[sel]
    z2, z3, z4, z6, z7, z8, z10 = z**2, z**3, z**4, z**6, z**7, z**8, z**10

    qlog = -_PI2 / 8.0 / z2
    tiny = qlog < -690.0  # exp underflow => cdf ~ 0 => sf ~ 1
    q = np.exp(np.where(tiny, -690.0, qlog))

    k1a = -z2
    k1b = _PI2 / 4.0
    k2a = 6 * z6 + 2 * z4
    k2b = (2 * z4 - 5 * z2) * _PI2 / 4.0
    k2c = _PI4 * (1 - 2 * z2) / 16.0
    k3d = _PI6 * (5 - 30 * z2) / 64.0
    k3c = _PI4 * (-60 * z2 + 212 * z4) / 16.0
    k3b = _PI2 * (135 * z4 - 96 * z6) / 4.0
    k3a = -30 * z6 - 90 * z8

    K0 = np.zeros_like(z)
    K1 = np.zeros_like(z)
    K2 = np.zeros_like(z)
    K3 = np.zeros_like(z)
    maxk = int(np.ceil(16 * z.max() / np.pi))
    for k in range(maxk, 0, -1):  # Horner over odd m = 2k-1 in powers of q
        m = 2 * k - 1
        m2, m4, m6 = m**2, m**4, m**6
        qp = q ** (8 * k)
        K0 = K0 * qp + 1.0
        K1 = K1 * qp + (k1a + k1b * m2)
        K2 = K2 * qp + (k2a + k2b * m2 + k2c * m4)
        K3 = K3 * qp + (k3a + k3b * m2 + k3c * m4 + k3d * m6)
    K0 *= q * _SQRT2PI / z
    K1 *= q * _SQRT2PI / (6 * z4)
    K2 *= q * _SQRT2PI / (72 * z7)
    K3 *= q * _SQRT2PI / (6480 * z10)

    # extra integer-k theta sums for K2, K3
    q2 = np.exp(-_PI2 / 2.0 / z2)
    ks = np.arange(1, maxk + 1, dtype=np.float64)
    k2_ = ks**2
    qpw = q2[:, None] ** k2_[None, :]
    K2 += (qpw @ k2_) * _PI2 * _SQRT2PI / (-36 * z3)
    kspi = np.pi * ks
    sqrt3z = _SQRT3 * z
    term = (sqrt3z[:, None] + kspi[None, :]) * (sqrt3z[:, None] - kspi[None, :])
    K3 += ((term * qpw) @ k2_) * _PI2 * _SQRT2PI / (216 * z6)

    cdf = K0 + K1 / np.sqrt(n) + K2 / n + K3 / n**1.5
    cdf = np.where(tiny, 0.0, cdf)
    out[sel] = 1.0 - cdf
    return np.clip(out, 0.0, 1.0)


def chi2_from_counts_many(
    ref_counts: np.ndarray, batch_counts: np.ndarray, offsets: np.ndarray
) -> np.ndarray:
    """Vectorized chi2_from_counts over concatenated per-feature count
    buffers (``offsets`` delimits features). One scipy chi2.sf call total;
    numerically identical to scipy.stats.chi2_contingency per feature
    (including the Yates continuity correction on 2x2 tables)."""
    nf = len(offsets) - 1
    stat = np.zeros(nf)
    dof = np.zeros(nf, dtype=np.int64)
    for j in range(nf):
        rc = np.asarray(ref_counts[offsets[j] : offsets[j + 1]], dtype=np.float64)
        bc = np.asarray(batch_counts[offsets[j] : offsets[j + 1]], dtype=np.float64)
        keep = (rc + bc) > 0
        rc, bc = rc[keep], bc[keep]
        k = len(rc)
        if k < 2 or rc.sum() == 0 or bc.sum() == 0:
            dof[j] = 0
            continue
        n = rc.sum() + bc.sum()
        exp_r = (rc + bc) * (rc.sum() / n)
        exp_b = (rc + bc) * (bc.sum() / n)
        dr = np.abs(rc - exp_r)
        db = np.abs(bc - exp_b)
        if k == 2:  # Yates continuity correction, as chi2_contingency applies
            dr = np.maximum(dr - 0.5, 0.0)
            db = np.maximum(db - 0.5, 0.0)
        stat[j] = (dr**2 / exp_r).sum() + (db**2 / exp_b).sum()
        dof[j] = k - 1
    pv = np.ones(nf)
    live = dof > 0
    if live.any():
        pv[live] = stats.chi2.sf(stat[live], dof[live])
    return pv


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
