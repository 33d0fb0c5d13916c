"""Statistical-primitive tests: the drift math must match scipy (which is
what alibi-detect 0.12 uses underneath — reference 02-register cell-6)."""

from __future__ import annotations

import numpy as np
import pytest
from scipy import stats

from creditcore.models.drift import (
    TabularDriftDetector,
    chi2_from_counts,
    ks_2samp_d,
    ks_asymp_pvalue,
)


@pytest.mark.parametrize("n,m,seed", [(1000, 64, 0), (5000, 1024, 1), (777, 333, 2)])
def test_ks_d_matches_scipy(n, m, seed):
    rng = np.random.default_rng(seed)
    ref = np.sort(rng.normal(size=n))
    batch = rng.normal(0.3, 1.2, size=m)
    d = ks_2samp_d(ref, batch)
    sp = stats.ks_2samp(ref, batch, method="asymp")
    assert abs(d - sp.statistic) < 1e-12


@pytest.mark.parametrize("n,m,seed", [(1000, 64, 0), (5000, 1024, 1)])
def test_ks_pvalue_matches_scipy(n, m, seed):
    rng = np.random.default_rng(seed)
    ref = np.sort(rng.normal(size=n))
    batch = rng.normal(0.1, 1.0, size=m)
    d = ks_2samp_d(ref, batch)
    p = ks_asymp_pvalue(d, n, m)
    sp = stats.ks_2samp(ref, batch, method="asymp")
    assert abs(p - sp.pvalue) < 1e-10


def test_ks_with_ties():
    ref = np.sort(np.repeat([1.0, 2.0, 3.0], 100))
    batch = np.repeat([2.0, 3.0, 4.0], 10)
    d = ks_2samp_d(ref, batch)
    sp = stats.ks_2samp(ref, batch, method="asymp")
    assert abs(d - sp.statistic) < 1e-12


def test_pelz_good_matches_exact_kstwo():
    """The serving-path vectorized Pelz-Good p-value vs scipy's exact
    kstwo.sf across the drift-relevant range."""
    from creditcore.models.drift import _pelz_good_sf, ks_asymp_pvalue_many

    for en in (300, 974, 5000):
        ds = np.linspace(0.002, 0.9, 150)
        np.testing.assert_allclose(
            _pelz_good_sf(ds, en), stats.kstwo.sf(ds, en), atol=3e-7
        )
    # below the accuracy cutoff it must be scipy-exact
    ds = np.array([0.05, 0.2, 0.4])
    got = ks_asymp_pvalue_many(ds, 1000, 100)  # en = 91 < 300
    en = round(1000 * 100 / 1100)
    np.testing.assert_allclose(got, stats.kstwo.sf(ds, en), atol=1e-14)


def test_vectorized_pvals_match_scalar(packed):
    """pvals_from_stats (vectorized serving path) vs the scalar exact path."""
    from creditcore.data import make_request_batch
    from creditcore.ops import cpu_ref
    from creditcore.pack import encode_batch

    recs = make_request_batch(512, seed=42)
    codes, nums = encode_batch(recs, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    hist, ks_d = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
    fast = cpu_ref.pvals_from_stats(packed, hist, ks_d, 512)
    exact = cpu_ref.drift_pvals_cpu(packed, codes, nums_imp)
    np.testing.assert_allclose(fast, exact, atol=1e-6)


def test_chi2_matches_scipy():
    rc = np.array([50, 30, 20, 5])
    bc = np.array([10, 25, 3, 1])
    p = chi2_from_counts(rc, bc)
    sp = stats.chi2_contingency(np.stack([rc, bc]))
    assert abs(p - sp.pvalue) < 1e-12


def test_chi2_drops_empty_categories():
    rc = np.array([50, 0, 30])
    bc = np.array([10, 0, 20])
    p = chi2_from_counts(rc, bc)
    sp = stats.chi2_contingency(np.stack([[50, 30], [10, 20]]))
    assert abs(p - sp.pvalue) < 1e-12


def test_chi2_degenerate_single_category():
    assert chi2_from_counts(np.array([10]), np.array([5])) == 1.0


def test_detector_no_drift_on_same_distribution():
    rng = np.random.default_rng(0)
    cats = rng.choice(["a", "b", "c"], size=(2000, 2)).astype(object)
    nums = rng.normal(size=(2000, 3)).astype(object)
    x = np.concatenate([cats, nums], axis=1)
    det = TabularDriftDetector(x, p_val=0.05, categorical_idx=(0, 1))
    same = np.concatenate(
        [
            rng.choice(["a", "b", "c"], size=(500, 2)).astype(object),
            rng.normal(size=(500, 3)).astype(object),
        ],
        axis=1,
    )
    out = det.predict(same)
    assert out["data"]["is_drift"] == 0


def test_detector_flags_drift():
    rng = np.random.default_rng(0)
    cats = rng.choice(["a", "b", "c"], size=(2000, 1)).astype(object)
    nums = rng.normal(size=(2000, 2)).astype(object)
    det = TabularDriftDetector(
        np.concatenate([cats, nums], axis=1), p_val=0.05, categorical_idx=(0,)
    )
    shifted = np.concatenate(
        [
            np.full((500, 1), "a", dtype=object),
            (rng.normal(size=(500, 2)) + 3.0).astype(object),
        ],
        axis=1,
    )
    out = det.predict(shifted)
    assert out["data"]["is_drift"] == 1
    assert (out["data"]["p_val"][1:] < 1e-6).all()


def test_detector_unseen_category_binned():
    rng = np.random.default_rng(1)
    cats = rng.choice(["a", "b"], size=(1000, 1)).astype(object)
    det = TabularDriftDetector(cats, p_val=0.05, categorical_idx=(0,))
    batch = np.full((100, 1), "zzz_new", dtype=object)
    out = det.predict(batch)
    assert out["data"]["p_val"][0] < 1e-6  # wholly new category = maximal drift
