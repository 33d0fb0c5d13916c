"""Golden-parity tests: the packed flat-buffer scorer (CPU reference, which
the HIP kernels are in turn tested against in test_gpu.py) must reproduce the
sklearn/detector outputs of the packaged pyfunc model — the reference's
serving numerics (02-register-model.ipynb cell-9)."""

from __future__ import annotations

import numpy as np
import pytest

from creditcore.ops import cpu_ref
from creditcore.pack import encode_batch
from creditcore.schema import CATEGORICAL_FEATURES, FEATURES, NUMERIC_FEATURES


@pytest.fixture(scope="module")
def encoded(packed, score_batch):
    codes, nums = encode_batch(score_batch, packed.vocabs)
    return codes, nums


def test_encode_semantics(packed, score_batch):
    codes, nums = encode_batch(score_batch, packed.vocabs)
    assert codes.shape == (len(score_batch), 9) and codes.dtype == np.int16
    assert nums.shape == (len(score_batch), 14) and nums.dtype == np.float32
    # unknown categories -> -1 (OneHotEncoder handle_unknown="ignore")
    j = CATEGORICAL_FEATURES.index("education")
    assert codes[2, j] == -1
    # NaN numerics pass through for in-kernel median imputation
    assert np.isnan(nums[0, NUMERIC_FEATURES.index("credit_limit")])


def test_classifier_parity(packed, loaded_pyfunc, encoded, score_batch):
    codes, nums = encoded
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ours = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    sk = loaded_pyfunc.python_model.classifier.predict_proba(score_batch[FEATURES])[:, 1]
    np.testing.assert_allclose(ours, sk, atol=1e-7)


def test_iforest_parity(packed, loaded_pyfunc, encoded, score_batch):
    codes, nums = encoded
    nums_imp = cpu_ref.impute_nums(packed, nums)
    iscore, flags = cpu_ref.score_iforest_cpu(packed, nums_imp)
    # detector scored on the same (median-imputed) numerics
    ref = loaded_pyfunc.python_model.outliers.predict(nums_imp.astype(np.float64))
    # packed leaf path-lengths are f32; sklearn computes in f64
    np.testing.assert_allclose(iscore, ref["data"]["instance_score"], atol=1e-6, rtol=1e-5)
    np.testing.assert_array_equal(flags, ref["data"]["is_outlier"].astype(np.float64))


def test_drift_parity(packed, train_df, encoded, score_batch):
    """Packed drift p-values vs a TabularDriftDetector at matched precision.

    The packed path stores the reference sorted values as f32 and scores f32
    request batches, so exact reference↔batch ties (the common no-drift case:
    batch rows drawn from the training distribution) are honored at f32.
    The comparison detector is therefore fitted on the f32-rounded reference
    matrix — mixed f32/f64 would silently break those ties.

    Uses a clean batch (no unknown categories): the detector bins unseen
    categories as new contingency columns while the packed path uses the
    OHE "unseen" bin — equivalent only when both are empty or the vocab
    matches.
    """
    from creditcore.models.drift import TabularDriftDetector
    from creditcore.schema import FEATURES

    clean = score_batch.iloc[8:].reset_index(drop=True)
    codes, nums = encode_batch(clean, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    pvals = cpu_ref.drift_pvals_cpu(packed, codes, nums_imp)

    ref_mat = np.concatenate(
        [
            train_df[CATEGORICAL_FEATURES].to_numpy(dtype=object),
            train_df[NUMERIC_FEATURES]
            .to_numpy(dtype=np.float64)
            .astype(np.float32)
            .astype(object),
        ],
        axis=1,
    )
    det32 = TabularDriftDetector(ref_mat, p_val=0.05, categorical_idx=tuple(range(9)))
    ref = det32.predict(
        np.concatenate(
            [
                clean[CATEGORICAL_FEATURES].to_numpy(dtype=object),
                nums_imp.astype(object),
            ],
            axis=1,
        )
    )
    np.testing.assert_allclose(pvals, ref["data"]["p_val"].astype(np.float64), atol=1e-6)


def test_full_pipeline_vs_pyfunc_predict(packed, loaded_pyfunc, score_batch):
    """End-to-end: packed pipeline vs CustomModel.predict response dict."""
    clean = score_batch.iloc[8:72].reset_index(drop=True)
    codes, nums = encode_batch(clean, packed.vocabs)
    out = cpu_ref.score_batch_cpu(packed, codes, nums)
    ref = loaded_pyfunc.predict(clean)
    np.testing.assert_allclose(out["predictions"], ref["predictions"], atol=1e-7)
    np.testing.assert_array_equal(out["outliers"], np.asarray(ref["outliers"], dtype=np.float64))
    ours_drift = {f: 1.0 - p for f, p in zip(FEATURES, out["p_vals"])}
    for f in FEATURES:
        assert abs(ours_drift[f] - ref["feature_drift_batch"][f]) < 1e-5


def test_packed_save_load_roundtrip(packed, tmp_path, encoded):
    from creditcore.pack import PackedModel

    p = str(tmp_path / "packed.npz")
    packed.save(p)
    re = PackedModel.load(p)
    codes, nums = encoded
    a = cpu_ref.score_batch_cpu(packed, codes, nums)
    b = cpu_ref.score_batch_cpu(re, codes, nums)
    np.testing.assert_array_equal(a["predictions"], b["predictions"])
    np.testing.assert_array_equal(a["p_vals"], b["p_vals"])
    assert re.vocabs == packed.vocabs


def test_tree_threshold_f32_rounding(packed):
    """The packed f32 thresholds must preserve the f64 decision boundary for
    every f32 input (pack.py ceil-to-next-float32 rule)."""
    bits = packed.cls_nodes[:, 1].view(np.float32)
    internal = packed.cls_nodes[:, 0] >= 0
    thr32 = bits[internal]
    assert np.isfinite(thr32).all()


def test_linear_scorer_matches_sklearn(train_df, packed, loaded_pyfunc):
    """LinearScorer (the small linear family utility) vs sklearn logistic
    regression on the encoded matrix."""
    from creditcore.models.linear import LinearScorer
    from creditcore.schema import FEATURES, TARGET

    pre = loaded_pyfunc.python_model.classifier.named_steps["preprocessor"]
    x = pre.transform(train_df[FEATURES].head(2000))
    if hasattr(x, "toarray"):
        x = x.toarray()
    y = train_df[TARGET].head(2000).to_numpy()
    sc = LinearScorer.fit(x, y, max_iter=50)
    p = sc.predict_proba1(x)
    assert p.shape == (2000,)
    assert 0.55 < ((p > 0.5) == y).mean()  # fits the training signal
