"""Gradient-boosted-tree family: pack + scorer parity vs sklearn
(the north star's "gradient-boosted-tree traversal" path — same node-SoA
format and HIP kernel as the reference's RandomForest, different finalize:
sigmoid(sum of lr-scaled leaves + prior))."""

from __future__ import annotations

import numpy as np
import pytest

from creditcore.models.forest import make_classifier_pipeline
from creditcore.ops import cpu_ref
from creditcore.pack import (
    PackedModel,
    encode_batch,
    pack_classifier_pipeline,
    pack_drift,
    pack_isolation_forest,
)
from creditcore.schema import FEATURES, TARGET
from creditcore.train import fit_detectors


@pytest.fixture(scope="module")
def gbt_packed(train_df):
    pipe = make_classifier_pipeline(
        {"n_estimators": 60, "max_depth": 3, "random_state": 0}, algorithm="gbt"
    )
    pipe.fit(train_df[FEATURES], train_df[TARGET].values.ravel())
    drift, outlier = fit_detectors(train_df)
    c = pack_classifier_pipeline(pipe)
    o = pack_isolation_forest(outlier)
    d = pack_drift(drift, c["vocabs"])
    return PackedModel(**c, **o, **d), pipe


def test_gbt_pack_kind(gbt_packed):
    packed, _ = gbt_packed
    assert packed.cls_kind == 1
    assert np.isfinite(packed.cls_bias)


def test_gbt_parity_vs_sklearn(gbt_packed, score_batch):
    packed, pipe = gbt_packed
    codes, nums = encode_batch(score_batch, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ours = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    sk = pipe.predict_proba(score_batch[FEATURES])[:, 1]
    np.testing.assert_allclose(ours, sk, atol=1e-7)


def test_gbt_save_load(gbt_packed, tmp_path, score_batch):
    packed, _ = gbt_packed
    p = str(tmp_path / "gbt.npz")
    packed.save(p)
    re = PackedModel.load(p)
    assert re.cls_kind == 1 and re.cls_bias == packed.cls_bias
    codes, nums = encode_batch(score_batch.head(64), packed.vocabs)
    a = cpu_ref.score_batch_cpu(packed, codes, nums)
    b = cpu_ref.score_batch_cpu(re, codes, nums)
    np.testing.assert_array_equal(a["predictions"], b["predictions"])


@pytest.mark.gpu
def test_gbt_gpu_parity(gbt_packed, score_batch):
    from creditcore.engine import ScoringEngine

    packed, _ = gbt_packed
    eng = ScoringEngine(packed, device="cuda")
    codes, nums = encode_batch(score_batch, packed.vocabs)
    out = eng.score_arrays(codes, nums, with_drift=False)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ref = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    np.testing.assert_allclose(out["predictions"], ref, atol=1e-9)


def test_gbt_serves_through_api(train_df, tmp_path):
    """A GBT-trained pyfunc dir serves through the same API stack."""
    from fastapi.testclient import TestClient

    from creditcore import registry
    from creditcore.config import ServeConfig
    from creditcore.schema import SAMPLE_REQUEST
    from creditcore.serve import create_app
    from creditcore.train import fit_detectors

    pipe = make_classifier_pipeline(
        {"n_estimators": 40, "max_depth": 3, "random_state": 1}, algorithm="gbt"
    )
    pipe.fit(train_df[FEATURES].head(1500), train_df[TARGET].head(1500).values.ravel())
    drift, outlier = fit_detectors(train_df.head(1500))
    model_dir = str(tmp_path / "gbt_model")
    registry.save_pyfunc_model(model_dir, pipe, drift, outlier)

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    with TestClient(create_app(cfg)) as client:
        r = client.post("/predict", json=SAMPLE_REQUEST)
        assert r.status_code == 200
        assert 0.0 <= r.json()["predictions"][0] <= 1.0
