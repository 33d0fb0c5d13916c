"""ExtraTrees family: sklearn ExtraTreesClassifier has the same tree_
structure and leaf-fraction-mean predict_proba as the reference's
RandomForest, so it rides the RF pack/score path (cls_kind 0) unchanged —
this suite proves that end to end."""

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
def et_packed(train_df):
    pipe = make_classifier_pipeline(
        {"n_estimators": 80, "max_depth": 8, "random_state": 0}, algorithm="et"
    )
    pipe.fit(train_df[FEATURES], train_df[TARGET].values.ravel())
    drift, outlier = fit_detectors(train_df)
    c = pack_classifier_pipeline(pipe)
    o = pack_isolation_forest(outlier)
    d = pack_drift(drift, c["vocabs"])
    return PackedModel(**c, **o, **d), pipe


def test_et_packs_as_rf_kind(et_packed):
    packed, _ = et_packed
    assert packed.cls_kind == 0  # leaf-fraction mean, same finalize as RF


def test_et_parity_vs_sklearn(et_packed, score_batch):
    packed, pipe = et_packed
    codes, nums = encode_batch(score_batch, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ours = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    sk = pipe.predict_proba(score_batch[FEATURES])[:, 1]
    np.testing.assert_allclose(ours, sk, atol=1e-7)


@pytest.mark.gpu
def test_et_gpu_parity(et_packed, score_batch):
    from creditcore.engine import ScoringEngine

    packed, _ = et_packed
    eng = ScoringEngine(packed, device="cuda")
    codes, nums = encode_batch(score_batch, packed.vocabs)
    out = eng.score_arrays(codes, nums, with_drift=False)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ref = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    np.testing.assert_allclose(out["predictions"], ref, atol=1e-9)
