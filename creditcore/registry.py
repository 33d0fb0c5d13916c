"""MLflow-pyfunc-compatible checkpoint format + local model registry.

The reference's only durable model artifact is an MLflow pyfunc directory
(written at 02-register-model.ipynb cell-12, loaded by app/main.py:26-28):

    model/
    ├─ MLmodel                       # pyfunc flavor descriptor
    ├─ conda.yaml / requirements.txt
    ├─ python_model.pkl              # cloudpickled CustomModel instance
    ├─ input_example.json
    └─ artifacts/
       ├─ classifier/model/model.pkl # sklearn Pipeline (joblib)
       ├─ drift.pkl                  # TabularDrift detector (incl. reference data)
       └─ outlier.pkl                # IForest detector

creditcore reads and writes this exact on-disk layout (mlflow itself is not a
runtime dependency — the loader here implements the subset of
``mlflow.pyfunc.load_model`` the serving path needs). The MLflow Model
Registry (02-register cell-13) is replaced by a local directory registry with
``models:/<name>/<version>`` URIs.
"""

from __future__ import annotations

import json
import os
import uuid
from dataclasses import dataclass
from datetime import datetime, timezone

import cloudpickle
import joblib
import pandas as pd
import yaml

from .models.drift import TabularDriftDetector
from .models.iforest import IForestDetector
from .schema import CATEGORICAL_FEATURES, NUMERIC_FEATURES

PYFUNC_ARTIFACTS_SUBDIR = "artifacts"
CLASSIFIER_PKL = "classifier/model/model.pkl"  # nesting per 02-register cell-9
DRIFT_PKL = "drift.pkl"
OUTLIER_PKL = "outlier.pkl"


class CustomModel:
    """The packaged 3-model composite (reference 02-register cell-9).

    Mirrors the reference CustomModel(mlflow.pyfunc.PythonModel): classifier
    probabilities + outlier flags + per-feature batch drift (1 - p_val).
    Implemented without an mlflow base class so it unpickles with only
    creditcore installed; the ``context`` argument accepts any object with an
    ``artifacts["artifacts_path"]`` mapping (mlflow-compatible call shape).
    """

    def __init__(self, categorical_feature_names: list[str], numeric_feature_names: list[str]):
        self.categorical_features = list(categorical_feature_names)
        self.numeric_features = list(numeric_feature_names)
        self.all_features = self.categorical_features + self.numeric_features
        self.classifier = None
        self.drift = None
        self.outliers = None

    def load_context(self, context) -> None:
        base = context.artifacts["artifacts_path"]
        self.classifier = joblib.load(os.path.join(base, CLASSIFIER_PKL))
        self.drift = joblib.load(os.path.join(base, DRIFT_PKL))
        self.outliers = joblib.load(os.path.join(base, OUTLIER_PKL))

    def predict(self, context, model_input) -> dict:
        df = pd.DataFrame(model_input)
        predictions = self.classifier.predict_proba(df[self.all_features])[:, 1].tolist()
        drift_results = self.drift.predict(df[self.all_features].values)
        outlier_results = self.outliers.predict(df[self.numeric_features].values)
        return {
            "predictions": predictions,
            "outliers": outlier_results["data"]["is_outlier"].tolist(),
            "feature_drift_batch": dict(
                zip(
                    self.all_features,
                    (1 - drift_results["data"]["p_val"]).tolist(),
                )
            ),
        }


@dataclass
class _Context:
    artifacts: dict


class LoadedPyfuncModel:
    """What ``load_pyfunc_model`` returns — mlflow-pyfunc-shaped
    (``.predict(df)``), plus access to the raw artifacts for GPU packing."""

    def __init__(self, model_dir: str, python_model: CustomModel):
        self.model_dir = model_dir
        self._model = python_model

    @property
    def artifacts_path(self) -> str:
        return os.path.join(self.model_dir, PYFUNC_ARTIFACTS_SUBDIR)

    @property
    def python_model(self) -> CustomModel:
        return self._model

    def predict(self, model_input) -> dict:
        return self._model.predict(None, model_input)


def save_pyfunc_model(
    model_dir: str,
    classifier_pipeline,
    drift: TabularDriftDetector,
    outlier: IForestDetector,
    input_example: list[dict] | None = None,
    run_id: str | None = None,
    extra_metadata: dict | None = None,
) -> str:
    """Write the MLflow pyfunc directory layout (02-register cell-12)."""
    os.makedirs(model_dir, exist_ok=True)
    art = os.path.join(model_dir, PYFUNC_ARTIFACTS_SUBDIR)
    os.makedirs(os.path.join(art, os.path.dirname(CLASSIFIER_PKL)), exist_ok=True)

    joblib.dump(classifier_pipeline, os.path.join(art, CLASSIFIER_PKL))
    joblib.dump(drift, os.path.join(art, DRIFT_PKL))
    joblib.dump(outlier, os.path.join(art, OUTLIER_PKL))

    model = CustomModel(CATEGORICAL_FEATURES, NUMERIC_FEATURES)
    with open(os.path.join(model_dir, "python_model.pkl"), "wb") as f:
        cloudpickle.dump(model, f)

    if input_example is not None:
        with open(os.path.join(model_dir, "input_example.json"), "w") as f:
            json.dump({"data": input_example}, f, indent=2)

    run_id = run_id or uuid.uuid4().hex
    mlmodel = {
        "artifact_path": "model",
        "flavors": {
            "python_function": {
                "artifacts": {
                    "artifacts_path": {
                        "path": PYFUNC_ARTIFACTS_SUBDIR,
                        "uri": PYFUNC_ARTIFACTS_SUBDIR,
                    }
                },
                "cloudpickle_version": cloudpickle.__version__,
                "env": {"conda": "conda.yaml", "virtualenv": "python_env.yaml"},
                "loader_module": "mlflow.pyfunc.model",
                "python_model": "python_model.pkl",
                "python_version": "3.10",
            }
        },
        "model_uuid": uuid.uuid4().hex,
        "run_id": run_id,
        "utc_time_created": datetime.now(timezone.utc).isoformat(),
        "mlflow_version": "2.10.0",  # layout-compatible version (reference pin)
    }
    if extra_metadata:
        mlmodel["metadata"] = extra_metadata
    with open(os.path.join(model_dir, "MLmodel"), "w") as f:
        yaml.safe_dump(mlmodel, f, sort_keys=False)

    conda_env = {
        "name": "creditcore-serving",
        "channels": ["conda-forge"],
        "dependencies": [
            "python=3.10",
            "pip",
            {
                "pip": [
                    "numpy",
                    "pandas",
                    "scikit-learn",
                    "scipy",
                    "joblib",
                    f"cloudpickle=={cloudpickle.__version__}",
                    "creditcore",
                ]
            },
        ],
    }
    with open(os.path.join(model_dir, "conda.yaml"), "w") as f:
        yaml.safe_dump(conda_env, f, sort_keys=False)

    # the rest of mlflow's standard file set (the MLmodel env block above
    # references python_env.yaml; real mlflow writes all three)
    import sys

    pyver = ".".join(map(str, sys.version_info[:3]))
    reqs = [
        "numpy",
        "pandas",
        "scikit-learn",
        "scipy",
        "joblib",
        f"cloudpickle=={cloudpickle.__version__}",
        "creditcore",
    ]
    with open(os.path.join(model_dir, "python_env.yaml"), "w") as f:
        yaml.safe_dump(
            {
                "python": pyver,
                "build_dependencies": ["pip", "setuptools", "wheel"],
                "dependencies": ["-r requirements.txt"],
            },
            f,
            sort_keys=False,
        )
    with open(os.path.join(model_dir, "requirements.txt"), "w") as f:
        f.write("\n".join(reqs) + "\n")

    return model_dir


def load_pyfunc_model(model_dir: str) -> LoadedPyfuncModel:
    """Load a pyfunc model directory (mlflow.pyfunc.load_model equivalent for
    this flavor — reference app/main.py:26-28)."""
    mlmodel_path = os.path.join(model_dir, "MLmodel")
    if not os.path.isfile(mlmodel_path):
        raise FileNotFoundError(f"not a pyfunc model dir (no MLmodel): {model_dir}")
    with open(mlmodel_path) as f:
        mlmodel = yaml.safe_load(f)
    flavor = mlmodel["flavors"]["python_function"]
    with open(os.path.join(model_dir, flavor["python_model"]), "rb") as f:
        model: CustomModel = cloudpickle.load(f)
    art_rel = flavor.get("artifacts", {}).get("artifacts_path", {}).get("path", PYFUNC_ARTIFACTS_SUBDIR)
    ctx = _Context(artifacts={"artifacts_path": os.path.join(model_dir, art_rel)})
    model.load_context(ctx)
    return LoadedPyfuncModel(model_dir, model)


# ---------------------------------------------------------------------------
# Local model registry (replaces the MLflow Model Registry, 02-register
# cell-13: registered name + version + tags, returns models:/<name>/<version>)
# ---------------------------------------------------------------------------

DEFAULT_REGISTRY_ROOT = os.environ.get("CREDITCORE_REGISTRY", "./registry")


def register_model(
    model_dir: str,
    name: str,
    registry_root: str | None = None,
    tags: dict | None = None,
) -> str:
    """Copy/record a pyfunc model dir into the local registry; returns the
    model URI ``models:/<name>/<version>`` (reference cell-13/15)."""
    import shutil

    registry_root = registry_root or DEFAULT_REGISTRY_ROOT
    base = os.path.join(registry_root, name)
    os.makedirs(base, exist_ok=True)
    versions = [int(v) for v in os.listdir(base) if v.isdigit()]
    version = max(versions, default=0) + 1
    dst = os.path.join(base, str(version))
    shutil.copytree(model_dir, dst)
    meta = {
        "name": name,
        "version": version,
        "tags": tags or {},
        "source": os.path.abspath(model_dir),
        "creation_time": datetime.now(timezone.utc).isoformat(),
    }
    with open(os.path.join(dst, "registered_model_meta.yaml"), "w") as f:
        yaml.safe_dump(meta, f, sort_keys=False)
    return f"models:/{name}/{version}"


def resolve_model_uri(uri: str, registry_root: str | None = None) -> str:
    """Resolve ``models:/<name>/<version|latest>`` or a plain path to a model
    directory."""
    if not uri.startswith("models:/"):
        return uri
    registry_root = registry_root or DEFAULT_REGISTRY_ROOT
    rest = uri[len("models:/") :]
    name, _, version = rest.partition("/")
    base = os.path.join(registry_root, name)
    if not version or version == "latest":
        versions = [int(v) for v in os.listdir(base) if v.isdigit()]
        if not versions:
            raise FileNotFoundError(f"no versions registered under {base}")
        version = str(max(versions))
    return os.path.join(base, version)
