"""Local trainer — replaces the Databricks training + registration job.

Reference behavior being reproduced (databricks/src/01-train-model.ipynb,
02-register-model.ipynb, orchestrated by
databricks/resources/train_register_model.yml):

1. hyperparameter search over {n_estimators in [100,1000),
   max_depth in [1,25), criterion in {gini, entropy}} with 10 evaluations
   (reference: hyperopt TPE, cell-8). hyperopt is not available offline, so
   the search here is a seeded random search with a TPE-style exploit phase:
   after ``n_startup`` random draws, later draws perturb the incumbent best.
2. per-eval metrics: accuracy, roc_auc, f1, precision, recall on an 80/20
   split with random_state=2024 (cell-7); runs recorded under ``runs/``
   (MLflow tracking replacement).
3. best run selected by validation_roc_auc_score (cell-10).
4. drift (TabularDrift p_val=0.05) + outlier (IForest threshold=0.95)
   detectors fitted on the training feature matrix (02-register cell-6).
5. the 3-model composite packaged in the MLflow pyfunc layout and registered
   (02-register cell-12/13); returns the model URI (cell-15).
"""

from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import dataclass, field

import numpy as np
import pandas as pd
from sklearn.metrics import (
    accuracy_score,
    f1_score,
    precision_score,
    recall_score,
    roc_auc_score,
)
from sklearn.model_selection import train_test_split

from . import registry
from .data import make_uci_shaped_frame
from .models.drift import TabularDriftDetector
from .models.forest import make_classifier_pipeline
from .models.iforest import IForestDetector
from .schema import CATEGORICAL_FEATURES, FEATURES, INPUT_SAMPLE, NUMERIC_FEATURES, TARGET

DEFAULT_MODEL_NAME = "credit-default-uci-custom"  # reference 02-register cell-13


class QualityGateError(RuntimeError):
    """Raised when the best run fails the registration quality gate."""


@dataclass
class EvalRun:
    run_id: str
    params: dict
    metrics: dict
    pipeline: object = field(repr=False, default=None)


def _evaluate(
    params: dict, df: pd.DataFrame, seed: int = 2024, algorithm: str = "rf"
) -> EvalRun:
    """One hyperparameter evaluation (reference cell-7, minus the dead
    double-fits)."""
    df_train, df_test = train_test_split(
        df[FEATURES + [TARGET]], test_size=0.20, random_state=seed
    )
    x_train, y_train = df_train[FEATURES], df_train[TARGET]
    x_test, y_test = df_test[FEATURES], df_test[TARGET]

    estimator = make_classifier_pipeline({**params, "random_state": seed}, algorithm)
    estimator.fit(x_train, y_train.values.ravel())
    y_pred = estimator.predict(x_test)

    metrics = {
        "validation_accuracy_score": accuracy_score(y_test, y_pred),
        # Reference quirk kept: roc_auc computed on hard labels (cell-7).
        "validation_roc_auc_score": roc_auc_score(y_test, y_pred),
        "validation_f1_score": f1_score(y_test, y_pred),
        "validation_precision_score": precision_score(y_test, y_pred, zero_division=0),
        "validation_recall_score": recall_score(y_test, y_pred),
    }
    return EvalRun(uuid.uuid4().hex, dict(params), metrics, estimator)


def train_model(
    df: pd.DataFrame | None = None,
    max_evals: int = 10,
    n_startup: int = 5,
    seed: int = 2024,
    runs_dir: str | None = None,
    n_rows: int = 20_000,
    algorithm: str = "rf",
) -> EvalRun:
    """Hyperparameter search with a real TPE sampler (reference cell-8:
    hyperopt fmin(tpe.suggest, max_evals=10); hyperopt itself is
    unavailable offline — creditcore.models.tpe implements the algorithm).
    Returns the best run by roc_auc (cell-10)."""
    from .models.tpe import TPESampler, reference_space

    if df is None:
        df = make_uci_shaped_frame(n_rows=n_rows, seed=seed)
    sampler = TPESampler(reference_space(), seed=seed, n_startup=n_startup)
    best: EvalRun | None = None
    t0 = time.time()
    for i in range(max_evals):
        params = sampler.suggest()
        run = _evaluate(params, df, seed=seed, algorithm=algorithm)
        # hyperopt minimizes; the notebook returns -roc_auc as the loss
        sampler.observe(params, -run.metrics["validation_roc_auc_score"])
        if runs_dir:
            rd = os.path.join(runs_dir, run.run_id)
            os.makedirs(rd, exist_ok=True)
            with open(os.path.join(rd, "run.json"), "w") as f:
                json.dump({"params": run.params, "metrics": run.metrics}, f, indent=2)
        if best is None or (
            run.metrics["validation_roc_auc_score"]
            > best.metrics["validation_roc_auc_score"]
        ):
            best = run
        print(
            f"[train] eval {i + 1}/{max_evals} params={params} "
            f"roc_auc={run.metrics['validation_roc_auc_score']:.4f} "
            f"({time.time() - t0:.1f}s)",
            flush=True,
        )
    assert best is not None
    return best


def fit_detectors(df: pd.DataFrame) -> tuple[TabularDriftDetector, IForestDetector]:
    """Fit drift + outlier detectors on the training feature matrix
    (reference 02-register cell-6)."""
    drift = TabularDriftDetector(
        df[FEATURES].values,
        p_val=0.05,
        categorical_idx=tuple(range(len(CATEGORICAL_FEATURES))),
    )
    outlier = IForestDetector(threshold=0.95)
    outlier.fit(df[NUMERIC_FEATURES].values.astype(np.float64))
    return drift, outlier


def train_and_register(
    model_dir: str = "./model",
    model_name: str = DEFAULT_MODEL_NAME,
    registry_root: str | None = None,
    max_evals: int = 10,
    n_rows: int = 20_000,
    seed: int = 2024,
    df: pd.DataFrame | None = None,
    register: bool = True,
    algorithm: str = "rf",
    min_roc_auc: float | None = None,
) -> str:
    """The full train -> package -> register job (the reference's 2-task
    Databricks DAG, train_register_model.yml:10-39). Returns the model URI
    (or the model dir when register=False)."""
    if df is None:
        df = make_uci_shaped_frame(n_rows=n_rows, seed=seed)
    best = train_model(df=df, max_evals=max_evals, seed=seed, algorithm=algorithm)
    if min_roc_auc is not None:
        auc = best.metrics["validation_roc_auc_score"]
        if auc < min_roc_auc:
            # the quality gate the reference lacks (metrics
            # were logged but never enforced before registration/deploy)
            raise QualityGateError(
                f"best validation_roc_auc_score {auc:.4f} < gate {min_roc_auc}"
            )
    drift, outlier = fit_detectors(df)
    registry.save_pyfunc_model(
        model_dir,
        best.pipeline,
        drift,
        outlier,
        input_example=INPUT_SAMPLE,
        run_id=best.run_id,
        extra_metadata={"best_params": best.params, "best_metrics": best.metrics},
    )
    if not register:
        return model_dir
    uri = registry.register_model(
        model_dir,
        model_name,
        registry_root=registry_root or registry.DEFAULT_REGISTRY_ROOT,
        tags={"best_classifier_model_run_id": best.run_id},
    )
    print(f"[train] registered {uri}", flush=True)
    return uri
