"""Classifier pipeline — training-side model definition.

Reproduces the reference's sklearn pipeline exactly
(reference databricks/src/01-train-model.ipynb cell-6):

- categorical: SimpleImputer(constant, "missing") -> OneHotEncoder(ignore)
- numeric:     SimpleImputer(median)
- estimator:   RandomForestClassifier(n_estimators, max_depth, criterion,
               n_jobs=-1)

Training happens on CPU with sklearn (like the reference); *scoring* happens
on MI355X via the packed flat-buffer representation (creditcore.pack) and the
HIP forest-traversal kernel (csrc/creditcore_kernels.hip (forest_kernel_ilpN)). This module is the
golden CPU path the GPU path is tested against.
"""

from __future__ import annotations

from sklearn.compose import ColumnTransformer
from sklearn.ensemble import RandomForestClassifier
from sklearn.impute import SimpleImputer
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import OneHotEncoder

from ..schema import CATEGORICAL_FEATURES, MISSING_CATEGORY, NUMERIC_FEATURES


def make_classifier_pipeline(params: dict, algorithm: str = "rf") -> Pipeline:
    """Build the classifier pipeline (reference 01-train cell-6).

    ``params`` are estimator kwargs (n_estimators / max_depth / random_state
    ...). ``algorithm``: "rf" (the reference's RandomForest), "gbt"
    (GradientBoostingClassifier — the north star's "gradient-boosted-tree
    traversal" family; same packed node-SoA format, scored by the same HIP
    kernel with a sigmoid(sum + prior) finalize), or "et"
    (ExtraTreesClassifier — identical tree structure and leaf-fraction-mean
    semantics to RF, so it rides the RF pack/score path unchanged).
    """
    if algorithm == "gbt":
        from sklearn.ensemble import GradientBoostingClassifier

        params = dict(params)
        params.pop("criterion", None)  # rf-only knob from the search space
        params.setdefault("n_estimators", 200)
        estimator = GradientBoostingClassifier(**params)
    elif algorithm == "et":
        from sklearn.ensemble import ExtraTreesClassifier

        estimator = ExtraTreesClassifier(**params, n_jobs=-1)
    elif algorithm == "rf":
        estimator = RandomForestClassifier(**params, n_jobs=-1)
    else:
        raise ValueError(f"unknown algorithm: {algorithm}")

    categorical_transformer = Pipeline(
        steps=[
            (
                "imputer",
                SimpleImputer(strategy="constant", fill_value=MISSING_CATEGORY),
            ),
            ("ohe", OneHotEncoder(handle_unknown="ignore")),
        ]
    )
    numeric_transformer = Pipeline(steps=[("imputer", SimpleImputer(strategy="median"))])
    preprocessor = ColumnTransformer(
        transformers=[
            ("categorical", categorical_transformer, CATEGORICAL_FEATURES),
            ("numeric", numeric_transformer, NUMERIC_FEATURES),
        ]
    )
    return Pipeline(
        [
            ("preprocessor", preprocessor),
            ("classifier", estimator),
        ]
    )
