"""Classifier pipeline — training-side model definition.

Reproduces the reference's sklearn pipeline exactly
(reference databricks/src/01-train-model.ipynb cell-6):

- categorical: SimpleImputer(constant, "missing") -> OneHotEncoder(ignore)
- numeric:     SimpleImputer(median)
- estimator:   RandomForestClassifier(n_estimators, max_depth, criterion,
               n_jobs=-1)

Training happens on CPU with sklearn (like the reference); *scoring* happens
on MI355X via the packed flat-buffer representation (creditcore.pack) and the
HIP forest-traversal kernel (csrc/kernels/score.hip). This module is the
golden CPU path the GPU path is tested against.
"""

from __future__ import annotations

from sklearn.compose import ColumnTransformer
from sklearn.ensemble import RandomForestClassifier
from sklearn.impute import SimpleImputer
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import OneHotEncoder

from ..schema import CATEGORICAL_FEATURES, MISSING_CATEGORY, NUMERIC_FEATURES


def make_classifier_pipeline(params: dict) -> Pipeline:
    """Build the classifier pipeline (reference 01-train cell-6).

    ``params`` are RandomForestClassifier kwargs
    (n_estimators / max_depth / criterion / random_state ...).
    """
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
            ("classifier", RandomForestClassifier(**params, n_jobs=-1)),
        ]
    )
