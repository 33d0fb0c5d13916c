"""Synthetic data generation.

The reference trains on the UCI Credit Card Default dataset
(reference README.md:13; loaded via a Hive table at 01-train-model.ipynb
cell-7). That CSV is not available offline, so creditcore generates a
UCI-shaped synthetic dataset with the exact feature schema, the categorical
vocabularies observed in the reference's own fixtures
(databricks/data/inference.csv + app/sample-request.json), and a target that
is a noisy logistic function of the features so that training is
non-degenerate.
"""

from __future__ import annotations

import numpy as np
import pandas as pd

from .schema import CATEGORICAL_FEATURES, NUMERIC_FEATURES, TARGET

# Observed vocabularies (reference databricks/data/inference.csv,
# app/sample-request.json).
REPAYMENT_VOCAB = [
    "duly_paid",
    "no_delay",
    "delay_1_month",
    "delay_2_months",
    "delay_3_months",
]

VOCABULARIES: dict[str, list[str]] = {
    "sex": ["male", "female"],
    "education": ["university", "graduate_school", "high_school"],
    "marriage": ["married", "single"],
    **{f"repayment_status_{i}": list(REPAYMENT_VOCAB) for i in range(1, 7)},
}

# Risk weight per repayment category (later delays -> higher default risk).
_REPAY_RISK = {
    "duly_paid": -0.6,
    "no_delay": -0.2,
    "delay_1_month": 0.5,
    "delay_2_months": 1.1,
    "delay_3_months": 1.8,
}


def make_uci_shaped_frame(
    n_rows: int = 20_000,
    seed: int = 2024,
    missing_rate: float = 0.0,
    include_target: bool = True,
) -> pd.DataFrame:
    """Generate an UCI-credit-default-shaped dataframe.

    ``missing_rate`` > 0 injects NaNs into both categorical and numeric
    columns to exercise the imputation paths (constant "missing" for
    categoricals, median for numerics — reference 01-train cell-6).
    """
    rng = np.random.default_rng(seed)
    n = int(n_rows)
    cols: dict[str, np.ndarray] = {}

    for c in CATEGORICAL_FEATURES:
        vocab = VOCABULARIES[c]
        # Mildly non-uniform category frequencies.
        w = np.linspace(1.0, 0.45, num=len(vocab))
        p = w / w.sum()
        cols[c] = rng.choice(np.asarray(vocab, dtype=object), size=n, p=p)

    credit_limit = rng.gamma(shape=2.2, scale=60_000.0, size=n) + 10_000.0
    age = rng.integers(21, 75, size=n).astype(np.float64)
    cols["credit_limit"] = np.round(credit_limit, 2)
    cols["age"] = age
    for i in range(1, 7):
        bill = rng.gamma(shape=1.5, scale=0.25 * credit_limit / 1.5, size=n)
        pay = bill * rng.beta(2.0, 5.0, size=n)
        cols[f"bill_amount_{i}"] = np.round(bill, 2)
        cols[f"payment_amount_{i}"] = np.round(pay, 2)

    df = pd.DataFrame(cols, columns=CATEGORICAL_FEATURES + NUMERIC_FEATURES)

    if include_target:
        # Latent default score: repayment delays dominate, utilisation and
        # low credit limit contribute.
        z = np.zeros(n)
        for i in range(1, 7):
            z += np.vectorize(_REPAY_RISK.__getitem__)(df[f"repayment_status_{i}"]) / 6.0
        util = df[[f"bill_amount_{i}" for i in range(1, 7)]].to_numpy().sum(axis=1) / (
            6.0 * df["credit_limit"].to_numpy()
        )
        z += 1.2 * (util - util.mean()) / (util.std() + 1e-9)
        z += -0.3 * (np.log(df["credit_limit"]) - np.log(df["credit_limit"]).mean())
        z += rng.normal(0.0, 0.8, size=n)
        p_default = 1.0 / (1.0 + np.exp(-(z - 0.8)))
        df[TARGET] = (rng.uniform(size=n) < p_default).astype(np.int64)

    if missing_rate > 0.0:
        for c in CATEGORICAL_FEATURES:
            mask = rng.uniform(size=n) < missing_rate
            col = df[c].astype(object)
            # np.nan (not None): sklearn's SimpleImputer default
            # missing_values=np.nan detects nan floats in object arrays via
            # x != x, which None does not satisfy.
            col[mask] = np.nan
            df[c] = col
        for c in NUMERIC_FEATURES:
            mask = rng.uniform(size=n) < missing_rate
            col = df[c].to_numpy(dtype=np.float64, copy=True)
            col[mask] = np.nan
            df[c] = col

    return df


def make_request_batch(n_rows: int, seed: int = 7, drifted: bool = False) -> list[dict]:
    """Generate a list of request records (the POST /predict body shape)."""
    df = make_uci_shaped_frame(n_rows, seed=seed, include_target=False)
    if drifted:
        # Shift numerics and skew categoricals so drift tests can assert
        # detection fires.
        for c in NUMERIC_FEATURES:
            df[c] = df[c] * 2.5 + 1_000.0
        df["sex"] = "female"
    return df.to_dict(orient="records")


def make_dense_synthetic(
    n_rows: int, n_feats: int, seed: int = 0, dtype=np.float32
) -> tuple[np.ndarray, np.ndarray]:
    """Large dense synthetic matrix + binary labels (the 10M x 1k scale
    config of BASELINE.json config 5)."""
    rng = np.random.default_rng(seed)
    x = rng.standard_normal((n_rows, n_feats), dtype=np.float32).astype(dtype, copy=False)
    w = rng.standard_normal(n_feats).astype(np.float64)
    logits = x.astype(np.float64) @ (w / np.sqrt(n_feats))
    y = (logits + rng.normal(0, 0.5, size=n_rows) > 0).astype(np.int64)
    return x, y
