"""Request/response schema — the wire contract of the credit-default API.

Mirrors the reference contract exactly (reference: app/model.py:8-71 and the
response assembled at databricks/src/02-register-model.ipynb cell-9):

- 23 input features: 9 categorical strings + 14 numeric floats;
- response = per-row default probability, per-row outlier flag, and a
  per-feature batch-level drift score (1 - p_val).

The reference's ``age`` default of 18000.0 (a copy-paste of ``credit_limit``,
reference app/model.py:22) is reproduced because it is part of the observable
wire contract (sample-request.json uses it).
"""

from __future__ import annotations

from pydantic import BaseModel, ConfigDict

# Feature sets (reference: 01-train-model.ipynb cell-4, 02-register cell-4).
CATEGORICAL_FEATURES: list[str] = [
    "sex",
    "education",
    "marriage",
    "repayment_status_1",
    "repayment_status_2",
    "repayment_status_3",
    "repayment_status_4",
    "repayment_status_5",
    "repayment_status_6",
]

NUMERIC_FEATURES: list[str] = [
    "credit_limit",
    "age",
    "bill_amount_1",
    "bill_amount_2",
    "bill_amount_3",
    "bill_amount_4",
    "bill_amount_5",
    "bill_amount_6",
    "payment_amount_1",
    "payment_amount_2",
    "payment_amount_3",
    "payment_amount_4",
    "payment_amount_5",
    "payment_amount_6",
]

FEATURES: list[str] = CATEGORICAL_FEATURES + NUMERIC_FEATURES

TARGET = "default_payment_next_month"

# The imputation category used for missing categorical values
# (reference: SimpleImputer(strategy="constant", fill_value="missing"),
# 01-train-model.ipynb cell-6).
MISSING_CATEGORY = "missing"


class LoanApplicant(BaseModel):
    """One scoring record (reference app/model.py:8-34, defaults included)."""

    sex: str = "male"
    education: str = "university"
    marriage: str = "married"
    repayment_status_1: str = "duly_paid"
    repayment_status_2: str = "duly_paid"
    repayment_status_3: str = "duly_paid"
    repayment_status_4: str = "duly_paid"
    repayment_status_5: str = "no_delay"
    repayment_status_6: str = "no_delay"
    credit_limit: float = 18000.0
    age: float = 18000.0  # reference quirk kept: app/model.py:22
    bill_amount_1: float = 764.95
    bill_amount_2: float = 2221.95
    bill_amount_3: float = 1131.85
    bill_amount_4: float = 5074.85
    bill_amount_5: float = 18000.0
    bill_amount_6: float = 1419.95
    payment_amount_1: float = 2236.5
    payment_amount_2: float = 1137.55
    payment_amount_3: float = 5084.55
    payment_amount_4: float = 111.65
    payment_amount_5: float = 306.9
    payment_amount_6: float = 805.65

    model_config = ConfigDict(extra="ignore")


class FeatureBatchDrift(BaseModel):
    """Per-feature batch drift scores = 1 - p_val (reference app/model.py:37-61)."""

    sex: float
    education: float
    marriage: float
    repayment_status_1: float
    repayment_status_2: float
    repayment_status_3: float
    repayment_status_4: float
    repayment_status_5: float
    repayment_status_6: float
    credit_limit: float
    age: float
    bill_amount_1: float
    bill_amount_2: float
    bill_amount_3: float
    bill_amount_4: float
    bill_amount_5: float
    bill_amount_6: float
    payment_amount_1: float
    payment_amount_2: float
    payment_amount_3: float
    payment_amount_4: float
    payment_amount_5: float
    payment_amount_6: float


class ModelOutput(BaseModel):
    """Response envelope (reference app/model.py:64-70)."""

    predictions: list[float]
    outliers: list[float]
    feature_drift_batch: FeatureBatchDrift


# The one-record CI smoke-test fixture (reference app/sample-request.json).
SAMPLE_REQUEST: list[dict] = [
    {
        "sex": "male",
        "education": "university",
        "marriage": "married",
        "repayment_status_1": "duly_paid",
        "repayment_status_2": "duly_paid",
        "repayment_status_3": "duly_paid",
        "repayment_status_4": "duly_paid",
        "repayment_status_5": "no_delay",
        "repayment_status_6": "no_delay",
        "credit_limit": 18000,
        "age": 18000,
        "bill_amount_1": 764.95,
        "bill_amount_2": 2221.95,
        "bill_amount_3": 1131.85,
        "bill_amount_4": 5074.85,
        "bill_amount_5": 18000,
        "bill_amount_6": 1419.95,
        "payment_amount_1": 2236.5,
        "payment_amount_2": 1137.55,
        "payment_amount_3": 5084.55,
        "payment_amount_4": 111.65,
        "payment_amount_5": 306.9,
        "payment_amount_6": 805.65,
    }
]

# The sane single-record example used at registration time
# (reference 02-register-model.ipynb cell-4, INPUT_SAMPLE — age=33.0 there).
INPUT_SAMPLE: list[dict] = [dict(SAMPLE_REQUEST[0], age=33.0, bill_amount_5=3448.0)]
