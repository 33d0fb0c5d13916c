"""Shared fixtures. The reference has no tests; this suite is
the test pyramid built in its place: golden-parity units (HIP/CPU-ref vs
sklearn), API contract replays, and multi-process collective tests.

Marker: ``gpu`` — tests needing a real MI355X; everything else runs on CPU.
"""

from __future__ import annotations

import numpy as np
import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (run on MI355X)")
    config.addinivalue_line("markers", "slow: multi-process / lifecycle tests")


def _has_gpu() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def pytest_collection_modifyitems(config, items):
    if _has_gpu():
        return
    skip = pytest.mark.skip(reason="no GPU on this machine")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def train_df():
    from creditcore.data import make_uci_shaped_frame

    return make_uci_shaped_frame(n_rows=3000, seed=7)


@pytest.fixture(scope="session")
def model_dir(tmp_path_factory, train_df):
    """A small trained + packaged model (pyfunc layout) shared by the suite."""
    from creditcore import train as T

    d = str(tmp_path_factory.mktemp("model") / "model")
    T.train_and_register(model_dir=d, max_evals=2, df=train_df, register=False)
    return d


@pytest.fixture(scope="session")
def packed(model_dir):
    from creditcore.pack import pack_pyfunc_dir

    return pack_pyfunc_dir(model_dir)


@pytest.fixture(scope="session")
def loaded_pyfunc(model_dir):
    from creditcore.registry import load_pyfunc_model

    return load_pyfunc_model(model_dir)


@pytest.fixture(scope="session")
def score_batch(train_df):
    """A 512-row scoring batch drawn from the training distribution, with a
    few adversarial rows: NaN numerics and unknown categories."""
    from creditcore.schema import FEATURES

    df = train_df[FEATURES].sample(512, random_state=3).reset_index(drop=True)
    df = df.copy()
    df.loc[0, "credit_limit"] = np.nan
    df.loc[1, "bill_amount_3"] = np.nan
    df.loc[2, "education"] = "phd_never_seen"
    df.loc[3, "repayment_status_4"] = "delay_99_months"
    return df
