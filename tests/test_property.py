"""Property-based tests (hypothesis): the native encoder and the drift
statistics must hold their invariants on adversarial inputs, not just the
fixtures."""

from __future__ import annotations

import json
import math

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st
from scipy import stats

from creditcore.models.drift import ks_2samp_d


@settings(max_examples=60, deadline=None)
@given(
    ref=st.lists(st.floats(-1e6, 1e6, allow_nan=False), min_size=2, max_size=200),
    batch=st.lists(st.floats(-1e6, 1e6, allow_nan=False), min_size=1, max_size=120),
)
def test_ks_d_matches_scipy_random(ref, batch):
    ref_sorted = np.sort(np.asarray(ref, dtype=np.float64))
    d = ks_2samp_d(ref_sorted, np.asarray(batch, dtype=np.float64))
    sp = stats.ks_2samp(ref_sorted, batch, method="asymp").statistic
    assert abs(d - sp) < 1e-12


@settings(max_examples=60, deadline=None)
@given(
    ref=st.lists(st.sampled_from([0.0, 1.0, 2.5]), min_size=2, max_size=100),
    batch=st.lists(st.sampled_from([0.0, 1.0, 2.5, 7.0]), min_size=1, max_size=80),
)
def test_ks_d_matches_scipy_heavy_ties(ref, batch):
    ref_sorted = np.sort(np.asarray(ref, dtype=np.float64))
    d = ks_2samp_d(ref_sorted, np.asarray(batch, dtype=np.float64))
    sp = stats.ks_2samp(ref_sorted, batch, method="asymp").statistic
    assert abs(d - sp) < 1e-12


# JSON scalar values a client could send for any field
_scalar = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(-(10**12), 10**12),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=20),
    st.lists(st.integers(0, 3), max_size=3),
)


@pytest.fixture(scope="module")
def vocabs(packed):
    return packed.vocabs


@settings(max_examples=80, deadline=None)
@given(
    records=st.lists(
        st.dictionaries(
            st.sampled_from(
                ["sex", "education", "credit_limit", "age", "bill_amount_1",
                 "payment_amount_6", "repayment_status_3", "unknown_field"]
            ),
            _scalar,
            max_size=6,
        ),
        max_size=5,
    )
)
def test_encoder_never_crashes_and_matches_pydantic_accept_set(
    packed, records
):
    """For any JSON body: the native parser either raises ValueError (and
    the pydantic fallback decides) or returns arrays of the right shape.
    Whenever BOTH accept, the encodings must agree."""
    from creditcore.engine import ScoringEngine
    from creditcore.pack import encode_batch
    from creditcore.schema import LoanApplicant

    eng = ScoringEngine(packed, device="cpu")
    body = json.dumps(records).encode()
    try:
        codes, nums = eng.encode_json_body(body)
        native_ok = True
    except ValueError:
        native_ok = False
    try:
        validated = [LoanApplicant(**r) for r in records]
        pyd_ok = True
    except Exception:
        pyd_ok = False
    if native_ok:
        assert codes.shape == (len(records), 9)
        assert nums.shape == (len(records), 14)
        assert codes.dtype == np.int16 and nums.dtype == np.float32
        if pyd_ok:
            c2, n2 = encode_batch([v.__dict__ for v in validated], packed.vocabs)
            np.testing.assert_array_equal(codes, c2)
            # allow f32 rounding between float(v) and the parser's path
            both_nan = np.isnan(nums) & np.isnan(n2)
            close = np.isclose(nums, n2, rtol=1e-6, atol=0, equal_nan=True)
            assert (both_nan | close).all()


@settings(max_examples=40, deadline=None)
@given(junk=st.binary(max_size=64))
def test_encoder_rejects_junk_bytes(packed, junk):
    """Arbitrary bytes must raise ValueError (or parse, iff valid JSON of
    the right shape) — never crash the process."""
    from creditcore.engine import ScoringEngine

    eng = ScoringEngine(packed, device="cpu")
    try:
        codes, nums = eng.encode_json_body(bytes(junk))
        assert codes.shape[1] == 9
    except ValueError:
        pass


@settings(max_examples=25, deadline=None)
@given(
    seed=st.integers(0, 10_000),
    rows=st.integers(1, 200),
    nan_frac=st.floats(0.0, 0.4),
)
def test_driftsync_torch_and_numpy_paths_agree(packed, seed, rows, nan_frac):
    """DriftSync.accumulate has a torch path (GPU tensors) and a numpy fast
    path — they must produce identical histograms, NaNs included."""
    import torch

    from creditcore.parallel import DriftSync

    rng = np.random.default_rng(seed)
    C = int(packed.ref_cat_offsets[-1])
    cat_hist = rng.integers(0, 50, size=C).astype(np.int32)
    nums = rng.normal(5000.0, 4000.0, size=(rows, 14)).astype(np.float32)
    nums[rng.uniform(size=nums.shape) < nan_frac] = np.nan

    a = DriftSync(packed, device="cpu", n_bins=16)
    b = DriftSync(packed, device="cpu", n_bins=16)
    a.accumulate(torch.from_numpy(cat_hist.copy()), torch.from_numpy(nums.copy()))
    b.accumulate(cat_hist, nums)  # numpy fast path
    np.testing.assert_array_equal(a.local.numpy(), b.local.numpy())
    assert a.batches == b.batches == 1


@settings(max_examples=80, deadline=None)
@given(
    body=st.binary(max_size=256),
    ctype=st.sampled_from(["application/json", "application/octet-stream", ""]),
)
def test_parse_dense_body_never_crashes(body, ctype):
    """Fuzz: parse_dense_body on arbitrary bytes either returns a valid
    [B, F] float32 array or raises ValueError — never any other exception."""
    from creditcore.dense import parse_dense_body

    try:
        x = parse_dense_body(body, ctype, 8)
    except ValueError:
        return
    assert x.ndim == 2 and x.shape[1] == 8 and x.dtype == np.float32


@settings(max_examples=40, deadline=None)
@given(
    rows=st.integers(min_value=1, max_value=9),
    cols=st.integers(min_value=1, max_value=12),
    seed=st.integers(min_value=0, max_value=2**16),
)
def test_parse_dense_body_binary_roundtrip(rows, cols, seed):
    import struct

    from creditcore.dense import parse_dense_body

    rng = np.random.default_rng(seed)
    x = rng.normal(size=(rows, cols)).astype("<f4")
    body = struct.pack("<II", rows, cols) + x.tobytes()
    got = parse_dense_body(body, "application/octet-stream", cols)
    np.testing.assert_array_equal(got, x)
    # wrong feature count must be rejected
    with pytest.raises(ValueError):
        parse_dense_body(body, "application/octet-stream", cols + 1)


@settings(max_examples=60, deadline=None)
@given(
    data=st.lists(
        st.lists(
            st.one_of(st.none(), st.floats(allow_nan=False, allow_infinity=False,
                                           width=32)),
            min_size=5, max_size=5,
        ),
        min_size=1, max_size=6,
    )
)
def test_parse_dense_body_json_nulls_to_nan(data):
    import json as _json

    from creditcore.dense import parse_dense_body

    x = parse_dense_body(_json.dumps({"rows": data}).encode(),
                         "application/json", 5)
    for i, row in enumerate(data):
        for j, v in enumerate(row):
            if v is None:
                assert np.isnan(x[i, j])
            else:
                assert x[i, j] == np.float32(v)
