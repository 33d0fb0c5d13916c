"""Native JSON request parser (csrc encode_json) vs the pydantic reference
path — the serving fast path must be byte-for-byte equivalent on valid
bodies and must *reject* (→ fallback) anything pydantic would treat
differently (nulls, wrong types)."""

from __future__ import annotations

import json

import numpy as np
import pytest

from creditcore.pack import encode_batch
from creditcore.schema import (
    CATEGORICAL_FEATURES,
    MISSING_CATEGORY,
    NUMERIC_FEATURES,
    LoanApplicant,
)


@pytest.fixture(scope="module")
def ext():
    import torch  # noqa: F401  (loads libc10 for the extension)

    from creditcore.ops import gpu

    if not gpu.available():
        pytest.skip("extension not built")
    return gpu.ext()


@pytest.fixture(scope="module")
def defaults(packed):
    dc, dn = encode_batch([LoanApplicant().__dict__], packed.vocabs)
    return np.ascontiguousarray(dc[0]), np.ascontiguousarray(dn[0])


def _enc(ext, packed, defaults, body: bytes):
    c, n = ext.encode_json(
        body, packed.vocabs, CATEGORICAL_FEATURES, NUMERIC_FEATURES,
        MISSING_CATEGORY, defaults[0], defaults[1],
    )
    return np.asarray(c), np.asarray(n)


def test_parity_with_pydantic_path(ext, packed, defaults):
    from creditcore.data import make_request_batch

    recs = make_request_batch(257, seed=3)
    body = json.dumps(recs).encode()
    c1, n1 = _enc(ext, packed, defaults, body)
    validated = [LoanApplicant(**r).__dict__ for r in recs]
    c2, n2 = encode_batch(validated, packed.vocabs)
    np.testing.assert_array_equal(c1, c2)
    np.testing.assert_array_equal(n1, n2)


def test_absent_fields_take_schema_defaults(ext, packed, defaults):
    c, n = _enc(ext, packed, defaults, b"[{}]")
    validated = [LoanApplicant().__dict__]
    c2, n2 = encode_batch(validated, packed.vocabs)
    np.testing.assert_array_equal(c, c2)
    np.testing.assert_array_equal(n, n2)


def test_unknown_category_and_extra_fields(ext, packed, defaults):
    body = b'[{"education": "zzz", "extra": {"deep": [1, "x", {}]}, "age": 44}]'
    c, n = _enc(ext, packed, defaults, body)
    assert c[0, CATEGORICAL_FEATURES.index("education")] == -1
    assert n[0, NUMERIC_FEATURES.index("age")] == 44.0


def test_escapes_and_numbers(ext, packed, defaults):
    body = b'[{"sex": "mal\\u0065", "credit_limit": 1.5e4, "age": -3}]'
    c, n = _enc(ext, packed, defaults, body)
    assert c[0, 0] == packed.vocabs[0].index("male")
    assert n[0, 0] == 15000.0
    assert n[0, 1] == -3.0


@pytest.mark.parametrize(
    "bad",
    [
        b"{}",  # not an array
        b'[{"sex": 5}]',  # number for string field
        b'[{"sex": null}]',  # null (pydantic rejects -> 422 via fallback)
        b'[{"credit_limit": "x"}]',  # string for numeric field
        b'[{"credit_limit": null}]',
        b"[",  # truncated
        b'[{"sex" "male"}]',  # missing colon
        b'[{"sex": "male"}] trailing',
    ],
)
def test_malformed_rejected(ext, packed, defaults, bad):
    with pytest.raises(ValueError):
        _enc(ext, packed, defaults, bad)


def test_empty_array(ext, packed, defaults):
    c, n = _enc(ext, packed, defaults, b" [ ] ")
    assert c.shape == (0, 9) and n.shape == (0, 14)


def test_response_bytes_matches_dict_path(packed):
    """score_json_bytes (C serializer) must produce the same JSON values as
    the dict path at double precision."""
    import json

    from creditcore.data import make_request_batch
    from creditcore.engine import ScoringEngine

    eng = ScoringEngine(packed, device="cpu")
    recs = make_request_batch(64, seed=21)
    body = json.dumps(recs).encode()
    d = eng.score_json(body)["response"]
    # CPU path serializes via json.dumps; on GPU the C path is exercised by
    # the gpu-marked test below. Round-trip equality check:
    b = json.loads(eng.score_json_bytes(body)["response_bytes"])
    assert b == json.loads(json.dumps(d))


@pytest.mark.gpu
def test_response_bytes_gpu_matches_dict_path(packed):
    import json

    from creditcore.data import make_request_batch
    from creditcore.engine import ScoringEngine

    eng = ScoringEngine(packed, device="cuda")
    recs = make_request_batch(257, seed=22)
    body = json.dumps(recs).encode()
    d = eng.score_json(body)["response"]
    b = json.loads(eng.score_json_bytes(body)["response_bytes"])
    np.testing.assert_allclose(b["predictions"], d["predictions"], rtol=0, atol=0)
    np.testing.assert_array_equal(b["outliers"], d["outliers"])
    for k, v in d["feature_drift_batch"].items():
        assert abs(b["feature_drift_batch"][k] - v) < 1e-12


@pytest.mark.gpu
def test_score_json_full_matches_staged(packed):
    import json

    from creditcore.data import make_request_batch
    from creditcore.engine import ScoringEngine

    eng = ScoringEngine(packed, device="cuda")
    for b in (1, 64, 1024):
        body = json.dumps(make_request_batch(b, seed=b)).encode()
        full = eng.score_json_full(body)
        staged = eng.score_json_bytes(body)
        assert full["rows"] == staged["rows"] == b
        a = json.loads(full["response_bytes"])
        c = json.loads(staged["response_bytes"])
        assert a == c
    with pytest.raises(ValueError):
        eng.score_json_full(b"[]")
    with pytest.raises(ValueError):
        eng.score_json_full(b"not json")


@pytest.mark.gpu
def test_slot_pipeline_matches_sync(packed):
    """submit_encoded_slot/finish_slot (double-buffered async path) must
    produce byte-identical responses to the synchronous path, including
    when slots alternate across differing batches."""
    import json as _json

    from creditcore.data import make_request_batch
    from creditcore.engine import ScoringEngine
    from creditcore.pack import encode_batch

    eng = ScoringEngine(packed, device="cuda")
    batches = [
        encode_batch(make_request_batch(b, seed=40 + b), packed.vocabs)
        for b in (64, 257, 64, 1024, 513)
    ]
    want = [eng.score_encoded_bytes(c, n)["response_bytes"] for c, n in batches]
    got = [None] * len(batches)
    pending = None
    for i, (c, n) in enumerate(batches):
        b = eng.submit_encoded_slot(c, n, i & 1)
        if pending is not None:
            got[i - 1] = eng.finish_slot(pending[0], pending[1])["response_bytes"]
        pending = (i & 1, b)
    got[-1] = eng.finish_slot(pending[0], pending[1])["response_bytes"]
    for w, g in zip(want, got):
        assert _json.loads(w) == _json.loads(g)


def test_serving_fallback_handles_lax_types(model_dir):
    """A numeric string coerces through the pydantic fallback exactly as the
    reference would (fast path rejects, fallback accepts)."""
    from fastapi.testclient import TestClient

    from creditcore.config import ServeConfig
    from creditcore.serve import create_app

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    app = create_app(cfg)
    with TestClient(app) as client:
        r = client.post("/predict", json=[{"credit_limit": "18000"}])
        assert r.status_code == 200
        r = client.post("/predict", json=[{"sex": None}])
        assert r.status_code == 422


def test_c_serializer_double_roundtrip(ext):
    """append_double (C, shortest-round-trip) must reproduce every double
    bit-exactly through JSON, including awkward magnitudes."""
    import json as _json

    import torch

    vals = [0.0, 1.0, 0.1, 2 / 3, 1e-300, 1.7976931348623157e308,
            5e-324, 123456789.123456789, 0.49999999999999994, 1e16, 1e-5]
    b = len(vals)
    outs = torch.zeros(3 * b, dtype=torch.float64)
    outs[:b] = torch.tensor(vals, dtype=torch.float64)
    outs[2 * b:] = torch.tensor([i % 2 for i in range(b)], dtype=torch.float64)
    pvals = np.linspace(0.0, 1.0, 23)
    from creditcore.schema import FEATURES

    raw = ext.build_response_json(outs, b, np.ascontiguousarray(pvals), FEATURES)
    doc = _json.loads(raw)
    for got, want in zip(doc["predictions"], vals):
        assert got == want, (got, want)
    assert doc["outliers"] == [float(i % 2) for i in range(b)]
    for f, p in zip(FEATURES, pvals):
        assert doc["feature_drift_batch"][f] == float(np.float32(1.0) - np.float32(p))


def test_cpu_engine_json_body_fills_schema_defaults(packed, monkeypatch):
    """device=cpu encode_json_body (extension absent) must fill absent
    fields with the schema defaults (pydantic default semantics, reference
    app/model.py:8-34) — not MISSING_CATEGORY (round-1 advisor low
    finding)."""
    import json

    from creditcore.engine import ScoringEngine
    from creditcore.ops import gpu
    from creditcore.pack import encode_batch
    from creditcore.schema import LoanApplicant

    monkeypatch.setattr(gpu, "available", lambda: False)
    eng = ScoringEngine(packed, device="cpu")
    partial = [{"credit_limit": 123.0}, {"sex": "female", "age": 40.0}]
    codes, nums = eng.encode_json_body(json.dumps(partial).encode())
    full = [{**LoanApplicant().__dict__, **r} for r in partial]
    codes_ref, nums_ref = encode_batch(full, packed.vocabs)
    np.testing.assert_array_equal(codes, codes_ref)
    np.testing.assert_array_equal(nums, nums_ref)
    # malformed shapes raise ValueError (callers map it to 4xx)
    import pytest as _pytest

    with _pytest.raises(ValueError):
        eng.encode_json_body(b'{"not": "a list"}')
    with _pytest.raises(ValueError):
        eng.encode_json_body(b'[1, 2]')


def test_c_serializer_arrays_roundtrip(ext):
    """build_response_json_arrays (merged-flush path) must reproduce every
    double bit-exactly and match the reference (1 - p) float32 semantics."""
    import json as _json

    from creditcore.schema import FEATURES

    vals = [0.0, 1.0, 2 / 3, 1e-300, 5e-324, 0.49999999999999994, 1e16]
    preds = np.array(vals, dtype=np.float64)
    outl = np.array([i % 2 for i in range(len(vals))], dtype=np.float64)
    pvals = np.linspace(0.0, 1.0, 23)
    raw = ext.build_response_json_arrays(preds, outl, pvals, FEATURES)
    doc = _json.loads(raw)
    assert doc["predictions"] == vals
    assert doc["outliers"] == [float(i % 2) for i in range(len(vals))]
    for f, p in zip(FEATURES, pvals):
        assert doc["feature_drift_batch"][f] == float(np.float32(1.0) - np.float32(p))
    # size mismatches are rejected loudly
    import pytest as _pytest

    with _pytest.raises(Exception):
        ext.build_response_json_arrays(preds, outl[:-1], pvals, FEATURES)
    with _pytest.raises(Exception):
        ext.build_response_json_arrays(preds, outl, pvals[:-1], FEATURES)
