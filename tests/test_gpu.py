"""GPU numerics tests — HIP kernels (csrc/creditcore_kernels.hip) vs the
plain-PyTorch/NumPy fp32/f64 CPU reference (creditcore.ops.cpu_ref) on the
same packed buffers. Run on a real MI355X (`pytest -m gpu`)."""

from __future__ import annotations

import numpy as np
import pytest

from creditcore.engine import ScoringEngine
from creditcore.ops import cpu_ref
from creditcore.pack import encode_batch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu_engine(packed):
    from creditcore.ops import gpu

    assert gpu.available(), (
        "HIP extension must be present on a GPU box (no silent CPU fallback)"
    )
    return ScoringEngine(packed, device="cuda", device_index=0)


@pytest.fixture(scope="module")
def encoded(packed, score_batch):
    return encode_batch(score_batch, packed.vocabs)


def test_forest_parity(gpu_engine, packed, encoded):
    codes, nums = encoded
    out = gpu_engine.score_arrays(codes, nums, with_drift=False)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    ref = cpu_ref.score_forest_cpu(packed, codes, nums_imp)
    np.testing.assert_allclose(out["predictions"], ref, atol=1e-9)


def test_iforest_parity(gpu_engine, packed, encoded):
    codes, nums = encoded
    out = gpu_engine.score_arrays(codes, nums, with_drift=False)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    iscore, flags = cpu_ref.score_iforest_cpu(packed, nums_imp)
    np.testing.assert_allclose(out["instance_score"], iscore, atol=1e-9)
    np.testing.assert_array_equal(out["outliers"], flags)


def test_drift_parity(gpu_engine, packed, encoded):
    codes, nums = encoded
    out = gpu_engine.score_arrays(codes, nums, with_drift=True)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    hists, ks_d = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
    np.testing.assert_array_equal(out["cat_hist"], hists)
    np.testing.assert_allclose(out["ks_d"], ks_d, atol=1e-6)
    pvals = cpu_ref.pvals_from_stats(packed, hists, ks_d, len(codes))
    np.testing.assert_allclose(out["p_vals"], pvals, atol=1e-6)


@pytest.mark.parametrize("b", [1, 2, 63, 64, 65, 1024, 4096, 8192, 16384])
def test_batch_size_sweep(gpu_engine, packed, b):
    rng = np.random.default_rng(b)
    codes = np.stack(
        [rng.integers(-1, len(v), size=b) for v in packed.vocabs], axis=1
    ).astype(np.int16)
    nums = rng.normal(5000.0, 3000.0, size=(b, 14)).astype(np.float32)
    nums[rng.uniform(size=nums.shape) < 0.02] = np.nan
    out = gpu_engine.score_arrays(codes, nums, with_drift=True)
    ref = cpu_ref.score_batch_cpu(packed, codes, nums)
    np.testing.assert_allclose(out["predictions"], ref["predictions"], atol=1e-9)
    np.testing.assert_array_equal(out["outliers"], ref["outliers"])
    np.testing.assert_allclose(out["p_vals"], ref["p_vals"], atol=1e-6)


def test_score_records_end_to_end(gpu_engine):
    from creditcore.schema import SAMPLE_REQUEST

    out = gpu_engine.score_records(SAMPLE_REQUEST * 32)
    resp = out["response"]
    assert len(resp["predictions"]) == 32
    assert all(0.0 <= p <= 1.0 for p in resp["predictions"])
    # identical rows -> identical predictions
    assert len({round(p, 12) for p in resp["predictions"]}) == 1


def test_engine_refuses_silent_fallback(packed, monkeypatch):
    """On a GPU box the HIP extension is mandatory — a missing extension must
    raise, never fall back to CPU (ops/gpu.py contract)."""
    from creditcore.ops import gpu

    monkeypatch.setattr(gpu, "_ext", None)
    monkeypatch.setattr(gpu, "_err", ImportError("forced"))
    with pytest.raises(gpu.ExtensionMissing):
        ScoringEngine(packed, device="cuda", device_index=0)


def test_irregular_batch_sizes_eager_path(gpu_engine, packed):
    """Non-bucketed sizes skip graph capture and run eagerly — results
    must be identical to the CPU reference."""
    rng = np.random.default_rng(0)
    for b in (3, 97, 300, 777, 1500):
        codes = np.stack(
            [rng.integers(-1, len(v), size=b) for v in packed.vocabs], axis=1
        ).astype(np.int16)
        nums = rng.normal(1000.0, 500.0, size=(b, 14)).astype(np.float32)
        out = gpu_engine.score_arrays(codes, nums)
        ref = cpu_ref.score_batch_cpu(packed, codes, nums)
        np.testing.assert_allclose(out["predictions"], ref["predictions"], atol=1e-9)
        np.testing.assert_allclose(out["p_vals"], ref["p_vals"], atol=1e-6)


def test_interleaved_batch_sizes(gpu_engine, packed):
    """Batch sizes alternating large -> small -> large on one session.
    Guards the accumulator all-zero invariant (finalize_kernel re-zeros the
    rows it consumed instead of a per-request memset): a stale row from a
    larger earlier batch would corrupt a later one."""
    rng = np.random.default_rng(11)
    for b in (4096, 64, 4096, 1, 2048, 300, 4096):
        codes = np.stack(
            [rng.integers(-1, len(v), size=b) for v in packed.vocabs], axis=1
        ).astype(np.int16)
        nums = rng.normal(5000.0, 3000.0, size=(b, 14)).astype(np.float32)
        out = gpu_engine.score_arrays(codes, nums, with_drift=True)
        ref = cpu_ref.score_batch_cpu(packed, codes, nums)
        np.testing.assert_allclose(out["predictions"], ref["predictions"], atol=1e-9)
        np.testing.assert_allclose(out["p_vals"], ref["p_vals"], atol=1e-6)


def test_serving_stack_on_gpu(model_dir):
    """Whole serving stack on the GPU engine (TestClient): contract-valid
    responses through both the solo-flush bytes path and the merged path."""
    from fastapi.testclient import TestClient

    from creditcore.config import ServeConfig
    from creditcore.schema import SAMPLE_REQUEST, ModelOutput
    from creditcore.serve import create_app
    from creditcore.data import make_request_batch

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cuda"
    app = create_app(cfg)
    with TestClient(app) as client:
        r = client.post("/predict", json=SAMPLE_REQUEST)
        assert r.status_code == 200
        ModelOutput.model_validate(r.json())
        r = client.post("/score", json=make_request_batch(300, seed=2))
        assert r.status_code == 200
        body = r.json()
        assert len(body["predictions"]) == 300
        assert all(0.0 <= p <= 1.0 for p in body["predictions"])
        assert client.get("/healthz").json()["status"] == "ok"
        drift = client.get("/drift").json()
        assert drift["rows"] >= 301


def _random_encoded(packed, b, seed=0):
    rng = np.random.default_rng(seed)
    codes = np.stack(
        [rng.integers(-1, len(v), size=b) for v in packed.vocabs], axis=1
    ).astype(np.int16)
    nums = rng.normal(5000.0, 3000.0, size=(b, 14)).astype(np.float32)
    return codes, nums


def test_oversized_batch_bytes_path(gpu_engine, packed):
    """b > DRIFT_MAX_ROWS through score_encoded_bytes: predictions for every
    row (including rows >= 16384) must match the CPU reference — guards the
    round-1 advisor finding where the capped drift second pass overwrote the
    b-packed pinned outputs before serialization."""
    import json

    b = 20000
    codes, nums = _random_encoded(packed, b, seed=7)
    out = gpu_engine.score_encoded_bytes(codes, nums)
    assert out["rows"] == b
    resp = json.loads(out["response_bytes"])
    assert len(resp["predictions"]) == b and len(resp["outliers"]) == b
    ref = cpu_ref.score_batch_cpu(packed, codes, nums)
    np.testing.assert_allclose(resp["predictions"], ref["predictions"], atol=1e-9)
    np.testing.assert_array_equal(resp["outliers"], ref["outliers"])
    # drift block reflects the capped 16384-row sample
    chists, ks_d = cpu_ref.drift_stats_cpu(
        packed,
        codes[: gpu_engine.DRIFT_MAX_ROWS],
        cpu_ref.impute_nums(packed, nums[: gpu_engine.DRIFT_MAX_ROWS]),
    )
    pvals = cpu_ref.pvals_from_stats(packed, chists, ks_d, gpu_engine.DRIFT_MAX_ROWS)
    from creditcore.schema import FEATURES

    got = np.array([resp["feature_drift_batch"][f] for f in FEATURES])
    np.testing.assert_allclose(got, 1.0 - np.asarray(pvals, dtype=np.float32),
                               atol=1e-6)


def test_oversized_batch_json_full_path(gpu_engine, packed):
    """Same guard for the one-call C++ path (score_json_full at
    csrc/creditcore_kernels.hip score_json_full): grow capacity first so the
    C++ oversized branch (not the Python fallback) executes."""
    import json

    from creditcore.data import make_request_batch

    b = 20000
    # grow the session past b so score_json_full takes the C++ branch
    gpu_engine._ensure_capacity(b)
    body = json.dumps(make_request_batch(b, seed=3)).encode()
    out = gpu_engine.score_json_full(body)
    assert out["rows"] == b
    resp = json.loads(out["response_bytes"])
    assert len(resp["predictions"]) == b
    codes, nums = gpu_engine.encode_json_body(body)
    ref = cpu_ref.score_batch_cpu(packed, codes, nums)
    np.testing.assert_allclose(resp["predictions"], ref["predictions"], atol=1e-9)
    np.testing.assert_array_equal(resp["outliers"], ref["outliers"])


def test_drift_cap_config_respected(packed):
    """A configured drift_max_batch below the hardware ceiling caps the
    drift sample while every row is still scored (config wiring —
    round-1 advisor low finding)."""
    import json

    eng = ScoringEngine(packed, device="cuda", device_index=0, drift_max_rows=4096)
    assert eng.DRIFT_MAX_ROWS == 4096
    b = 6000
    codes, nums = _random_encoded(packed, b, seed=9)
    out = eng.score_encoded_bytes(codes, nums)
    resp = json.loads(out["response_bytes"])
    assert len(resp["predictions"]) == b
    ref = cpu_ref.score_batch_cpu(packed, codes, nums)
    np.testing.assert_allclose(resp["predictions"], ref["predictions"], atol=1e-9)


def test_gpu_fault_injection_failover_and_recovery(model_dir, monkeypatch):
    """Poison one GPU replica mid-traffic (faults surface exactly like HIP
    errors do — RuntimeError out of the scoring call): requests keep
    answering 200 on the survivor, the poisoned replica leaves rotation,
    and the probation loop re-admits it once it answers again
    (SURVEY §5.3 detection + recovery on hardware)."""
    import time as _time

    from fastapi.testclient import TestClient

    from creditcore.config import ServeConfig
    from creditcore.schema import SAMPLE_REQUEST
    from creditcore.serve import create_app, state

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cuda"
    cfg.n_gpus = 2  # two independent sessions (oversubscribed on 1 GPU)
    cfg.replica_probe_period_s = 0.3
    with TestClient(create_app(cfg)) as c:
        assert len(state["engines"]) == 2
        e0 = state["engines"][0]
        b0 = state["batchers"][0]
        orig_arrays = e0.score_arrays
        orig_barrays = b0.score_arrays
        orig_bsingle = b0.score_single

        def boom(*a, **k):
            raise RuntimeError("hipErrorInjected: device fault (test)")

        monkeypatch.setattr(e0, "score_arrays", boom)
        monkeypatch.setattr(b0, "score_arrays", boom)
        if b0.score_single is not None:
            monkeypatch.setattr(b0, "score_single", boom)
        for _ in range(8):
            assert c.post("/predict", json=SAMPLE_REQUEST).status_code == 200
        assert state["pool"].alive == [False, True]

        monkeypatch.setattr(e0, "score_arrays", orig_arrays)
        monkeypatch.setattr(b0, "score_arrays", orig_barrays)
        monkeypatch.setattr(b0, "score_single", orig_bsingle)
        deadline = _time.time() + 20
        while _time.time() < deadline and not all(state["pool"].alive):
            _time.sleep(0.1)
        assert all(state["pool"].alive), "replica 0 never revived"
        for _ in range(4):
            assert c.post("/predict", json=SAMPLE_REQUEST).status_code == 200


def test_native_code_is_loaded(gpu_engine):
    """The loaded extension must be the in-tree .so (native-code check)."""
    import creditcore._ccore as ccore

    assert "creditcore" in ccore.__file__
    assert ccore.__file__.endswith(".so")


def test_submit_arrays_strided_inputs(gpu_engine, packed):
    """submit_arrays must handle non-contiguous inputs (row-wise copy
    path) identically to contiguous ones."""
    b = 256
    codes, nums = _random_encoded(packed, 2 * b, seed=21)
    # every other row: stride-2 views, non-contiguous
    cs, ns = codes[::2], nums[::2]
    assert not cs.flags["C_CONTIGUOUS"]
    out_strided = gpu_engine.score_arrays(cs, ns, with_drift=False)
    out_contig = gpu_engine.score_arrays(
        np.ascontiguousarray(cs), np.ascontiguousarray(ns), with_drift=False
    )
    np.testing.assert_array_equal(out_strided["predictions"], out_contig["predictions"])
    np.testing.assert_array_equal(out_strided["outliers"], out_contig["outliers"])
