"""Ops surface: Prometheus-format /metrics and drift-state persistence
across service restarts (both absent from the reference, which shipped logs
to Log Analytics and lost all drift context on pod restart)."""

from __future__ import annotations

import numpy as np
import pytest
import torch

from creditcore.utils.metrics import Metrics


def test_prometheus_rendering():
    m = Metrics()
    m.observe_request(64, 1.5)
    m.observe_request(64, 2.5)
    m.observe_error()
    text = m.prometheus()
    assert "# TYPE creditcore_requests_total counter" in text
    assert "creditcore_requests_total 2" in text
    assert "creditcore_rows_total 128" in text
    assert "creditcore_errors_total 1" in text
    assert 'creditcore_latency_ms{quantile="0.5"}' in text
    assert text.endswith("\n")


def test_metrics_endpoint_prometheus(model_dir):
    from fastapi.testclient import TestClient

    from creditcore.config import ServeConfig
    from creditcore.schema import SAMPLE_REQUEST
    from creditcore.serve import create_app

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    with TestClient(create_app(cfg)) as client:
        assert client.post("/score", json=SAMPLE_REQUEST).status_code == 200
        r = client.get("/metrics?format=prometheus")
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/plain")
        assert "creditcore_requests_total 1" in r.text
        # default stays JSON
        assert client.get("/metrics").json()["requests_total"] == 1


def _fold_batch(ds, packed, seed):
    from creditcore.data import make_request_batch
    from creditcore.ops import cpu_ref
    from creditcore.pack import encode_batch

    recs = make_request_batch(90, seed=seed)
    codes, nums = encode_batch(recs, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    hist, _ = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
    ds.accumulate(torch.from_numpy(hist), torch.from_numpy(nums))


def test_driftsync_state_roundtrip(packed, tmp_path):
    from creditcore.parallel import DriftSync

    p = str(tmp_path / "state" / "drift.npz")
    a = DriftSync(packed, device="cpu", n_bins=16)
    _fold_batch(a, packed, 1)
    _fold_batch(a, packed, 2)
    a.save_state(p)

    b = DriftSync(packed, device="cpu", n_bins=16)
    assert b.load_state(p)
    np.testing.assert_array_equal(a.local.numpy(), b.local.numpy())
    assert b.batches == 2
    b.allreduce()
    assert b.snapshot()["rows"] == 180

    # layout mismatch (different binning) must refuse and start fresh
    c = DriftSync(packed, device="cpu", n_bins=8)
    assert not c.load_state(p)
    assert int(c.local.sum()) == 0
    # missing file
    assert not c.load_state(str(tmp_path / "nope.npz"))


def test_drift_state_survives_service_restart(model_dir, tmp_path):
    from fastapi.testclient import TestClient

    from creditcore.config import ServeConfig
    from creditcore.data import make_request_batch
    from creditcore.serve import create_app

    path = str(tmp_path / "drift_state.npz")
    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    cfg.drift_state_path = path

    with TestClient(create_app(cfg)) as client:
        assert client.post("/score", json=make_request_batch(50, seed=3)).status_code == 200
        rows_before = client.get("/drift").json()["rows"]
    assert rows_before == 50

    # fresh app instance = restarted service: state is restored from disk
    with TestClient(create_app(cfg)) as client:
        assert client.get("/drift").json()["rows"] == 50
        client.post("/score", json=make_request_batch(25, seed=4))
        assert client.get("/drift").json()["rows"] == 75


def test_hot_reload_swaps_model(model_dir, train_df, tmp_path):
    """/admin/reload swaps to a new model version without a restart (the
    reference's only path to a new version was a full redeploy)."""
    from fastapi.testclient import TestClient

    from creditcore import registry
    from creditcore.config import ServeConfig
    from creditcore.data import make_request_batch
    from creditcore.models.forest import make_classifier_pipeline
    from creditcore.schema import FEATURES, TARGET
    from creditcore.serve import create_app
    from creditcore.train import fit_detectors

    # a deliberately different second model (tiny, different seed/shape)
    pipe = make_classifier_pipeline(
        {"n_estimators": 7, "max_depth": 2, "random_state": 99}
    )
    sub = train_df.head(1200)
    pipe.fit(sub[FEATURES], sub[TARGET].values.ravel())
    drift, outlier = fit_detectors(sub)
    v2_dir = str(tmp_path / "v2")
    registry.save_pyfunc_model(v2_dir, pipe, drift, outlier)

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    batch = make_request_batch(64, seed=5)
    with TestClient(create_app(cfg)) as client:
        before = client.post("/score", json=batch).json()["predictions"]
        r = client.post("/admin/reload", json={"model_uri": v2_dir})
        assert r.status_code == 200
        assert r.json()["status"] == "reloaded"
        after = client.post("/score", json=batch).json()["predictions"]
        # different model -> different scores; service never restarted
        assert before != after
        assert client.get("/healthz").json()["status"] == "ok"
        # bad URI is rejected and the live model keeps serving
        assert client.post(
            "/admin/reload", json={"model_uri": str(tmp_path / "missing")}
        ).status_code == 422
        again = client.post("/score", json=batch).json()["predictions"]
        assert again == after


@pytest.mark.slow
@pytest.mark.timeout(120)
def test_raw_server_sigterm_persists_drift(model_dir, tmp_path):
    """SIGTERM (the K8s pod-stop signal) must shut the raw server down
    through close(), persisting drift state."""
    import os
    import signal
    import socket
    import subprocess
    import sys
    import time

    import httpx

    import creditcore
    from creditcore.data import make_request_batch

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    state_path = str(tmp_path / "drift.npz")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(creditcore.__file__)))
    env = dict(os.environ, CREDITCORE_LOG_INFERENCE_DATA="0",
               CREDITCORE_LOG_RESPONSES="0",
               PYTHONPATH=repo + os.pathsep + os.environ.get("PYTHONPATH", ""))
    proc = subprocess.Popen(
        [sys.executable, "-m", "creditcore", "serve", "--raw-http",
         "--device", "cpu", "--host", "127.0.0.1", "--port", str(port),
         "--model-directory", model_dir, "--drift-state-path", state_path],
        env=env, cwd=str(tmp_path),
    )
    try:
        deadline = time.monotonic() + 90
        ok = False
        while time.monotonic() < deadline and not ok:
            try:
                r = httpx.post(f"http://127.0.0.1:{port}/score",
                               json=make_request_batch(40, seed=9), timeout=10.0)
                ok = r.status_code == 200
            except Exception:
                time.sleep(0.5)
        assert ok
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=30) == 0
        assert os.path.isfile(state_path)

        from creditcore.pack import pack_pyfunc_dir
        from creditcore.parallel import DriftSync

        ds = DriftSync(pack_pyfunc_dir(model_dir), device="cpu")
        assert ds.load_state(state_path)
        assert ds.snapshot()["rows"] == 40
    finally:
        try:
            proc.kill()
        except Exception:
            pass
