"""Replica health / failure-detection tests and the
node-global drift monitor endpoint."""

from __future__ import annotations

import pytest
from fastapi.testclient import TestClient

from creditcore.config import ServeConfig
from creditcore.schema import SAMPLE_REQUEST
from creditcore.serve import ReplicaPool, create_app, state


def test_pool_round_robin():
    pool = ReplicaPool(3)
    picks = [pool.pick() for _ in range(6)]
    assert picks == [0, 1, 2, 0, 1, 2]


def test_pool_drops_failed_replica():
    pool = ReplicaPool(2)
    for _ in range(3):
        pool.report_fail(0)
    assert pool.alive == [False, True]
    assert {pool.pick() for _ in range(4)} == {1}


def test_pool_ok_resets_fail_count():
    pool = ReplicaPool(1)
    pool.report_fail(0)
    pool.report_fail(0)
    pool.report_ok(0)
    pool.report_fail(0)
    assert pool.alive == [True]


def test_pool_all_dead_raises():
    pool = ReplicaPool(1)
    for _ in range(3):
        pool.report_fail(0)
    with pytest.raises(RuntimeError):
        pool.pick()


@pytest.fixture()
def client(model_dir):
    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    cfg.drift_sync_period = 1
    app = create_app(cfg)
    with TestClient(app) as c:
        yield c


def test_drift_endpoint(client):
    from creditcore.data import make_request_batch

    client.post("/predict", json=make_request_batch(128, seed=2))
    snap = client.get("/drift").json()
    assert snap["rows"] >= 128
    assert len(snap["node_feature_drift"]) == 23


def test_dead_replicas_give_503(client):
    pool = state["pool"]
    for _ in range(3):
        pool.report_fail(0)
    r = client.post("/predict", json=SAMPLE_REQUEST)
    assert r.status_code == 503
    assert client.get("/healthz").json()["status"] == "dead"


def test_engine_failure_marks_replica(client, monkeypatch):
    """Engine faults drop the replica from rotation. With per-request
    failover, a single request burns through the sole replica's failure
    budget and surfaces 503 (no healthy replicas) — faster detection than
    the old one-failure-per-request accounting."""
    batcher = state["batchers"][0]

    def boom(codes, nums):
        raise RuntimeError("injected HIP fault")

    monkeypatch.setattr(batcher, "score_arrays", boom)
    r = client.post("/predict", json=SAMPLE_REQUEST)
    assert r.status_code == 503
    assert state["pool"].alive == [False]
    assert client.post("/predict", json=SAMPLE_REQUEST).status_code == 503


def test_failover_keeps_service_503_free(model_dir, monkeypatch):
    """With 2 replicas and one poisoned mid-traffic, every request must
    still answer 200 off the survivor (the K8s-Service-retry analog);
    the poisoned replica leaves rotation after its failure budget."""
    from fastapi.testclient import TestClient

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    cfg.n_gpus = 2  # two CPU replicas
    cfg.replica_probe_period_s = 3600  # keep probation out of this test
    with TestClient(create_app(cfg)) as c:
        assert len(state["engines"]) == 2
        b0 = state["batchers"][0]

        def boom(codes, nums):
            raise RuntimeError("injected HIP fault")

        monkeypatch.setattr(b0, "score_arrays", boom)
        for _ in range(8):
            assert c.post("/predict", json=SAMPLE_REQUEST).status_code == 200
        assert state["pool"].alive == [False, True]
        hz = c.get("/healthz").json()
        assert hz["status"] == "ok"  # service healthy on the survivor


def test_probation_revives_recovered_replica(model_dir, monkeypatch):
    """A dead replica that starts answering again is re-admitted by the
    probation loop (SURVEY §5.3 recovery — round-1 had detection only)."""
    import time as _time

    from fastapi.testclient import TestClient

    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    cfg.n_gpus = 2
    cfg.replica_probe_period_s = 0.3
    with TestClient(create_app(cfg)) as c:
        e0 = state["engines"][0]
        orig = e0.score_arrays

        def boom(*a, **k):
            raise RuntimeError("injected HIP fault")

        # poison the engine itself so both the batcher and the probe fail
        monkeypatch.setattr(e0, "score_arrays", boom)
        b0 = state["batchers"][0]
        monkeypatch.setattr(b0, "score_arrays", boom)
        for _ in range(6):
            assert c.post("/predict", json=SAMPLE_REQUEST).status_code == 200
        assert state["pool"].alive == [False, True]

        # replica recovers: probe succeeds, pool re-admits it
        monkeypatch.setattr(e0, "score_arrays", orig)
        monkeypatch.setattr(b0, "score_arrays", orig)
        deadline = _time.time() + 15
        while _time.time() < deadline and not all(state["pool"].alive):
            _time.sleep(0.1)
        assert all(state["pool"].alive), "probation loop never revived replica 0"
        for _ in range(4):
            assert c.post("/predict", json=SAMPLE_REQUEST).status_code == 200


def test_deep_health_probe(client):
    body = client.get("/healthz?deep=1").json()
    assert body["status"] == "ok"
    assert body["engines"][0]["probe"] == "ok"
