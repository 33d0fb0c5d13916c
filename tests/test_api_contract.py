"""API contract tests — the reference's smoke test (POST sample-request →
HTTP 200 + schema-valid body, reference .github/workflows/deploy-kubernetes.yml:206-271)
promoted to a unit test, plus the full wire-contract surface of
reference app/main.py / app/model.py."""

from __future__ import annotations

import json

import pytest
from fastapi.testclient import TestClient

from creditcore.config import ServeConfig
from creditcore.schema import FEATURES, SAMPLE_REQUEST
from creditcore.serve import create_app


@pytest.fixture(scope="module")
def client(model_dir):
    cfg = ServeConfig()
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    app = create_app(cfg)
    with TestClient(app) as c:
        yield c


def _check_body(body, n_rows):
    assert set(body.keys()) == {"predictions", "outliers", "feature_drift_batch"}
    assert len(body["predictions"]) == n_rows
    assert len(body["outliers"]) == n_rows
    assert all(0.0 <= p <= 1.0 for p in body["predictions"])
    assert all(o in (0.0, 1.0) for o in body["outliers"])
    assert set(body["feature_drift_batch"].keys()) == set(FEATURES)
    for v in body["feature_drift_batch"].values():
        assert -1e-6 <= v <= 1.0 + 1e-6


def test_smoke_sample_request(client):
    """The CI smoke test: the exact sample-request.json body → 200."""
    r = client.post("/predict", json=SAMPLE_REQUEST)
    assert r.status_code == 200
    _check_body(r.json(), 1)


def test_score_alias(client):
    r = client.post("/score", json=SAMPLE_REQUEST)
    assert r.status_code == 200
    _check_body(r.json(), 1)


def test_batch_request(client):
    from creditcore.data import make_request_batch

    batch = make_request_batch(64, seed=11)
    r = client.post("/predict", json=batch)
    assert r.status_code == 200
    _check_body(r.json(), 64)


def test_defaults_fill_missing_fields(client):
    """Pydantic defaults mirror the reference schema (app/model.py:8-34):
    an empty record is valid and scores."""
    r = client.post("/predict", json=[{}])
    assert r.status_code == 200
    _check_body(r.json(), 1)


def test_unknown_category_tolerated(client):
    """OneHotEncoder(handle_unknown='ignore') semantics — unseen categories
    must not 500 (reference 01-train cell-6)."""
    rec = dict(SAMPLE_REQUEST[0], education="unheard_of_degree")
    r = client.post("/predict", json=[rec])
    assert r.status_code == 200


def test_extra_fields_ignored(client):
    rec = dict(SAMPLE_REQUEST[0], bogus_field=123)
    r = client.post("/predict", json=[rec])
    assert r.status_code == 200


def test_empty_batch_rejected(client):
    r = client.post("/predict", json=[])
    assert r.status_code == 400


def test_invalid_type_rejected(client):
    rec = dict(SAMPLE_REQUEST[0], credit_limit="not-a-number")
    r = client.post("/predict", json=[rec])
    assert r.status_code == 422


def test_docs_at_root(client):
    """Swagger UI served at / (reference app/main.py:37)."""
    r = client.get("/")
    assert r.status_code == 200
    assert "swagger" in r.text.lower()


def test_healthz_and_metrics(client):
    assert client.get("/healthz").json()["status"] == "ok"
    m = client.post("/predict", json=SAMPLE_REQUEST) and client.get("/metrics").json()
    assert m["requests_total"] >= 1
    assert m["rows_total"] >= 1


def test_drifted_batch_reports_high_drift(client):
    from creditcore.data import make_request_batch

    r = client.post("/predict", json=make_request_batch(256, seed=5, drifted=True))
    assert r.status_code == 200
    drift = r.json()["feature_drift_batch"]
    # Shifted numerics → p ≈ 0 → 1-p ≈ 1 on every numeric feature.
    assert drift["credit_limit"] > 0.99
    assert drift["bill_amount_1"] > 0.99


def test_two_json_log_lines_per_request(client, capfd):
    """The observability contract: InferenceData + ModelOutput lines sharing
    a request_id (reference app/main.py:59-84)."""
    import logging

    records = []

    class Capture(logging.Handler):
        def emit(self, record):
            records.append(record.getMessage())

    h = Capture()
    lg = logging.getLogger("creditcore.requests")
    old_level = lg.level
    lg.setLevel(logging.INFO)
    lg.addHandler(h)
    try:
        client.post("/predict", json=SAMPLE_REQUEST)
    finally:
        lg.removeHandler(h)
        lg.setLevel(old_level)
    docs = [json.loads(m) for m in records]
    types = [d["type"] for d in docs]
    assert "InferenceData" in types and "ModelOutput" in types
    rid = {d["request_id"] for d in docs}
    assert len(rid) == 1
