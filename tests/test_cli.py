"""CLI entrypoint coverage (python -m creditcore ...) — the reference's
job/workflow invocations as subprocess smoke tests."""

from __future__ import annotations

import json
import subprocess
import sys


def _run(*args, timeout=240):
    return subprocess.run(
        [sys.executable, "-m", "creditcore", *args],
        capture_output=True,
        text=True,
        timeout=timeout,
    )


def test_usage_on_unknown_command():
    r = _run("frobnicate")
    assert r.returncode == 2
    assert "usage" in r.stderr


def test_train_pack_roundtrip(tmp_path):
    model_dir = str(tmp_path / "model")
    r = _run(
        "train", "--model-dir", model_dir, "--max-evals", "1",
        "--n-rows", "1200", "--no-register",
    )
    assert r.returncode == 0, r.stderr[-500:]
    assert r.stdout.strip().endswith(model_dir)

    r = _run("pack", "--model-dir", model_dir)
    assert r.returncode == 0, r.stderr[-500:]
    info = json.loads(r.stdout.strip().splitlines()[-1])
    assert info["cls_trees"] >= 100
    assert info["if_trees"] == 100
    assert (tmp_path / "model" / "packed.npz").exists()


def test_generate_data(tmp_path):
    out = str(tmp_path / "c.csv")
    r = _run("generate-data", "--out", out, "--n-rows", "500")
    assert r.returncode == 0
    head = open(out).readline()
    assert head.startswith("sex,education,marriage")


def test_score_batch(tmp_path):
    model_dir = str(tmp_path / "model")
    _run("train", "--model-dir", model_dir, "--max-evals", "1",
         "--n-rows", "1200", "--no-register")
    inp = str(tmp_path / "in.csv")
    _run("generate-data", "--out", str(tmp_path / "_c.csv"),
         "--n-rows", "10", "--inference-sample", inp)
    out = str(tmp_path / "scored.csv")
    r = _run("score-batch", "--model-dir", model_dir, "--input", inp,
             "--output", out, "--device", "cpu", "--chunk-rows", "32")
    assert r.returncode == 0, r.stderr[-500:]
    import pandas as pd

    df = pd.read_csv(out)
    assert "prediction" in df and "is_outlier" in df and len(df) == 80
    assert df["prediction"].between(0, 1).all()


def test_config_env_and_cli_precedence(monkeypatch):
    from creditcore.config import ServeConfig

    monkeypatch.setenv("CREDITCORE_PORT", "5055")
    monkeypatch.setenv("MODEL_DIRECTORY", "/env/model")  # reference env name
    cfg = ServeConfig()
    assert cfg.port == 5055
    assert cfg.model_directory == "/env/model"
    cfg2 = ServeConfig.from_args(["--port", "6000"])
    assert cfg2.port == 6000  # CLI wins over env default


def test_compose_deploy_target_is_valid():
    """Second deploy target (the ACA analog, reference
    deploy-container-app.yml): compose file parses and carries the
    staging->production gate + health checks."""
    import os

    import yaml

    p = os.path.join(os.path.dirname(os.path.dirname(__file__)),
                     "deploy", "docker-compose.yml")
    doc = yaml.safe_load(open(p))
    svcs = doc["services"]
    assert {"staging", "production"} <= set(svcs)
    assert svcs["production"]["depends_on"]["staging"]["condition"] == "service_healthy"
    for s in ("staging", "production"):
        assert "healthcheck" in svcs[s]
        assert any("5000" in p for p in svcs[s]["ports"])


def test_lint_gate_clean():
    """Static-check analog of the reference's bicep lint + dependabot
    (SURVEY §4 static checks): AST lint + dependency-pin freshness."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(__file__))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "lint.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stdout + r.stderr


def test_config_new_round2_fields(monkeypatch):
    """Round-2 config fields parse from env and CLI with the same
    precedence as the rest of ServeConfig."""
    from creditcore.config import ServeConfig

    monkeypatch.setenv("CREDITCORE_ADMIN_TOKEN", "envtok")
    monkeypatch.setenv("CREDITCORE_REPLICA_PROBE_PERIOD_S", "2.5")
    monkeypatch.setenv("CREDITCORE_DRIFT_MAX_BATCH", "4096")
    cfg = ServeConfig()
    assert cfg.admin_token == "envtok"
    assert cfg.replica_probe_period_s == 2.5
    assert cfg.drift_max_batch == 4096
    # CLI overrides env
    cfg2 = ServeConfig.from_args(
        ["--admin-token", "clitok", "--replica-probe-period-s", "7"]
    )
    assert cfg2.admin_token == "clitok"
    assert cfg2.replica_probe_period_s == 7.0
    assert cfg2.drift_max_batch == 4096  # env still applies where no flag


def test_engine_drift_cap_clamps_to_hardware(packed):
    """drift_max_batch above the 16384 LDS ceiling clamps instead of
    launching an unsupported K-S shape."""
    from creditcore.engine import ScoringEngine

    e = ScoringEngine(packed, device="cpu", drift_max_rows=99999)
    assert e.DRIFT_MAX_ROWS == ScoringEngine.HW_DRIFT_MAX_ROWS == 16384
    e2 = ScoringEngine(packed, device="cpu", drift_max_rows=2048)
    assert e2.DRIFT_MAX_ROWS == 2048
