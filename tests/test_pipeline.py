"""End-to-end lifecycle test: the local CD pipeline (train → register →
staging → smoke → approve → production) — the reference's deploy workflow
semantics as a single-node integration test."""

from __future__ import annotations

import pytest

from creditcore.pipeline import PipelineError, run_pipeline, smoke_test


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.slow
@pytest.mark.timeout(300)
def test_full_pipeline(tmp_path):
    report = run_pipeline(
        model_dir=str(tmp_path / "model"),
        registry_root=str(tmp_path / "registry"),
        max_evals=1,
        n_rows=1200,
        staging_port=_free_port(),
        production_port=_free_port(),
        device="cpu",
        auto_approve=True,
    )
    stages = [s["stage"] for s in report["stages"]]
    assert report["status"] == "deployed"
    assert stages == [
        "train+register",
        "containerize",
        "staging+smoke",
        "approval",
        "production+smoke",
    ]
    assert report["stages"][0]["model_uri"].startswith("models:/")


def test_smoke_test_fails_on_dead_service():
    with pytest.raises(Exception):
        smoke_test("http://127.0.0.1:9")  # nothing listens on port 9
