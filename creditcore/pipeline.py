"""Local CD pipeline — the reference's GitHub Actions deploy workflow
(train → containerize → staging → smoke test → approval → production,
reference .github/workflows/deploy-kubernetes.yml:31-299) re-expressed as a
one-command lifecycle for a single MI355X node.

Stages:
1. train      — hyperparameter search + detector fitting + pyfunc packaging
                (the Databricks 2-task job)
2. register   — local model registry version bump (MLflow registry analog)
3. containerize (optional) — `docker build` of the serving image with the
                model baked in (reference bakes the model into the image at
                containerize time, app/Dockerfile:18) — skipped when docker
                is unavailable, the staging deploy then serves the model dir
4. staging    — launch the serving process on the staging port
5. smoke      — POST sample-request to staging; require HTTP 200 +
                schema-valid body (the reference's only test)
6. approval   — manual gate unless --auto-approve (reference: GH environment
                approval, deploy-kubernetes.yml:275-279)
7. production — launch on the production port + production smoke test
"""

from __future__ import annotations

import os
import shutil
import subprocess
import sys
import time


class PipelineError(RuntimeError):
    pass


def _wait_healthy(url: str, timeout_s: float = 60.0) -> None:
    import httpx

    t0 = time.time()
    while time.time() - t0 < timeout_s:
        try:
            r = httpx.get(f"{url}/healthz", timeout=2.0)
            if r.status_code == 200 and r.json().get("status") == "ok":
                return
        except Exception:
            pass
        time.sleep(0.5)
    raise PipelineError(f"service at {url} did not become healthy in {timeout_s}s")


def smoke_test(url: str) -> dict:
    """The reference CI smoke test: sample request → 200 + schema-valid."""
    import httpx

    from .schema import SAMPLE_REQUEST, ModelOutput

    r = httpx.post(f"{url}/predict", json=SAMPLE_REQUEST, timeout=30.0)
    if r.status_code != 200:
        raise PipelineError(f"smoke test failed: HTTP {r.status_code}: {r.text[:500]}")
    ModelOutput.model_validate(r.json())
    return r.json()


def _spawn_server(model_dir: str, port: int, device: str, extra_env: dict | None = None):
    env = dict(os.environ, **(extra_env or {}))
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "creditcore",
            "serve",
            "--model-directory",
            model_dir,
            "--port",
            str(port),
            "--host",
            "127.0.0.1",
            "--device",
            device,
        ],
        env=env,
    )
    return proc


def run_pipeline(
    model_dir: str = "./model",
    registry_root: str | None = None,
    model_name: str = "credit-default-uci-custom",
    max_evals: int = 10,
    n_rows: int = 20_000,
    staging_port: int = 5001,
    production_port: int = 5000,
    device: str = "auto",
    auto_approve: bool = False,
    keep_production: bool = False,
    docker_image: str | None = None,
    log=print,
) -> dict:
    from . import registry as reg
    from .train import train_and_register

    report: dict = {"stages": []}

    def stage(name, **kv):
        log(f"[pipeline] {name}: {kv}")
        report["stages"].append({"stage": name, **kv})

    # 1+2: train + register
    t0 = time.time()
    uri = train_and_register(
        model_dir=model_dir,
        model_name=model_name,
        registry_root=registry_root or reg.DEFAULT_REGISTRY_ROOT,
        max_evals=max_evals,
        n_rows=n_rows,
        register=True,
    )
    stage("train+register", model_uri=uri, seconds=round(time.time() - t0, 1))

    # 3: containerize (model baked into the image, reference app/Dockerfile:18)
    if docker_image:
        if shutil.which("docker") is None:
            raise PipelineError("docker not available for --docker-image")
        t0 = time.time()
        subprocess.run(
            ["docker", "build", "-f", "docker/Dockerfile", "-t", docker_image,
             "--build-arg", f"MODEL_DIR={model_dir}", "."],
            check=True,
        )
        stage("containerize", image=docker_image, seconds=round(time.time() - t0, 1))
    else:
        stage("containerize", skipped="no --docker-image (serving model dir directly)")

    staging = production = None
    try:
        # 4+5: staging deploy + smoke
        staging = _spawn_server(model_dir, staging_port, device)
        url_s = f"http://127.0.0.1:{staging_port}"
        _wait_healthy(url_s)
        body = smoke_test(url_s)
        stage("staging+smoke", url=url_s, predictions=body["predictions"])

        # 6: approval gate
        if not auto_approve:
            answer = input("[pipeline] promote to production? [y/N] ").strip().lower()
            if answer not in ("y", "yes"):
                stage("approval", approved=False)
                report["status"] = "stopped-at-approval"
                return report
        stage("approval", approved=True, auto=auto_approve)

        # 7: production deploy + smoke
        production = _spawn_server(model_dir, production_port, device)
        url_p = f"http://127.0.0.1:{production_port}"
        _wait_healthy(url_p)
        smoke_test(url_p)
        stage("production+smoke", url=url_p)
        report["status"] = "deployed"
        if keep_production:
            stage("production-running", pid=production.pid, url=url_p)
            production = None  # leave it running
        return report
    finally:
        for proc in (staging, production):
            if proc is not None:
                proc.terminate()
                try:
                    proc.wait(timeout=15)
                except subprocess.TimeoutExpired:
                    proc.kill()
