"""FastAPI serving app — the reference's REST contract, GPU-backed.

Contract parity with reference app/main.py:
- ``POST /predict`` accepting ``list[LoanApplicant]``, returning ``ModelOutput``
  (app/main.py:42-86); ``/score`` is an alias (BASELINE.json names the
  endpoint /score).
- Swagger docs at ``/`` (app/main.py:37).
- Env config MODEL_DIRECTORY / SERVICE_NAME honoured (app/main.py:27,36).
- Two JSON log lines per request correlated by request_id (app/main.py:57-84).

MI355X-native additions: lifespan loads the packed model onto every visible
GPU (one ScoringEngine + micro-batcher per GPU, round-robin dispatch — the
in-node replacement for K8s replica scaling), /healthz and /metrics
endpoints, per-request latency in the ModelOutput log line.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from contextlib import asynccontextmanager
from itertools import count

from fastapi import FastAPI, HTTPException

from .config import ServeConfig
from .batching import MicroBatcher
from .engine import ScoringEngine, load_engine
from .pack import encode_batch
from .schema import FEATURES, LoanApplicant, ModelOutput
from .utils import logging as reqlog
from .utils.metrics import Metrics

state: dict = {}


def _build_engines(cfg: ServeConfig) -> list[ScoringEngine]:
    device = cfg.resolve_device()
    if device == "cpu":
        return [load_engine(cfg.model_directory, device="cpu")]
    import torch

    n = cfg.n_gpus or torch.cuda.device_count()
    first = load_engine(cfg.model_directory, device="cuda", device_index=0)
    engines = [first]
    for i in range(1, n):
        engines.append(ScoringEngine(first.packed, device="cuda", device_index=i))
    return engines


@asynccontextmanager
async def lifespan(app: FastAPI):
    cfg: ServeConfig = app.state.cfg
    engines = _build_engines(cfg)
    batchers = [
        MicroBatcher(
            e.score_arrays, max_rows=cfg.max_batch_rows, max_wait_us=cfg.batch_wait_us
        )
        for e in engines
    ]
    for b in batchers:
        await b.start()
    state["engines"] = engines
    state["batchers"] = batchers
    state["rr"] = count()
    state["metrics"] = Metrics()
    state["cfg"] = cfg
    yield
    for b in batchers:
        await b.close()
    state.clear()


def create_app(cfg: ServeConfig | None = None) -> FastAPI:
    cfg = cfg or ServeConfig()
    app = FastAPI(title=cfg.service_name, docs_url="/", lifespan=lifespan)
    app.state.cfg = cfg

    async def _predict_impl(data: list[LoanApplicant]) -> dict:
        if not data:
            raise HTTPException(status_code=400, detail="empty request batch")
        cfg: ServeConfig = state["cfg"]
        metrics: Metrics = state["metrics"]
        request_id = uuid.uuid4().hex
        # pydantic v2 keeps validated fields in __dict__; avoids the
        # per-request model_dump() copy on the hot path
        records = [r.__dict__ for r in data]

        if cfg.log_inference_data:
            reqlog.log_inference_data(
                cfg.service_name, request_id, json.dumps(records)
            )

        engines = state["engines"]
        batchers = state["batchers"]
        idx = next(state["rr"]) % len(engines)
        codes, nums = encode_batch(records, engines[idx].packed.vocabs)

        t0 = time.perf_counter()
        try:
            out = await batchers[idx].submit(codes, nums)
        except Exception as e:
            metrics.observe_error()
            raise HTTPException(status_code=500, detail=f"scoring failed: {e}")
        latency_ms = (time.perf_counter() - t0) * 1e3

        import numpy as np

        response = {
            "predictions": [float(x) for x in out["predictions"]],
            "outliers": [float(x) for x in out["outliers"]],
            "feature_drift_batch": {
                f: float(np.float32(1.0) - np.float32(p))
                for f, p in zip(FEATURES, out["p_vals"])
            },
        }
        metrics.observe_request(len(records), latency_ms)
        reqlog.log_model_output(
            cfg.service_name,
            request_id,
            response,
            latency_ms=latency_ms,
            rows=len(records),
            device=f"{engines[idx].device}:{engines[idx].device_index}",
        )
        return response

    @app.post("/predict", response_model=ModelOutput)
    async def predict(data: list[LoanApplicant]):
        """Reference endpoint (app/main.py:42)."""
        return await _predict_impl(data)

    @app.post("/score", response_model=ModelOutput)
    async def score(data: list[LoanApplicant]):
        """Alias — BASELINE.json names the endpoint /score."""
        return await _predict_impl(data)

    @app.get("/healthz")
    async def healthz():
        return {
            "status": "ok",
            "engines": [
                {"device": e.device, "index": e.device_index} for e in state["engines"]
            ],
        }

    @app.get("/metrics")
    async def metrics_endpoint():
        return state["metrics"].snapshot()

    return app


def main(argv: list[str] | None = None):
    import logging

    import uvicorn

    logging.basicConfig(level=logging.INFO)
    cfg = ServeConfig.from_args(argv)
    app = create_app(cfg)
    uvicorn.run(app, host=cfg.host, port=cfg.port, log_level="info")


if __name__ == "__main__":
    main()
