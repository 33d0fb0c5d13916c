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

import numpy as np
from fastapi import FastAPI, HTTPException, Request, Response

from .config import ServeConfig
from .batching import MicroBatcher
from .engine import ScoringEngine, load_engine
from .pack import encode_batch
from .schema import FEATURES, LoanApplicant, ModelOutput
from .utils import logging as reqlog
from .utils.metrics import Metrics

state: dict = {}


def _build_engines(cfg: ServeConfig) -> list[ScoringEngine]:
    """One ScoringEngine replica per serving slot. n_gpus picks the replica
    count (0 = one per visible GPU); asking for more replicas than devices
    oversubscribes round-robin (multiple independent sessions per GPU —
    used by failover tests and latency isolation)."""
    device = cfg.resolve_device()
    if device == "cpu":
        n = max(cfg.n_gpus, 1)
        first = load_engine(
            cfg.model_directory, device="cpu", drift_max_rows=cfg.drift_max_batch
        )
        return [first] + [
            ScoringEngine(first.packed, device="cpu", drift_max_rows=cfg.drift_max_batch)
            for _ in range(n - 1)
        ]
    import torch

    n_dev = max(torch.cuda.device_count(), 1)
    n = cfg.n_gpus or n_dev
    first = load_engine(
        cfg.model_directory,
        device="cuda",
        device_index=0,
        drift_max_rows=cfg.drift_max_batch,
    )
    engines = [first]
    for i in range(1, n):
        engines.append(
            ScoringEngine(
                first.packed,
                device="cuda",
                device_index=i % n_dev,
                drift_max_rows=cfg.drift_max_batch,
            )
        )
    return engines


# in-process test clients report these pseudo-hosts; real sockets give the
# peer IP ("testclient" is FastAPI's TestClient, None = no transport info
# from an in-process ASGI call)
_LOOPBACK_HOSTS = {"127.0.0.1", "::1", "localhost", "testclient"}


def admin_authorized(
    cfg: ServeConfig,
    client_host: str | None,
    authorization: str | None = None,
    x_admin_token: str | None = None,
) -> bool:
    """Gate for /admin/* (model hot-swap deserializes an arbitrary on-disk
    path — it must not be open to the public listener). Token configured →
    require it on every call; no token → loopback clients only."""
    if cfg.admin_token:
        presented = x_admin_token
        if presented is None and authorization and authorization.lower().startswith("bearer "):
            presented = authorization[7:]
        import secrets

        return presented is not None and secrets.compare_digest(presented, cfg.admin_token)
    return client_host is None or client_host in _LOOPBACK_HOSTS


class ReplicaPool:
    """Per-GPU replica health + dispatch (the in-node analog of K8s dropping
    a crashed pod from the Service — SURVEY.md §5.3). A replica is removed
    from rotation after ``max_failures`` consecutive scoring errors; scoring
    keeps flowing to the survivors."""

    def __init__(self, n: int, max_failures: int = 3):
        self.alive = [True] * n
        self.fails = [0] * n
        self.max_failures = max_failures
        self._rr = count()

    def pick(self) -> int:
        n = len(self.alive)
        for _ in range(n):
            i = next(self._rr) % n
            if self.alive[i]:
                return i
        raise RuntimeError("no healthy replicas")

    def report_ok(self, i: int) -> None:
        self.fails[i] = 0

    def report_fail(self, i: int) -> None:
        self.fails[i] += 1
        if self.fails[i] >= self.max_failures:
            self.alive[i] = False

    def revive(self, i: int) -> None:
        """Re-admit a dead replica after a successful probation probe (the
        recovery half of SURVEY.md §5.3 — detection/redistribution alone
        would shrink the pool monotonically until a full reload)."""
        self.fails[i] = 0
        self.alive[i] = True

    def dead_indices(self) -> list[int]:
        return [i for i, a in enumerate(self.alive) if not a]

    @property
    def attempt_budget(self) -> int:
        """Upper bound on per-request failover attempts that guarantees a
        healthy replica is reached if one exists (round-robin pick can hit
        a failing replica at most max_failures times before it drops)."""
        return len(self.alive) * self.max_failures + 1


async def probe_revive(pool: ReplicaPool, engines, logger=None) -> list[int]:
    """Probe every dead replica with the schema-default record and re-admit
    the ones that answer. Returns the revived indices."""
    import asyncio

    dead = pool.dead_indices()
    if not dead:
        return []
    codes, nums = encode_batch(
        [LoanApplicant().__dict__], engines[0].packed.vocabs
    )
    loop = asyncio.get_running_loop()
    revived = []
    for i in dead:
        try:
            await loop.run_in_executor(
                None, lambda e=engines[i]: e.score_arrays(codes, nums, False)
            )
            pool.revive(i)
            revived.append(i)
            if logger is not None:
                logger.info("replica %d revived after probation probe", i)
        except Exception:
            pass  # still dead; next period re-probes
    return revived


def _make_batchers(cfg: ServeConfig, engines, drift_sync) -> list[MicroBatcher]:
    """One micro-batcher per engine, each folding its drift stats into the
    shared node-level accumulator (allreduce every drift_sync_period)."""

    def _fold_drift(out, nums):
        if "cat_hist" in out:
            drift_sync.accumulate(out["cat_hist"], nums)
            if drift_sync.batches % max(cfg.drift_sync_period, 1) == 0:
                drift_sync.allreduce()
                m = state.get("metrics")
                if m is not None:
                    m.observe_drift_sync()

    def scorer(e: ScoringEngine):
        def run(codes, nums):
            out = e.score_arrays(codes, nums)
            _fold_drift(out, nums)
            return out

        return run

    def scorer_single(e: ScoringEngine):
        # GPU engines serialize the wire response in C for solo flushes
        if e.device != "cuda":
            return None

        def run(codes, nums):
            out = e.score_encoded_bytes(codes, nums)
            _fold_drift(out, nums)
            return out

        return run

    return [
        MicroBatcher(
            scorer(e),
            max_rows=cfg.max_batch_rows,
            max_wait_us=cfg.batch_wait_us,
            score_single=scorer_single(e),
        )
        for e in engines
    ]


@asynccontextmanager
async def lifespan(app: FastAPI):
    from .parallel import DriftSync

    cfg: ServeConfig = app.state.cfg
    engines = _build_engines(cfg)
    # One shared node-level drift accumulator (host-side; 23×bins int64 —
    # the per-request drift stats feeding it come from the HIP kernels).
    drift_sync = DriftSync(engines[0].packed, device="cpu")
    if cfg.drift_state_path:
        drift_sync.load_state(cfg.drift_state_path)

    batchers = _make_batchers(cfg, engines, drift_sync)
    for b in batchers:
        await b.start()
    # Dense wide-tabular family (BASELINE config 5) as a first-class served
    # path: one DenseEngine replica per GPU (each holding the full sorted
    # drift reference in its own HBM — the 8×40 GB data-parallel sizing
    # story), micro-batched and pool-dispatched like the credit path.
    dense_engines: list = []
    dense_batchers: list = []
    if cfg.dense_model_dir:
        from .dense import DenseEngine, DenseModel

        dm = DenseModel.load(cfg.dense_model_dir)
        device = cfg.resolve_device()
        if device == "cpu":
            dense_engines = [
                DenseEngine(dm, device="cpu") for _ in range(max(cfg.n_gpus, 1))
            ]
        else:
            import torch

            n_dev = max(torch.cuda.device_count(), 1)
            n = cfg.n_gpus or n_dev
            dense_engines = [
                DenseEngine(dm, device="cuda", device_index=i % n_dev)
                for i in range(n)
            ]

        def _dense_scorer(e):
            def run(x, _nums):
                return e.score_arrays(x)

            return run

        dense_batchers = [
            MicroBatcher(
                _dense_scorer(e),
                max_rows=cfg.max_batch_rows,
                max_wait_us=cfg.batch_wait_us,
            )
            for e in dense_engines
        ]
        for b in dense_batchers:
            await b.start()
    state["dense_engines"] = dense_engines
    state["dense_batchers"] = dense_batchers
    state["dense_pool"] = ReplicaPool(len(dense_engines)) if dense_engines else None
    state["engines"] = engines
    state["batchers"] = batchers
    state["pool"] = ReplicaPool(len(engines))
    state["drift_sync"] = drift_sync
    state["metrics"] = Metrics()
    state["cfg"] = cfg

    # probation loop: periodically re-probe dead replicas and re-admit the
    # ones that recover (reads state each tick, so /admin/reload swaps are
    # picked up)
    import asyncio
    import logging as _logging

    async def _revival_loop():
        log = _logging.getLogger("creditcore.replicas")
        loop = asyncio.get_running_loop()
        while True:
            await asyncio.sleep(max(cfg.replica_probe_period_s, 0.25))
            pool = state.get("pool")
            live_engines = state.get("engines")
            if pool is not None and live_engines:
                try:
                    await probe_revive(pool, live_engines, log)
                except Exception:
                    pass
            # dense replicas get the same probation treatment
            dpool = state.get("dense_pool")
            dengines = state.get("dense_engines")
            if dpool is not None and dengines:
                for i in dpool.dead_indices():
                    x = np.zeros((1, dengines[i].model.n_features), np.float32)
                    try:
                        await loop.run_in_executor(
                            None, lambda e=dengines[i]: e.score_arrays(x, False)
                        )
                        dpool.revive(i)
                        log.info("dense replica %d revived", i)
                    except Exception:
                        pass

    revival_task = asyncio.create_task(_revival_loop())
    yield
    revival_task.cancel()
    # use the *current* objects — /admin/reload may have swapped them
    for b in state.get("batchers", batchers):
        await b.close()
    for b in state.get("dense_batchers") or []:
        await b.close()
    if cfg.drift_state_path:
        state.get("drift_sync", drift_sync).save_state(cfg.drift_state_path)
    state.clear()


def create_app(cfg: ServeConfig | None = None) -> FastAPI:
    cfg = cfg or ServeConfig()
    app = FastAPI(title=cfg.service_name, docs_url="/", lifespan=lifespan)
    app.state.cfg = cfg

    def _encode_body(body: bytes, engine: ScoringEngine) -> tuple:
        """Native JSON fast path; pydantic fallback keeps the reference's
        422/validation semantics for anything the strict parser rejects
        (e.g. nulls, lax-coercible values like numeric strings)."""
        from .ops import gpu

        vocabs = engine.packed.vocabs
        if gpu.available():
            try:
                return engine.encode_json_body(body)
            except ValueError:
                pass  # fall through to full validation
        from pydantic import TypeAdapter, ValidationError

        try:
            data = TypeAdapter(list[LoanApplicant]).validate_json(body)
        except ValidationError as e:
            raise HTTPException(status_code=422, detail=e.errors(include_url=False))
        return encode_batch([r.__dict__ for r in data], vocabs)

    async def _predict_impl(body: bytes) -> dict:
        cfg: ServeConfig = state["cfg"]
        metrics: Metrics = state["metrics"]
        request_id = uuid.uuid4().hex

        if cfg.log_inference_data:
            reqlog.log_inference_data(
                cfg.service_name, request_id, body.decode("utf-8", "replace")
            )

        engines = state["engines"]
        batchers = state["batchers"]
        pool: ReplicaPool = state["pool"]
        try:
            codes, nums = _encode_body(body, engines[0])
        except HTTPException:
            raise
        except (ValueError, TypeError) as e:
            raise HTTPException(status_code=422, detail=f"bad record: {e}")
        if len(codes) == 0:
            raise HTTPException(status_code=400, detail="empty request batch")

        # failover: a replica failing mid-flight must not fail the request
        # while healthy replicas exist (the K8s-Service-retries analog);
        # the attempt budget guarantees a healthy replica is reached if any
        t0 = time.perf_counter()
        out = None
        last_exc: Exception | None = None
        for _ in range(pool.attempt_budget):
            try:
                idx = pool.pick()
            except RuntimeError:
                metrics.observe_error()
                raise HTTPException(status_code=503, detail="no healthy replicas")
            try:
                out = await batchers[idx].submit(codes, nums)
                pool.report_ok(idx)
                break
            except Exception as e:
                metrics.observe_error()
                pool.report_fail(idx)
                last_exc = e
        if out is None:
            raise HTTPException(status_code=500, detail=f"scoring failed: {last_exc}")
        latency_ms = (time.perf_counter() - t0) * 1e3

        if "response_bytes" in out:  # solo-flush wire-out fast path
            rb = out["response_bytes"]
            metrics.observe_request(out["rows"], latency_ms)
            if cfg.log_responses:
                reqlog.log_model_output_raw(
                    cfg.service_name,
                    request_id,
                    rb.decode("utf-8", "replace"),
                    latency_ms=latency_ms,
                    rows=out["rows"],
                    device=f"{engines[idx].device}:{engines[idx].device_index}",
                )
            return rb

        # merged-flush responses: C serializer (GIL released) beats Python
        # json.dumps by ~10x on 1024-row responses
        from .ops import gpu as _gpu

        if _gpu.available():
            response = bytes(
                _gpu.ext().build_response_json_arrays(
                    np.ascontiguousarray(out["predictions"], dtype=np.float64),
                    np.ascontiguousarray(out["outliers"], dtype=np.float64),
                    np.ascontiguousarray(out["p_vals"], dtype=np.float64),
                    FEATURES,
                )
            )
            log_payload = response.decode("utf-8", "replace")
        else:
            one_minus = (
                np.float32(1.0) - np.asarray(out["p_vals"], dtype=np.float32)
            ).astype(np.float64)
            response = {
                "predictions": np.asarray(out["predictions"]).tolist(),
                "outliers": np.asarray(out["outliers"]).tolist(),
                "feature_drift_batch": dict(zip(FEATURES, one_minus.tolist())),
            }
            log_payload = None
        metrics.observe_request(len(codes), latency_ms)
        if cfg.log_responses:
            if log_payload is not None:
                reqlog.log_model_output_raw(
                    cfg.service_name,
                    request_id,
                    log_payload,
                    latency_ms=latency_ms,
                    rows=len(codes),
                    device=f"{engines[idx].device}:{engines[idx].device_index}",
                )
            else:
                reqlog.log_model_output(
                    cfg.service_name,
                    request_id,
                    response,
                    latency_ms=latency_ms,
                    rows=len(codes),
                    device=f"{engines[idx].device}:{engines[idx].device_index}",
                )
        return response

    # The endpoints take the raw body (native JSON fast path) but keep the
    # reference's documented request/response schema in OpenAPI.
    _openapi_extra = {
        "requestBody": {
            "required": True,
            "content": {
                "application/json": {
                    "schema": {
                        "type": "array",
                        "items": LoanApplicant.model_json_schema(),
                    }
                }
            },
        }
    }

    def _json_response(payload) -> Response:
        # shape is correct by construction; skip response-model revalidation
        if isinstance(payload, (bytes, str)):
            return Response(payload, media_type="application/json")
        return Response(json.dumps(payload), media_type="application/json")

    @app.post("/predict", response_model=ModelOutput, openapi_extra=_openapi_extra)
    async def predict(request: Request):
        """Reference endpoint (app/main.py:42)."""
        return _json_response(await _predict_impl(await request.body()))

    @app.post("/score", response_model=ModelOutput, openapi_extra=_openapi_extra)
    async def score(request: Request):
        """Alias — BASELINE.json names the endpoint /score."""
        return _json_response(await _predict_impl(await request.body()))

    @app.get("/healthz")
    async def healthz(deep: bool = False):
        pool: ReplicaPool = state["pool"]
        any_alive = any(pool.alive)
        body = {
            "status": "ok" if any_alive else "dead",
            "engines": [
                {"device": e.device, "index": e.device_index, "alive": pool.alive[i]}
                for i, e in enumerate(state["engines"])
            ],
        }
        if deep and any_alive:
            # active probe: score the schema-default record on every live
            # replica (SURVEY.md §5.3 — HIP error surfacing on demand)
            import asyncio

            from .schema import LoanApplicant

            codes, nums = encode_batch(
                [LoanApplicant().__dict__], state["engines"][0].packed.vocabs
            )
            loop = asyncio.get_running_loop()
            for i, e in enumerate(state["engines"]):
                if not pool.alive[i]:
                    continue
                try:
                    await loop.run_in_executor(
                        None, lambda e=e: e.score_arrays(codes, nums, False)
                    )
                    body["engines"][i]["probe"] = "ok"
                except Exception as exc:
                    pool.report_fail(i)
                    body["engines"][i]["probe"] = f"failed: {exc}"
                    body["status"] = "degraded"
        return body

    @app.get("/metrics")
    async def metrics_endpoint(format: str = "json"):
        if format == "prometheus":
            return Response(
                state["metrics"].prometheus(),
                media_type="text/plain; version=0.0.4",
            )
        return state["metrics"].snapshot()

    def _parse_dense_body(body: bytes, content_type: str, n_features: int):
        from .dense import parse_dense_body

        try:
            return parse_dense_body(body, content_type, n_features)
        except ValueError as e:
            raise HTTPException(status_code=422, detail=str(e))

    @app.post("/predict_dense")
    async def predict_dense(request: Request):
        """Dense wide-tabular scoring (BASELINE config 5), first-class:
        micro-batched, dispatched round-robin across the per-GPU dense
        replicas (each holds the full sorted drift reference in its own
        HBM), with the same failover semantics as /score."""
        engines = state.get("dense_engines") or []
        if not engines:
            raise HTTPException(status_code=404, detail="no dense model configured")
        body = await request.body()
        x = _parse_dense_body(
            body, request.headers.get("content-type", ""), engines[0].model.n_features
        )
        pool: ReplicaPool = state["dense_pool"]
        batchers = state["dense_batchers"]
        t0 = time.perf_counter()
        out = None
        last_exc = None
        for _ in range(pool.attempt_budget):
            try:
                idx = pool.pick()
            except RuntimeError:
                state["metrics"].observe_error()
                raise HTTPException(status_code=503, detail="no healthy dense replicas")
            try:
                out = await batchers[idx].submit(x, None)
                pool.report_ok(idx)
                break
            except Exception as e:
                state["metrics"].observe_error()
                pool.report_fail(idx)
                last_exc = e
        if out is None:
            raise HTTPException(status_code=500, detail=f"scoring failed: {last_exc}")
        latency_ms = (time.perf_counter() - t0) * 1e3
        state["metrics"].observe_request(len(x), latency_ms)
        return _json_response(
            {
                "predictions": np.asarray(out["predictions"]).tolist(),
                "outliers": np.asarray(out["outliers"]).tolist(),
                "feature_drift_batch": (
                    np.float32(1.0) - np.asarray(out["p_vals"], dtype=np.float32)
                ).astype(np.float64).tolist(),
            }
        )

    @app.post("/admin/reload")
    async def reload_model(request: Request):
        """Hot-swap the served model without a restart — the in-process
        equivalent of the reference's rolling redeploy (its only path to a
        new model version was rebuilding + redeploying every container).
        Body: {"model_uri": "models:/<name>/<version|latest>" | <dir>}.
        Builds the new engines first, then swaps atomically; in-flight
        requests finish on the old engines."""
        from .parallel import DriftSync

        cfg: ServeConfig = state["cfg"]
        if not admin_authorized(
            cfg,
            request.client.host if request.client else None,
            request.headers.get("authorization"),
            request.headers.get("x-admin-token"),
        ):
            raise HTTPException(
                status_code=403,
                detail="admin endpoint: loopback client or admin token required",
            )
        try:
            payload = json.loads(await request.body() or b"{}")
        except ValueError:
            raise HTTPException(status_code=422, detail="body must be JSON")
        uri = payload.get("model_uri") or cfg.model_directory
        import copy
        import dataclasses

        new_cfg = dataclasses.replace(copy.copy(cfg), model_directory=uri)
        try:
            engines = _build_engines(new_cfg)
        except Exception as e:
            raise HTTPException(status_code=422, detail=f"cannot load {uri!r}: {e}")

        drift_sync = DriftSync(engines[0].packed, device="cpu")
        batchers = _make_batchers(new_cfg, engines, drift_sync)
        for b in batchers:
            await b.start()
        old_batchers = state["batchers"]
        state["engines"] = engines
        state["batchers"] = batchers
        state["pool"] = ReplicaPool(len(engines))
        state["drift_sync"] = drift_sync  # new model ⇒ new drift reference
        state["cfg"] = new_cfg
        for b in old_batchers:  # drain + stop after the swap
            await b.close()
        return {"status": "reloaded", "model_uri": uri, "engines": len(engines)}

    @app.get("/drift")
    async def drift_endpoint():
        """Node-global drift state (the cross-replica monitor the reference's
        per-pod-isolated drift lacks — SURVEY.md §2.4)."""
        ds = state["drift_sync"]
        ds.allreduce()
        return ds.snapshot()

    return app


def app_from_env() -> FastAPI:
    """uvicorn factory target for multi-worker serving (config from env)."""
    return create_app(ServeConfig())


def main(argv: list[str] | None = None):
    import logging

    import uvicorn

    logging.basicConfig(level=logging.INFO)
    cfg = ServeConfig.from_args(argv)
    if cfg.raw_http:
        # throughput frontend: minimal asyncio HTTP/1.1, native JSON path.
        # No create_app here: a multi-worker parent must not load engines
        # (each worker owns its own GPU + engine).
        from .rawserve import main as raw_main

        raw_main(cfg)
        return
    # access_log off: the app writes its own two JSON lines per request
    if cfg.workers > 1:
        # process-level scaling: each worker owns its engines and GIL
        # (SO_REUSEPORT fan-in by uvicorn). Config rides the env.
        import dataclasses

        for f in dataclasses.fields(cfg):
            os.environ[f"CREDITCORE_{f.name.upper()}"] = str(getattr(cfg, f.name))
        uvicorn.run(
            "creditcore.serve:app_from_env",
            factory=True,
            host=cfg.host,
            port=cfg.port,
            log_level="info",
            access_log=False,
            workers=cfg.workers,
        )
    else:
        uvicorn.run(
            create_app(cfg), host=cfg.host, port=cfg.port,
            log_level="info", access_log=False,
        )


if __name__ == "__main__":
    main()
