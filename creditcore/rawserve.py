"""Raw asyncio HTTP/1.1 frontend — the serving-layer fast path.

uvicorn in this stack parses requests with pure-Python h11, which caps the
wire layer far below the engine (~0.65 MB request bodies). This frontend
speaks just enough HTTP/1.1 for the scoring contract — request line,
headers, Content-Length bodies, keep-alive — and hands the body bytes
straight to the native JSON fast path. The FastAPI app (`creditcore.serve`)
remains the contract-complete default (docs, OpenAPI, pydantic fallback
validation); enable this one with `creditcore serve --raw-http` where
throughput matters.

Endpoints: POST /score, /predict (wire-format fast path with pydantic
fallback semantics preserved via the same `_encode` logic),
POST /predict_dense (binary bulk or JSON rows), POST /admin/reload
(loopback/token gated), GET /healthz[?deep], /metrics, /drift. Bodies may
use Content-Length or chunked transfer encoding.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid

import numpy as np

from .batching import MicroBatcher
from .config import ServeConfig
from .serve import ReplicaPool
from .schema import FEATURES, LoanApplicant
from .utils import logging as reqlog
from .utils.metrics import Metrics


class RawScoreServer:
    def __init__(self, cfg: ServeConfig):
        self.cfg = cfg
        self.metrics = Metrics()
        self.engines = []
        self.batchers = []
        self.drift_sync = None
        self.pool = None
        self.dense_engines = []
        self.dense_batchers = []
        self.dense_pool = None

    def _wire_batchers(self, engines) -> list[MicroBatcher]:
        cfg = self.cfg

        def scorer(e):
            def run(codes, nums):
                out = e.score_arrays(codes, nums)
                self._fold_drift(out, nums)
                return out

            return run

        def scorer_single(e):
            if e.device != "cuda":
                return None

            def run(codes, nums):
                out = e.score_encoded_bytes(codes, nums)
                self._fold_drift(out, nums)
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

    # ------------------------------------------------------------ lifecycle
    async def start(self):
        from .parallel import DriftSync
        from .serve import _build_engines

        cfg = self.cfg
        self.engines = _build_engines(cfg)
        self.drift_sync = DriftSync(self.engines[0].packed, device="cpu")
        if cfg.drift_state_path and cfg.workers <= 1:
            # single-process serving only: SO_REUSEPORT workers already share
            # live state via the tmpfs publish dir, and one shared file would
            # lose all but the last worker's shard on shutdown
            self.drift_sync.load_state(cfg.drift_state_path)

        self.batchers = self._wire_batchers(self.engines)
        for b in self.batchers:
            await b.start()
        self.pool = ReplicaPool(len(self.engines))

        # dense wide-tabular family (BASELINE config 5) on the raw
        # frontend: same replica/batcher/failover wiring as serve.py
        self.dense_engines = []
        self.dense_batchers = []
        self.dense_pool = None
        if cfg.dense_model_dir:
            from .dense import DenseEngine, DenseModel

            dm = DenseModel.load(cfg.dense_model_dir)
            device = cfg.resolve_device()
            if device == "cpu":
                self.dense_engines = [
                    DenseEngine(dm, device="cpu") for _ in range(max(cfg.n_gpus, 1))
                ]
            else:
                import torch

                n_dev = max(torch.cuda.device_count(), 1)
                n = cfg.n_gpus or n_dev
                self.dense_engines = [
                    DenseEngine(dm, device="cuda", device_index=i % n_dev)
                    for i in range(n)
                ]

            def _dscorer(e):
                def run(x, _nums):
                    return e.score_arrays(x)

                return run

            self.dense_batchers = [
                MicroBatcher(
                    _dscorer(e),
                    max_rows=cfg.max_batch_rows,
                    max_wait_us=cfg.batch_wait_us,
                )
                for e in self.dense_engines
            ]
            for b in self.dense_batchers:
                await b.start()
            self.dense_pool = ReplicaPool(len(self.dense_engines))

        # probation loop: re-probe dead replicas, re-admit on success
        async def _revival_loop():
            from .serve import probe_revive

            while True:
                await asyncio.sleep(max(cfg.replica_probe_period_s, 0.25))
                try:
                    await probe_revive(self.pool, self.engines)
                except Exception:
                    pass

        self._revival_task = asyncio.ensure_future(_revival_loop())
        self._server = await asyncio.start_server(
            self._handle, cfg.host, cfg.port, backlog=512,
            reuse_port=(cfg.workers > 1)
            or __import__("os").environ.get("CREDITCORE_RAW_REUSE_PORT") == "1",
        )
        return self._server

    async def close(self):
        if getattr(self, "_revival_task", None) is not None:
            self._revival_task.cancel()
        self._server.close()
        await self._server.wait_closed()
        for b in self.batchers:
            await b.close()
        for b in getattr(self, "dense_batchers", []) or []:
            await b.close()
        if self.cfg.drift_state_path and self.cfg.workers <= 1:
            self.drift_sync.save_state(self.cfg.drift_state_path)

    @property
    def _drift_dir(self) -> str:
        import os
        import tempfile

        base = "/dev/shm" if os.path.isdir("/dev/shm") else tempfile.gettempdir()
        return os.path.join(base, f"creditcore_drift_{self.cfg.port}")

    def _fold_drift(self, out, nums):
        if "cat_hist" in out:
            self.drift_sync.accumulate(out["cat_hist"], nums)
            if self.drift_sync.batches % max(self.cfg.drift_sync_period, 1) == 0:
                self.drift_sync.allreduce()
                self.drift_sync.publish(self._drift_dir)
                self.metrics.observe_drift_sync()

    # ------------------------------------------------------------ handlers
    async def _score(self, body: bytes) -> tuple[int, bytes]:
        cfg = self.cfg
        request_id = uuid.uuid4().hex
        if cfg.log_inference_data:
            reqlog.log_inference_data(
                cfg.service_name, request_id, body.decode("utf-8", "replace")
            )
        engine = self.engines[0]
        loop = asyncio.get_running_loop()
        try:
            codes, nums = await loop.run_in_executor(
                None, engine.encode_json_body, body
            )
        except (ValueError, TypeError):
            # full pydantic validation for reference 422/coercion semantics
            from pydantic import TypeAdapter, ValidationError

            from .pack import encode_batch

            try:
                data = TypeAdapter(list[LoanApplicant]).validate_json(body)
            except ValidationError as e:
                return 422, json.dumps({"detail": e.errors(include_url=False)}).encode()
            codes, nums = encode_batch([r.__dict__ for r in data], engine.packed.vocabs)
        if len(codes) == 0:
            return 400, b'{"detail": "empty request batch"}'

        # failover across replicas: one failing GPU must not fail requests
        # while healthy replicas remain (see serve._predict_impl)
        t0 = time.perf_counter()
        out = None
        last_exc = None
        for _ in range(self.pool.attempt_budget):
            try:
                idx = self.pool.pick()
            except RuntimeError:
                self.metrics.observe_error()
                return 503, b'{"detail": "no healthy replicas"}'
            engine = self.engines[idx]
            try:
                out = await self.batchers[idx].submit(codes, nums)
                self.pool.report_ok(idx)
                break
            except Exception as e:
                self.metrics.observe_error()
                self.pool.report_fail(idx)
                last_exc = e
        if out is None:
            return 500, json.dumps({"detail": f"scoring failed: {last_exc}"}).encode()
        latency_ms = (time.perf_counter() - t0) * 1e3
        self.metrics.observe_request(len(codes), latency_ms)

        if "response_bytes" in out:
            rb = out["response_bytes"]
            if cfg.log_responses:
                reqlog.log_model_output_raw(
                    cfg.service_name, request_id, rb.decode("utf-8", "replace"),
                    latency_ms=latency_ms, rows=out["rows"],
                    device=f"{engine.device}:{engine.device_index}",
                )
            return 200, rb
        # merged-flush responses: C serializer (GIL released) — Python
        # json.dumps of a 1024-row response costs ~150+ µs ON the event
        # loop, which capped the multi-worker HTTP throughput
        from .ops import gpu as _gpu

        if _gpu.available():
            payload = bytes(
                _gpu.ext().build_response_json_arrays(
                    np.ascontiguousarray(out["predictions"], dtype=np.float64),
                    np.ascontiguousarray(out["outliers"], dtype=np.float64),
                    np.ascontiguousarray(out["p_vals"], dtype=np.float64),
                    FEATURES,
                )
            )
        else:
            one_minus = (
                np.float32(1.0) - np.asarray(out["p_vals"], dtype=np.float32)
            ).astype(np.float64)
            response = {
                "predictions": np.asarray(out["predictions"]).tolist(),
                "outliers": np.asarray(out["outliers"]).tolist(),
                "feature_drift_batch": dict(zip(FEATURES, one_minus.tolist())),
            }
            payload = json.dumps(response).encode()
        if cfg.log_responses:
            reqlog.log_model_output_raw(
                cfg.service_name, request_id, payload.decode("utf-8", "replace"),
                latency_ms=latency_ms, rows=len(codes),
                device=f"{engine.device}:{engine.device_index}",
            )
        return 200, payload

    async def _score_dense(self, body: bytes, ctype: bytes) -> tuple[int, bytes]:
        """Dense wide-tabular scoring on the raw frontend (binary bulk or
        JSON rows bodies; micro-batched, replica failover)."""
        if not self.dense_engines:
            return 404, b'{"detail": "no dense model configured"}'
        from .dense import parse_dense_body

        try:
            x = parse_dense_body(
                body, ctype.decode("ascii", "replace"),
                self.dense_engines[0].model.n_features,
            )
        except ValueError as e:
            return 422, json.dumps({"detail": str(e)}).encode()
        t0 = time.perf_counter()
        out = None
        last_exc = None
        for _ in range(self.dense_pool.attempt_budget):
            try:
                idx = self.dense_pool.pick()
            except RuntimeError:
                self.metrics.observe_error()
                return 503, b'{"detail": "no healthy dense replicas"}'
            try:
                out = await self.dense_batchers[idx].submit(x, None)
                self.dense_pool.report_ok(idx)
                break
            except Exception as e:
                self.metrics.observe_error()
                self.dense_pool.report_fail(idx)
                last_exc = e
        if out is None:
            return 500, json.dumps({"detail": f"scoring failed: {last_exc}"}).encode()
        self.metrics.observe_request(len(x), (time.perf_counter() - t0) * 1e3)
        one_minus = (
            np.float32(1.0) - np.asarray(out["p_vals"], dtype=np.float32)
        ).astype(np.float64)
        return 200, json.dumps(
            {
                "predictions": np.asarray(out["predictions"]).tolist(),
                "outliers": np.asarray(out["outliers"]).tolist(),
                "feature_drift_batch": one_minus.tolist(),
            }
        ).encode()

    async def _reload(self, body: bytes) -> tuple[int, bytes]:
        """Hot model swap (per worker): build new engines first, then swap
        atomically; in-flight requests finish on the old ones. Body:
        {"model_uri": "models:/<name>/<version|latest>" | <dir>}."""
        import dataclasses

        from .parallel import DriftSync
        from .serve import _build_engines

        try:
            payload = json.loads(body or b"{}")
        except ValueError:
            return 422, b'{"detail": "body must be JSON"}'
        uri = payload.get("model_uri") or self.cfg.model_directory
        new_cfg = dataclasses.replace(self.cfg, model_directory=uri)
        try:
            engines = _build_engines(new_cfg)
        except Exception as e:
            return 422, json.dumps({"detail": f"cannot load {uri!r}: {e}"}).encode()
        drift_sync = DriftSync(engines[0].packed, device="cpu")
        batchers = self._wire_batchers(engines)
        for b in batchers:
            await b.start()
        old = self.batchers
        self.engines = engines
        self.batchers = batchers
        self.pool = ReplicaPool(len(engines))
        self.drift_sync = drift_sync  # new model => new drift reference
        self.cfg = new_cfg
        for b in old:
            await b.close()
        return 200, json.dumps(
            {"status": "reloaded", "model_uri": uri, "engines": len(engines)}
        ).encode()

    async def _get(self, path: bytes) -> tuple[int, bytes]:
        path, _, query = path.partition(b"?")
        if path == b"/metrics" and b"format=prometheus" in query:
            return 200, self.metrics.prometheus().encode(), b"text/plain; version=0.0.4"
        if path == b"/healthz":
            any_alive = any(self.pool.alive)
            body = {
                "status": "ok" if any_alive else "dead",
                "engines": len(self.engines),
                "alive": self.pool.alive,
            }
            if b"deep" in query and any_alive:
                # active probe parity with the FastAPI frontend: score the
                # schema-default record on every live replica
                from .pack import encode_batch
                from .schema import LoanApplicant

                codes, nums = encode_batch(
                    [LoanApplicant().__dict__], self.engines[0].packed.vocabs
                )
                loop = asyncio.get_running_loop()
                probes = []
                for i, e in enumerate(self.engines):
                    if not self.pool.alive[i]:
                        probes.append("dead")
                        continue
                    try:
                        await loop.run_in_executor(
                            None, lambda e=e: e.score_arrays(codes, nums, False)
                        )
                        probes.append("ok")
                    except Exception as exc:
                        self.pool.report_fail(i)
                        probes.append(f"failed: {exc}")
                        body["status"] = "degraded"
                body["probe"] = probes
            return 200, json.dumps(body).encode()
        if path == b"/metrics":
            return 200, json.dumps(self.metrics.snapshot()).encode()
        if path == b"/drift":
            # node-global: this worker's live state + every worker's
            # published snapshot (SO_REUSEPORT workers share no memory)
            self.drift_sync.merge_published(self._drift_dir)
            return 200, json.dumps(self.drift_sync.snapshot()).encode()
        return 404, b'{"detail": "not found"}'

    # ------------------------------------------------------------ HTTP/1.1
    async def _read_chunked(self, reader):
        """Decode a Transfer-Encoding: chunked body (RFC 9112 §7.1) so any
        standard HTTP client works, matching the reference's
        accept-anything serving contract (reference app/main.py:42).
        Returns (body, None) or (None, (status, payload))."""
        parts = []
        total = 0
        cap = self.cfg.max_body_bytes
        while True:
            size_line = await reader.readline()
            if not size_line:
                return None, (400, b'{"detail": "truncated chunked body"}')
            # chunk extensions after ';' are ignored per spec
            size_tok = size_line.split(b";", 1)[0].strip()
            try:
                size = int(size_tok, 16)
            except ValueError:
                return None, (400, b'{"detail": "invalid chunk size"}')
            if size < 0:
                return None, (400, b'{"detail": "invalid chunk size"}')
            if size == 0:
                break
            total += size
            if total > cap:
                return None, (413, b'{"detail": "body too large"}')
            parts.append(await reader.readexactly(size))
            crlf = await reader.readexactly(2)
            if crlf != b"\r\n":
                return None, (400, b'{"detail": "malformed chunk"}')
        # trailer section: read until the blank line
        while True:
            t = await reader.readline()
            if t in (b"\r\n", b"\n", b""):
                break
        return b"".join(parts), None

    async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        try:
            while True:
                # one awaited read for request line + all headers (readline
                # per header cost ~6 event-loop awaits per request)
                try:
                    head = await reader.readuntil(b"\r\n\r\n")
                except asyncio.LimitOverrunError:
                    await self._respond(writer, 431, b'{"detail": "headers too large"}')
                    return
                except asyncio.IncompleteReadError as e:
                    if e.partial:  # garbage without a header terminator
                        break
                    break  # clean EOF between keep-alive requests
                lines = head.split(b"\r\n")
                try:
                    method, path, _ = lines[0].split(b" ", 2)
                except ValueError:
                    break
                clen = 0
                keep_alive = True
                chunked = False
                admin_hdrs = {}
                ctype_in = b""
                for h in lines[1:]:
                    if not h:
                        continue
                    k, _, v = h.partition(b":")
                    lk = k.lower()
                    if lk == b"content-length":
                        v = v.strip()
                        # digits only: negative/malformed values must be a
                        # 400, not an unhandled ValueError (and a negative
                        # length would sail past the max_body_bytes check)
                        if not v.isdigit():
                            await self._respond(
                                writer, 400, b'{"detail": "invalid Content-Length"}'
                            )
                            return
                        clen = int(v)
                    elif lk == b"connection" and b"close" in v.lower():
                        keep_alive = False
                    elif lk in (b"authorization", b"x-admin-token"):
                        admin_hdrs[lk] = v.strip()
                    elif lk == b"content-type":
                        ctype_in = v.strip()
                    elif lk == b"transfer-encoding":
                        te = v.strip().lower()
                        if te == b"chunked":
                            chunked = True
                        else:
                            # compressed transfer codings are out of contract
                            await self._respond(
                                writer, 501, b'{"detail": "unsupported transfer-encoding"}'
                            )
                            return
                if clen > self.cfg.max_body_bytes:
                    # refuse before reading: an unbounded Content-Length must
                    # not drive readexactly into allocating it
                    await self._respond(writer, 413, b'{"detail": "body too large"}')
                    return
                if chunked:
                    body, err = await self._read_chunked(reader)
                    if err is not None:
                        await self._respond(writer, err[0], err[1])
                        return
                else:
                    body = await reader.readexactly(clen) if clen else b""
                ctype = b"application/json"
                ppath = path.partition(b"?")[0]
                if method == b"POST" and ppath in (b"/score", b"/predict"):
                    status, payload = await self._score(body)
                elif method == b"POST" and ppath == b"/predict_dense":
                    status, payload = await self._score_dense(body, ctype_in)
                elif method == b"POST" and ppath == b"/admin/reload":
                    from .serve import admin_authorized

                    peer = writer.get_extra_info("peername")
                    auth = admin_hdrs.get(b"authorization")
                    xtok = admin_hdrs.get(b"x-admin-token")
                    if admin_authorized(
                        self.cfg,
                        peer[0] if peer else None,
                        auth.decode("ascii", "replace") if auth else None,
                        xtok.decode("ascii", "replace") if xtok else None,
                    ):
                        status, payload = await self._reload(body)
                    else:
                        status, payload = 403, (
                            b'{"detail": "admin endpoint: loopback client'
                            b' or admin token required"}'
                        )
                elif method == b"GET":
                    res = await self._get(path)
                    status, payload = res[0], res[1]
                    if len(res) > 2:
                        ctype = res[2]
                else:
                    status, payload = 405, b'{"detail": "method not allowed"}'
                await self._respond(writer, status, payload, ctype)
                if not keep_alive:
                    break
        except (asyncio.IncompleteReadError, ConnectionResetError):
            pass
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:
                pass

    _REASONS = {200: b"OK", 400: b"Bad Request", 403: b"Forbidden", 404: b"Not Found",
                405: b"Method Not Allowed", 411: b"Length Required",
                413: b"Payload Too Large", 422: b"Unprocessable Entity",
                431: b"Request Header Fields Too Large", 501: b"Not Implemented",
                500: b"Internal Server Error", 503: b"Service Unavailable"}

    async def _respond(self, writer, status: int, payload: bytes,
                       ctype: bytes = b"application/json"):
        head = b"HTTP/1.1 %d %s\r\nContent-Type: %s\r\nContent-Length: %d\r\n\r\n" % (
            status, self._REASONS.get(status, b"OK"), ctype, len(payload),
        )
        writer.write(head + payload)
        await writer.drain()


async def serve_raw(cfg: ServeConfig):
    import signal as _signal

    server = RawScoreServer(cfg)
    await server.start()
    print(f"[rawserve] listening on {cfg.host}:{cfg.port}", flush=True)
    # graceful SIGTERM/SIGINT (K8s pod stop): close the listener, then run
    # the shutdown hooks — drift-state persistence happens in close()
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (_signal.SIGTERM, _signal.SIGINT):
        try:
            loop.add_signal_handler(sig, stop.set)
        except (ValueError, NotImplementedError):  # non-main thread
            pass
    serve_task = asyncio.create_task(server._server.serve_forever())
    await stop.wait()
    serve_task.cancel()
    try:
        await serve_task
    except asyncio.CancelledError:
        pass
    await server.close()
    print("[rawserve] shut down cleanly", flush=True)


def main(cfg: ServeConfig):
    if cfg.workers > 1:
        _spawn_workers(cfg)
        return
    asyncio.run(serve_raw(cfg))


def _spawn_workers(cfg: ServeConfig):
    """Production multi-GPU shape: N worker processes share the port via
    SO_REUSEPORT (the kernel is the request queue), each pinned to one GPU
    through HIP_VISIBLE_DEVICES — the reference's K8s-replica scaling
    collapsed into one node."""
    import dataclasses
    import os
    import signal
    import subprocess
    import sys

    env_base = dict(os.environ)
    for f in dataclasses.fields(cfg):
        env_base[f"CREDITCORE_{f.name.upper()}"] = str(getattr(cfg, f.name))
    env_base["CREDITCORE_WORKERS"] = "1"  # workers serve single-process
    # workers share live drift via the tmpfs publish dir; a shared state
    # file would keep only the last worker's shard — disable per-worker
    env_base["CREDITCORE_DRIFT_STATE_PATH"] = ""
    try:
        import torch

        n_dev = torch.cuda.device_count()
    except Exception:
        n_dev = 0

    procs = []
    for w in range(cfg.workers):
        env = dict(env_base)
        if n_dev > 0:
            env["HIP_VISIBLE_DEVICES"] = str(w % n_dev)
            env["CREDITCORE_N_GPUS"] = "1"
        env["CREDITCORE_RAW_HTTP"] = "1"
        env["CREDITCORE_RAW_REUSE_PORT"] = "1"
        def _die_with_parent():  # orphan guard (Linux PR_SET_PDEATHSIG)
            import ctypes

            try:
                ctypes.CDLL("libc.so.6").prctl(1, signal.SIGTERM)
            except OSError:  # pragma: no cover
                pass

        procs.append(subprocess.Popen(
            [sys.executable, "-m", "creditcore", "serve"], env=env,
            preexec_fn=_die_with_parent,
        ))
    print(f"[rawserve] {cfg.workers} workers on port {cfg.port} "
          f"(SO_REUSEPORT, one GPU each)", flush=True)

    def shutdown(*_):
        for pr in procs:
            pr.terminate()

    signal.signal(signal.SIGTERM, shutdown)
    signal.signal(signal.SIGINT, shutdown)
    for pr in procs:
        pr.wait()
