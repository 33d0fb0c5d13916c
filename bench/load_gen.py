"""HTTP load generator for the serving stack.

Closed-loop workers POST wire-format /score requests against a running
server and report whole-service requests/sec + latency percentiles —
the serving-layer view of the headline metric (bench.py measures the
engine path; this measures through uvicorn/HTTP).

    python -m creditcore serve --port 5000 &
    python bench/load_gen.py --url http://127.0.0.1:5000 \
        --rows 1024 --concurrency 8 --duration 30
"""

from __future__ import annotations

import argparse
import asyncio
import json
import time

import numpy as np

# runnable from the repo root or anywhere: put the repo on sys.path
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


async def worker(client, url, body, stop_at, stats, endpoint="/score",
                 ctype="application/json"):
    while time.perf_counter() < stop_at:
        t0 = time.perf_counter()
        r = await client.post(f"{url}{endpoint}", content=body,
                              headers={"content-type": ctype})
        dt = (time.perf_counter() - t0) * 1e3
        if r.status_code == 200:
            stats["lat"].append(dt)
        else:
            stats["errors"] += 1


async def raw_worker(host, port, body, stop_at, stats, endpoint=b"/score",
                     ctype=b"application/json"):
    """Minimal HTTP/1.1 keep-alive client on a raw socket: the httpx stack
    costs ~1.4 ms/request of client CPU, which caps what the *server* can
    be measured at; this one is ~50 µs/request."""
    reader, writer = await asyncio.open_connection(host, port)
    req = (
        b"POST " + endpoint + b" HTTP/1.1\r\nHost: l\r\nContent-Type: " + ctype
        + b"\r\nContent-Length: " + str(len(body)).encode() + b"\r\n\r\n" + body
    )
    try:
        while time.perf_counter() < stop_at:
            t0 = time.perf_counter()
            writer.write(req)
            await writer.drain()
            head = await reader.readuntil(b"\r\n\r\n")
            clen = 0
            for line in head.split(b"\r\n"):
                if line[:15].lower() == b"content-length:":
                    clen = int(line[15:])
                    break
            payload = await reader.readexactly(clen)
            dt = (time.perf_counter() - t0) * 1e3
            if head.startswith(b"HTTP/1.1 200") and payload:
                stats["lat"].append(dt)
            else:
                stats["errors"] += 1
    finally:
        writer.close()


async def run(args):
    import httpx

    from creditcore.data import make_request_batch

    if getattr(args, "dense_features", 0):
        # binary dense body for /predict_dense (BASELINE config 5 served)
        import struct

        rng = np.random.default_rng(1)
        x = rng.normal(size=(args.rows, args.dense_features)).astype("<f4")
        body = struct.pack("<II", args.rows, args.dense_features) + x.tobytes()
    else:
        body = json.dumps(make_request_batch(args.rows, seed=1)).encode()
    stats = {"lat": [], "errors": 0}
    if getattr(args, "raw_client", False):
        from urllib.parse import urlparse

        u = urlparse(args.url)
        host, port = u.hostname, u.port or 80
        # warmup (also verifies the server answers)
        w = {"lat": [], "errors": 0}
        ep = args.endpoint.encode()
        ct = (b"application/octet-stream" if getattr(args, "dense_features", 0)
              else b"application/json")
        await raw_worker(host, port, body, time.perf_counter() + 0.5, w, ep, ct)
        stop_at = time.perf_counter() + args.duration
        t0 = time.perf_counter()
        await asyncio.gather(
            *(raw_worker(host, port, body, stop_at, stats, ep, ct)
              for _ in range(args.concurrency))
        )
        elapsed = time.perf_counter() - t0
    else:
      async with httpx.AsyncClient(timeout=60.0) as client:
        # warmup
        ct = ("application/octet-stream" if getattr(args, "dense_features", 0)
              else "application/json")
        await client.post(f"{args.url}{args.endpoint}", content=body,
                          headers={"content-type": ct})
        stop_at = time.perf_counter() + args.duration
        t0 = time.perf_counter()
        await asyncio.gather(
            *(worker(client, args.url, body, stop_at, stats, args.endpoint, ct)
              for _ in range(args.concurrency))
        )
        elapsed = time.perf_counter() - t0
    lat = np.sort(np.asarray(stats["lat"]))
    n = len(lat)
    out = {
        "metric": "HTTP requests/sec, credit-default /score",
        "value": round(n / elapsed, 2),
        "unit": "requests/s",
        "rows_per_request": args.rows,
        "rows_per_sec": round(n * args.rows / elapsed, 1),
        "concurrency": args.concurrency,
        "duration_s": round(elapsed, 1),
        "errors": stats["errors"],
        "latency_ms_p50": round(float(lat[n // 2]), 2) if n else None,
        "latency_ms_p90": round(float(lat[int(0.9 * n)]), 2) if n else None,
        "latency_ms_p99": round(float(lat[min(n - 1, int(0.99 * n))]), 2) if n else None,
    }
    print(json.dumps(out))


def _proc_worker(args, q):
    import io
    import contextlib

    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        asyncio.run(run(args))
    q.put(buf.getvalue())


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--url", default="http://127.0.0.1:5000")
    p.add_argument("--rows", type=int, default=1024)
    p.add_argument("--concurrency", type=int, default=8)
    p.add_argument("--duration", type=float, default=20.0)
    p.add_argument("--processes", type=int, default=1,
                   help="client processes (a single event loop saturates "
                        "before the server does on large bodies)")
    p.add_argument("--endpoint", default="/score")
    p.add_argument("--dense-features", type=int, default=0,
                   help="send binary dense bodies with this many features "
                        "(use with --endpoint /predict_dense)")
    p.add_argument("--raw-client", action="store_true",
                   help="raw-socket HTTP client (httpx costs ~1.4 ms/req "
                        "of client CPU and caps the measurement)")
    args = p.parse_args()
    if args.processes <= 1:
        asyncio.run(run(args))
        return
    import copy
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    sub = copy.copy(args)
    sub.concurrency = max(1, args.concurrency // args.processes)
    procs = [ctx.Process(target=_proc_worker, args=(sub, q)) for _ in range(args.processes)]
    for pr in procs:
        pr.start()
    results = [json.loads(q.get(timeout=args.duration + 120)) for _ in procs]
    for pr in procs:
        pr.join(timeout=30)
    total = sum(r["value"] for r in results)
    out = dict(results[0])
    out["value"] = round(total, 2)
    out["rows_per_sec"] = round(total * args.rows, 1)
    out["concurrency"] = sub.concurrency * args.processes
    out["client_processes"] = args.processes
    for k in ("latency_ms_p50", "latency_ms_p90", "latency_ms_p99"):
        vals = [r[k] for r in results if r[k] is not None]
        out[k] = round(float(np.median(vals)), 2) if vals else None
    print(json.dumps(out))


if __name__ == "__main__":
    main()
