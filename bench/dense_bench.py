"""Dense-family benchmark (BASELINE.json config 5): large synthetic tabular
train + serve with the drift reference resident in HBM.

    python bench/dense_bench.py --train-rows 10000000 --feats 1000 \
        --ref-rows 10000000 --score-rows 16384 --steps 50

Reports one JSON line: train seconds, resident HBM bytes, scoring rows/s
(fused dense_score kernel + 1k-feature exact K-S drift per step).
"""

from __future__ import annotations

import argparse
import json
import sys
import time

import numpy as np

# runnable from the repo root or anywhere: put the repo on sys.path
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--train-rows", type=int, default=10_000_000)
    p.add_argument("--feats", type=int, default=1000)
    p.add_argument("--ref-rows", type=int, default=None)
    p.add_argument("--score-rows", type=int, default=16384)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--device", default="auto")
    a = p.parse_args()

    import torch

    from creditcore.dense import DenseEngine, train_dense

    device = a.device
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"

    t0 = time.time()
    model = train_dense(
        n_rows=a.train_rows,
        n_feats=a.feats,
        ref_rows=a.ref_rows,
        epochs=a.epochs,
        device=device,
        keep_on_device=(device == "cuda"),
        log=lambda *x: print(*x, file=sys.stderr),
    )
    if device == "cuda":
        torch.cuda.synchronize()  # train queues async work (ref-build sorts)
    train_s = time.time() - t0
    eng = DenseEngine(model, device=device)

    rng = np.random.default_rng(0)
    xs = [
        rng.standard_normal((a.score_rows, a.feats)).astype(np.float32)
        for _ in range(2)
    ]
    for i in range(a.warmup):
        eng.score_arrays(xs[i % 2])
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(a.steps):
        out = eng.score_arrays(xs[i % 2])
    if device == "cuda":
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    hbm = eng.hbm_bytes()
    alloc = torch.cuda.memory_allocated() if device == "cuda" else 0
    print(
        json.dumps(
            {
                "metric": "dense scoring rows/s (10Mx1k family)",
                "value": round(a.score_rows * a.steps / elapsed, 1),
                "unit": "rows/s",
                "ms_per_step": round(elapsed / a.steps * 1e3, 3),
                "train_seconds": round(train_s, 1),
                "train_rows": a.train_rows,
                "n_feats": a.feats,
                "ref_rows": model.n_ref,
                "model_hbm_bytes": hbm,
                "torch_hbm_allocated": alloc,
                "score_rows": a.score_rows,
                "with_drift": True,
                "device": device,
            }
        )
    )


if __name__ == "__main__":
    main()
