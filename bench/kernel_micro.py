"""Per-kernel micro-benchmarks (hipEvent timing via torch.cuda.Event).

Sweeps launch configurations of the hot kernels on a real MI355X so the
session picks measured-best settings:

    python bench/kernel_micro.py            # K-S sweep + forest timing
"""

from __future__ import annotations

import json

import numpy as np

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


def timed(fn, iters=50, warmup=10):
    import torch

    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3  # µs


def main():
    import torch

    from creditcore.ops import gpu
    from creditcore.pack import PackedModel, encode_batch
    from creditcore.data import make_request_batch

    ext = gpu.ext()
    p = PackedModel.load("/tmp/bench_packed_500x16_20000.npz")
    dev = torch.device("cuda", 0)

    medians = torch.from_numpy(p.medians).to(dev)
    ref = torch.from_numpy(p.ref_sorted).to(dev)
    rs_off = torch.from_numpy(p.ref_sorted_offsets).to(dev)

    results = {}
    for b in (1024, 4096, 16384):
        recs = make_request_batch(b, seed=b)
        codes, nums = encode_batch(recs, p.vocabs)
        d_nums = torch.from_numpy(nums).to(dev)
        for block in (256, 512):
            key = f"ks b={b} block={block}"
            try:
                us = timed(lambda: ext.ks_stats(d_nums, medians, ref, rs_off, block, 0))
                results[key] = round(us, 2)
            except Exception as exc:
                results[key] = f"error: {exc}"

    # forest via the functional pipeline (classifier + iforest + finalize)
    cls_nodes = torch.from_numpy(np.ascontiguousarray(p.cls_nodes)).to(dev)
    cls_off = torch.from_numpy(p.cls_tree_offsets).to(dev)
    if_nodes = torch.from_numpy(np.ascontiguousarray(p.if_nodes)).to(dev)
    if_off = torch.from_numpy(p.if_tree_offsets).to(dev)
    fc = torch.from_numpy(p.feat_col).to(dev)
    fk = torch.from_numpy(p.feat_code).to(dev)
    for b in (1024, 16384):
        recs = make_request_batch(b, seed=b)
        codes, nums = encode_batch(recs, p.vocabs)
        d_codes = torch.from_numpy(codes).to(dev)
        d_nums = torch.from_numpy(nums).to(dev)
        us = timed(
            lambda: ext.score_forest_pipeline(
                d_codes, d_nums, cls_nodes, cls_off, fc, fk, medians,
                p.n_onehot, if_nodes, if_off,
                p.if_denom, p.if_offset, p.if_threshold,
            )
        )
        results[f"forest_pipeline b={b}"] = round(us, 2)
        # A/B: ILP depth x grid layout (classifier forest alone)
        # 0=1-tree, 1=ilp2, 2=ilp4, 3=ilp2+XCD-transposed grid, 4=ilp4+transposed
        for ilp in (0, 1, 2, 3, 4):
            us = timed(
                lambda: ext.forest_ilp_bench(
                    d_codes, d_nums, cls_nodes, cls_off, fc, fk, medians, ilp
                )
            )
            results[f"forest_cls b={b} ilp={ilp}"] = round(us, 2)
        a = ext.forest_ilp_bench(d_codes, d_nums, cls_nodes, cls_off, fc, fk, medians, 0)
        for v in (1, 2, 3, 4):
            c = ext.forest_ilp_bench(d_codes, d_nums, cls_nodes, cls_off, fc, fk, medians, v)
            results[f"forest_ilp_parity b={b} v={v}"] = float((a - c).abs().max().item())

    print(json.dumps(results, indent=2))


if __name__ == "__main__":
    main()
