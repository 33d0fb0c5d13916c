"""Scoring engine — one model replica on one device.

The GPU path uploads the packed flat buffers (creditcore.pack) to HBM once and
scores request batches with the HIP kernels (csrc/kernels/*.hip) on a private
HIP stream with pinned staging buffers:

    host:   encode strings -> codes (int16), nums (f32)      [request parse]
    H2D:    codes + nums via pinned staging                  [engine stream]
    GPU:    score_pipeline kernel set:
              forest traversal  -> P(default) per row
              iforest traversal -> instance score + outlier flag
              drift             -> per-cat histograms + per-num K-S D
    D2H:    proba/outlier/stat buffers (small)
    host:   chi2 / K-S statistic -> p-value (scipy, 23 scalars)

Everything between H2D and D2H is asynchronous on the engine stream; the only
sync is the final D2H, which is part of request latency anyway.

The CPU path (device="cpu") uses the same packed buffers via
creditcore.ops.cpu_ref — identical numerics (the HIP kernels are tested
against it). On a machine with a GPU the engine REFUSES to silently fall back
to CPU: the HIP extension must be present (ops/gpu.py).
"""

from __future__ import annotations

import time

import numpy as np

from .ops import cpu_ref
from .pack import N_CAT, N_NUM, PackedModel, encode_batch
from .schema import FEATURES


class ScoringEngine:
    def __init__(self, packed: PackedModel, device: str = "cpu", device_index: int = 0):
        self.packed = packed
        self.device = device
        self.device_index = device_index
        self.n_features = N_CAT + N_NUM
        self._gpu = None
        if device == "cuda":
            self._init_gpu()

    # ------------------------------------------------------------------ GPU
    def _init_gpu(self):
        import torch

        from .ops import gpu

        ext = gpu.ext()  # raises loudly if the HIP extension is missing
        dev = torch.device("cuda", self.device_index)
        p = self.packed

        def up(a, dtype):
            return torch.from_numpy(np.ascontiguousarray(a)).to(dtype).to(dev)

        g = {
            "torch": torch,
            "ext": ext,
            "dev": dev,
            "stream": torch.cuda.Stream(device=dev),
            "cls_nodes": up(p.cls_nodes, torch.int32),
            "cls_offsets": up(p.cls_tree_offsets, torch.int32),
            "if_nodes": up(p.if_nodes, torch.int32),
            "if_offsets": up(p.if_tree_offsets, torch.int32),
            "feat_col": up(p.feat_col, torch.int32),
            "feat_code": up(p.feat_code, torch.int32),
            "medians": up(p.medians, torch.float32),
            "ref_sorted": up(p.ref_sorted, torch.float32),
            "rs_offsets": up(p.ref_sorted_offsets, torch.int32),
            "cat_offsets": up(p.ref_cat_offsets, torch.int32),
        }
        # pinned staging (grown on demand)
        g["pin_codes"] = None
        g["pin_nums"] = None
        self._gpu = g

    def _ensure_staging(self, b: int):
        import torch

        g = self._gpu
        if g["pin_codes"] is None or g["pin_codes"].shape[0] < b:
            cap = max(1024, 1 << (b - 1).bit_length())
            g["pin_codes"] = torch.empty((cap, N_CAT), dtype=torch.int16, pin_memory=True)
            g["pin_nums"] = torch.empty((cap, N_NUM), dtype=torch.float32, pin_memory=True)

    def _score_gpu(self, codes: np.ndarray, nums: np.ndarray, with_drift: bool = True) -> dict:
        torch = self._gpu["torch"]
        g = self._gpu
        b = len(codes)
        self._ensure_staging(b)
        g["pin_codes"][:b].copy_(torch.from_numpy(codes))
        g["pin_nums"][:b].copy_(torch.from_numpy(nums))
        with torch.cuda.stream(g["stream"]):
            d_codes = g["pin_codes"][:b].to(g["dev"], non_blocking=True)
            d_nums = g["pin_nums"][:b].to(g["dev"], non_blocking=True)
            proba, iscore, outlier = g["ext"].score_forest_pipeline(
                d_codes,
                d_nums,
                g["cls_nodes"],
                g["cls_offsets"],
                g["feat_col"],
                g["feat_code"],
                g["medians"],
                int(self.packed.n_onehot),
                g["if_nodes"],
                g["if_offsets"],
                float(self.packed.if_denom),
                float(self.packed.if_offset),
                float(self.packed.if_threshold),
            )
            if with_drift:
                # The K-S kernel sorts the batch column in LDS; cap the drift
                # sample at its LDS capacity (drift is a batch-population
                # statistic — a 16k-row sample of a larger batch is ample).
                db = min(b, 16384)
                cat_hist, ks_d = g["ext"].drift_stats(
                    d_codes[:db],
                    d_nums[:db],
                    g["medians"],
                    g["ref_sorted"],
                    g["rs_offsets"],
                    g["cat_offsets"],
                    int(self.packed.ref_cat_offsets[-1]),
                )
            proba_h = proba.to("cpu", non_blocking=True)
            iscore_h = iscore.to("cpu", non_blocking=True)
            outlier_h = outlier.to("cpu", non_blocking=True)
            if with_drift:
                cat_hist_h = cat_hist.to("cpu", non_blocking=True)
                ks_d_h = ks_d.to("cpu", non_blocking=True)
        g["stream"].synchronize()
        out = {
            "predictions": proba_h.double().numpy(),
            "outliers": outlier_h.double().numpy(),
            "instance_score": iscore_h.double().numpy(),
        }
        if with_drift:
            out["p_vals"] = cpu_ref.pvals_from_stats(
                self.packed, cat_hist_h.numpy(), ks_d_h.numpy(), min(b, 16384)
            )
            out["cat_hist"] = cat_hist_h.numpy()
            out["ks_d"] = ks_d_h.numpy()
        return out

    # ------------------------------------------------------------------ API
    def score_arrays(self, codes: np.ndarray, nums: np.ndarray, with_drift: bool = True) -> dict:
        """Score an encoded batch; returns predictions/outliers/p_vals arrays."""
        if self.device == "cuda":
            return self._score_gpu(codes, nums, with_drift=with_drift)
        nums_imp = cpu_ref.impute_nums(self.packed, nums)
        proba = cpu_ref.score_forest_cpu(self.packed, codes, nums_imp)
        iscore, outliers = cpu_ref.score_iforest_cpu(self.packed, nums_imp)
        out = {"predictions": proba, "outliers": outliers, "instance_score": iscore}
        if with_drift:
            cat_hist, ks_d = cpu_ref.drift_stats_cpu(self.packed, codes, nums_imp)
            out["p_vals"] = cpu_ref.pvals_from_stats(self.packed, cat_hist, ks_d, len(codes))
            out["cat_hist"] = cat_hist
            out["ks_d"] = ks_d
        return out

    def score_records(self, records) -> dict:
        """Score a request body (list of dicts / DataFrame); returns the
        reference response shape (02-register cell-9)."""
        codes, nums = encode_batch(records, self.packed.vocabs)
        t0 = time.perf_counter()
        raw = self.score_arrays(codes, nums)
        latency_ms = (time.perf_counter() - t0) * 1e3
        # (1 - p_val) computed in float32, like the reference's
        # (1 - drift_results["data"]["p_val"]).tolist() on the float32
        # p_val array (02-register cell-9); .tolist() is C-speed.
        one_minus = (
            np.float32(1.0) - np.asarray(raw["p_vals"], dtype=np.float32)
        ).astype(np.float64)
        resp = {
            "predictions": np.asarray(raw["predictions"]).tolist(),
            "outliers": np.asarray(raw["outliers"]).tolist(),
            "feature_drift_batch": dict(zip(FEATURES, one_minus.tolist())),
        }
        return {"response": resp, "latency_ms": latency_ms, "rows": len(codes)}


def load_engine(model_directory: str, device: str = "auto", device_index: int = 0) -> ScoringEngine:
    """Load a pyfunc model dir (or packed .npz) into a ScoringEngine."""
    import os

    from . import pack as packmod
    from .config import ServeConfig

    if device == "auto":
        device = ServeConfig().resolve_device()
    if model_directory.endswith(".npz") and os.path.isfile(model_directory):
        packed = PackedModel.load(model_directory)
    else:
        packed = packmod.pack_pyfunc_dir(model_directory)
    return ScoringEngine(packed, device=device, device_index=device_index)
