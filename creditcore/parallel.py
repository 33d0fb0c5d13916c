"""Cross-GPU parallelism: RCCL weight broadcast + node-global drift state.

The reference scales by identical K8s pod replicas that share nothing at
runtime: the model is baked into every container image
(reference app/Dockerfile:18) and each pod computes drift statistics on its
own request batches in isolation. The MI355X-native
equivalents here:

- ``broadcast_packed`` — one-shot distribution of the packed model buffers
  from rank 0 to all replica ranks over RCCL/xGMI (``torch.distributed``
  backend "nccl" IS RCCL on ROCm; tests use "gloo" on CPU). Replaces N
  container images each carrying a model copy.
- ``DriftSync`` — per-replica drift histograms (categorical bins exact,
  numerics binned on a reference-quantile grid) merged across the node with
  a bucketed all-reduce, so the node reports *global* drift instead of the
  reference's per-pod-isolated statistics. The payload is one small int64
  tensor (≈23×bins) — latency-, not bandwidth-bound on xGMI, so a single
  flat buffer + one all_reduce per sync is the right shape (no ring
  bucketing needed at this size).

Both work in two modes:
- one process per GPU via ``torch.distributed`` (RCCL over xGMI) — the
  serving/bench scale-out mode;
- single process, no process group — ``DriftSync.allreduce`` degrades to a
  local no-op merge and in-process multi-GPU serving sums engine histograms
  directly (cross-GPU tensor copies also ride xGMI).
"""

from __future__ import annotations

import numpy as np

from .models.drift import chi2_from_counts, ks_asymp_pvalue
from .pack import N_CAT, N_NUM, PackedModel
from .schema import FEATURES

_META_FIELDS = [
    "n_onehot",
    "cls_kind",
    "cls_bias",
    "if_denom",
    "if_offset",
    "if_threshold",
    "n_ref",
    "drift_p_val",
    "lin_bias",
    "vocabs",
    "meta",
]
_ARRAY_FIELDS = [
    ("medians", np.float32),
    ("feat_col", np.int32),
    ("feat_code", np.int32),
    ("cls_nodes", np.int32),
    ("cls_tree_offsets", np.int32),
    ("if_nodes", np.int32),
    ("if_tree_offsets", np.int32),
    ("ref_sorted", np.float32),
    ("ref_sorted_offsets", np.int32),
    ("ref_cat_counts", np.int32),
    ("ref_cat_offsets", np.int32),
]


def broadcast_packed(
    packed: PackedModel | None, device: str = "cpu", src: int = 0, group=None
) -> PackedModel:
    """Broadcast a PackedModel from rank ``src`` to every rank.

    Rank src passes the model; other ranks pass None and receive it. Large
    buffers go as tensor broadcasts (RCCL over xGMI on "nccl"); shapes,
    vocabularies and scalars ride one broadcast_object_list.
    """
    import torch
    import torch.distributed as dist

    rank = dist.get_rank(group)
    is_src = rank == src

    if is_src:
        assert packed is not None
        meta = {f: getattr(packed, f) for f in _META_FIELDS}
        meta["shapes"] = {
            name: list(getattr(packed, name).shape) for name, _ in _ARRAY_FIELDS
        }
        meta["has_lin"] = packed.lin_weight is not None
        if meta["has_lin"]:
            meta["lin_shape"] = list(packed.lin_weight.shape)
        obj = [meta]
    else:
        obj = [None]
    dist.broadcast_object_list(obj, src=src, group=group)
    meta = obj[0]

    torch_dtype = {np.float32: torch.float32, np.int32: torch.int32}
    tensors = {}
    for name, npdt in _ARRAY_FIELDS:
        if is_src:
            t = torch.from_numpy(
                np.ascontiguousarray(getattr(packed, name).astype(npdt, copy=False))
            ).to(device)
        else:
            t = torch.empty(meta["shapes"][name], dtype=torch_dtype[npdt], device=device)
        dist.broadcast(t, src=src, group=group)
        tensors[name] = t

    # optional linear-predict weights (dense/linear family models)
    lin_weight = None
    if meta["has_lin"]:
        if is_src:
            lw = torch.from_numpy(
                np.ascontiguousarray(packed.lin_weight.astype(np.float32, copy=False))
            ).to(device)
        else:
            lw = torch.empty(meta["lin_shape"], dtype=torch.float32, device=device)
        dist.broadcast(lw, src=src, group=group)
        lin_weight = lw.cpu().numpy()

    if is_src:
        return packed
    kw = {name: tensors[name].cpu().numpy() for name, _ in _ARRAY_FIELDS}
    for f in _META_FIELDS:
        kw[f] = meta[f]
    kw.pop("meta", None)
    return PackedModel(lin_weight=lin_weight, meta=meta["meta"], **kw)


class DriftSync:
    """Node-global drift accumulator.

    Layout of the flat histogram buffer (int64):
        [0, C)            categorical bins (ref_cat_offsets layout, C = total)
        [C, C + 14*K)     numeric bins, K quantile bins per feature

    Numeric bin edges are the reference distribution's (K-1) interior
    quantiles, so under no drift each bin holds ≈1/K of the mass. The
    node-global numeric test is a binned two-sample K-S (exact up to grid
    resolution 1/K); the categorical test is the same chi-square as the
    per-request path.
    """

    def __init__(self, packed: PackedModel, device: str = "cpu", n_bins: int = 64):
        import torch

        self.packed = packed
        self.device = device
        self.n_bins = int(n_bins)
        self.C = int(packed.ref_cat_offsets[-1])

        edges, ref_num_counts = [], []
        for j in range(N_NUM):
            lo, hi = packed.ref_sorted_offsets[j], packed.ref_sorted_offsets[j + 1]
            ref = packed.ref_sorted[lo:hi].astype(np.float64)
            qs = np.quantile(ref, np.linspace(0, 1, self.n_bins + 1)[1:-1])
            edges.append(qs.astype(np.float32))
            idx = np.searchsorted(qs, ref, side="right")
            ref_num_counts.append(np.bincount(idx, minlength=self.n_bins))
        # [N_NUM, K-1] per-row sorted boundaries for batched searchsorted
        self.edges = torch.from_numpy(np.stack(edges)).to(device)
        self.ref_num_counts = np.stack(ref_num_counts)  # [N_NUM, K]
        self.medians_t = torch.from_numpy(packed.medians).to(device)

        self.size = self.C + N_NUM * self.n_bins
        self.local = torch.zeros(self.size, dtype=torch.int64, device=device)
        self.global_ = self.local.clone()
        self.batches = 0

    def accumulate(self, cat_hist, nums) -> None:
        """Fold one scored batch into the local histogram.

        ``cat_hist``: int array/tensor [C] (the drift kernel's output);
        ``nums``: float32 array/tensor [B, 14] (raw, NaNs allowed).
        numpy inputs take a numpy fast path (per-request torch-CPU op
        overhead was ~6 ms at bs=1024); tensors keep the torch path (GPU).
        """
        import torch

        if not torch.is_tensor(nums):
            if self.local.device.type == "cpu":
                self._accumulate_np(np.asarray(cat_hist), np.asarray(nums))
                return
            # device-resident accumulator: route numpy inputs through the
            # torch path (the numpy fast path writes through .numpy(),
            # which only exists for CPU tensors)
            cat_hist = torch.from_numpy(np.ascontiguousarray(cat_hist))
            nums = torch.from_numpy(np.ascontiguousarray(nums))
        self.local[: self.C] += cat_hist.to(self.local.device, torch.int64)
        x = nums.to(self.edges.device).t().contiguous()  # [N_NUM, B]
        x = torch.where(torch.isnan(x), self.medians_t[:, None], x)
        idx = torch.searchsorted(self.edges, x, right=True)  # [N_NUM, B]
        offs = torch.arange(N_NUM, device=idx.device)[:, None] * self.n_bins
        flat = (idx + offs).reshape(-1)
        binc = torch.bincount(flat, minlength=N_NUM * self.n_bins)
        self.local[self.C :] += binc.to(self.local.device)
        self.batches += 1

    def _accumulate_np(self, cat_hist: np.ndarray, nums: np.ndarray) -> None:
        if getattr(self, "_edges_np", None) is None:
            self._edges_np = self.edges.cpu().numpy()
            self._medians_np = self.medians_t.cpu().numpy()
        local = self.local.numpy()
        local[: self.C] += cat_hist.astype(np.int64)
        x = np.where(np.isnan(nums), self._medians_np[None, :], nums)
        offs = np.arange(N_NUM) * self.n_bins
        idx = np.empty((N_NUM, len(x)), dtype=np.int64)
        for j in range(N_NUM):
            idx[j] = np.searchsorted(self._edges_np[j], x[:, j], side="right")
        binc = np.bincount((idx + offs[:, None]).reshape(-1), minlength=N_NUM * self.n_bins)
        local[self.C :] += binc
        self.batches += 1

    def allreduce(self, group=None) -> None:
        """Merge local histograms across replicas (RCCL all-reduce when a
        process group is initialized; local copy otherwise)."""
        import torch.distributed as dist

        buf = self.local.clone()
        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=group)
        self.global_ = buf

    def node_pvals(self) -> np.ndarray:
        """Per-feature p-values from the node-global histograms, schema
        feature order (categoricals then numerics)."""
        g = self.global_.cpu().numpy()
        pvals = np.ones(N_CAT + N_NUM)
        for j in range(N_CAT):
            lo, hi = self.packed.ref_cat_offsets[j], self.packed.ref_cat_offsets[j + 1]
            pvals[j] = chi2_from_counts(self.packed.ref_cat_counts[lo:hi], g[lo:hi])
        m_total = int(g[self.C : self.C + self.n_bins].sum())
        if m_total > 0:
            for j in range(N_NUM):
                bc = g[self.C + j * self.n_bins : self.C + (j + 1) * self.n_bins]
                rc = self.ref_num_counts[j]
                f_b = np.cumsum(bc) / max(bc.sum(), 1)
                f_r = np.cumsum(rc) / rc.sum()
                d = np.abs(f_b - f_r).max()
                pvals[N_CAT + j] = ks_asymp_pvalue(float(d), int(rc.sum()), int(bc.sum()))
        return pvals

    def snapshot(self) -> dict:
        pvals = self.node_pvals()
        return {
            "batches": self.batches,
            "rows": int(self.global_[self.C : self.C + self.n_bins].sum()),
            "node_feature_drift": {
                f: float(np.float32(1.0) - np.float32(p))
                for f, p in zip(FEATURES, pvals)
            },
        }

    def merge_from(self, other: "DriftSync") -> None:
        """In-process multi-GPU merge (no process group): sum another
        replica's local histogram into this one (cross-GPU copy over xGMI)."""
        self.local += other.local.to(self.local.device)

    # -------------------------------------------------- restart persistence
    def save_state(self, path: str) -> None:
        """Persist the local histogram so drift state survives a service
        restart (the reference's pods lose all drift context on restart)."""
        import os
        import tempfile

        d = os.path.dirname(os.path.abspath(path))
        os.makedirs(d, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
        try:
            with os.fdopen(fd, "wb") as f:
                np.savez(f, local=self.local.cpu().numpy(),
                         batches=np.int64(self.batches), n_bins=np.int64(self.n_bins))
            os.replace(tmp, path)
        except BaseException:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise

    def load_state(self, path: str) -> bool:
        """Restore a saved histogram; returns False (and starts fresh) on a
        missing file or a bin-layout mismatch (e.g. model/bins changed)."""
        import os

        import torch

        if not os.path.isfile(path):
            return False
        try:
            with np.load(path) as z:
                if int(z["n_bins"]) != self.n_bins or z["local"].shape != (self.size,):
                    return False
                self.local = torch.from_numpy(z["local"].copy()).to(self.local.device)
                self.batches = int(z["batches"])
        except (OSError, ValueError, KeyError):
            return False
        self.allreduce()
        return True

    # -------------------------------------------------- cross-process merge
    # SO_REUSEPORT serving workers are separate processes with no process
    # group; they publish local histograms to a shared directory (tmpfs)
    # and /drift sums every worker's latest snapshot.
    def publish(self, publish_dir: str) -> None:
        import os
        import tempfile

        os.makedirs(publish_dir, exist_ok=True)
        arr = self.local.cpu().numpy()
        fd, tmp = tempfile.mkstemp(dir=publish_dir, suffix=".tmp")
        try:
            with os.fdopen(fd, "wb") as f:
                np.save(f, arr)
            os.replace(tmp, os.path.join(publish_dir, f"drift_{os.getpid()}.npy"))
        except BaseException:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise

    def merge_published(self, publish_dir: str) -> None:
        """Set global_ to the sum of every published worker snapshot (this
        worker's live local histogram replaces its own stale file)."""
        import glob
        import os

        import torch

        total = self.local.cpu().clone()
        me = f"drift_{os.getpid()}.npy"
        for f in glob.glob(os.path.join(publish_dir, "drift_*.npy")):
            if os.path.basename(f) == me:
                continue
            try:
                total += torch.from_numpy(np.load(f))
            except (OSError, ValueError):
                continue  # mid-replace or removed
        self.global_ = total.to(self.local.device)
