"""Scoring engine — one model replica on one device.

The GPU path holds a C++ ScoreSession (csrc/creditcore_kernels.hip):
resident HBM model buffers, two private HIP streams, double-buffered pinned
staging, and one hipGraph per recurring request shape. A request is:

    host:   native JSON parse -> codes (int16), nums (f32)   [C, GIL-free]
    graph:  pinned H2D -> 2-tree-ILP forest kernels (classifier + isolation
            forest) -> finalize (re-zeros its accumulator rows) -> outs D2H,
            ∥ drift branch on stream 2 (categorical histogram + K-S) ->
            one packed drift D2H; no memset nodes
    host:   drift p-values (C: chi2 closed forms + MTW/Pelz-Good K-S sf)
            -> response JSON bytes (C, shortest-round-trip doubles)

Entry points by use: score_json_full (single C++ call, lowest latency),
submit_encoded_slot/finish_slot (double-buffered pipelining: step i's
epilogue overlaps step i+1's graph), score_arrays (array in/out for the
micro-batcher), score_records/score_json (dict/response-shaped outputs).

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
    # K-S kernel LDS sort capacity (csrc MAX_DRIFT_ROWS) — the hardware
    # ceiling; the per-engine cap (config drift_max_batch) can only lower it
    HW_DRIFT_MAX_ROWS = 16384

    def __init__(
        self,
        packed: PackedModel,
        device: str = "cpu",
        device_index: int = 0,
        drift_max_rows: int | None = None,
    ):
        self.packed = packed
        self.device = device
        self.device_index = device_index
        self.n_features = N_CAT + N_NUM
        self.DRIFT_MAX_ROWS = min(
            drift_max_rows or self.HW_DRIFT_MAX_ROWS, self.HW_DRIFT_MAX_ROWS
        )
        self._gpu = None
        if device == "cuda":
            self._init_gpu()

    # ------------------------------------------------------------------ GPU

    def _init_gpu(self, capacity: int = 16384):
        import torch

        from .ops import gpu

        ext = gpu.ext()  # raises loudly if the HIP extension is missing
        p = self.packed
        model = {
            "cls_nodes": torch.from_numpy(np.ascontiguousarray(p.cls_nodes)),
            "cls_tree_offsets": torch.from_numpy(p.cls_tree_offsets),
            "feat_col": torch.from_numpy(p.feat_col),
            "feat_code": torch.from_numpy(p.feat_code),
            "medians": torch.from_numpy(p.medians),
            "if_nodes": torch.from_numpy(np.ascontiguousarray(p.if_nodes)),
            "if_tree_offsets": torch.from_numpy(p.if_tree_offsets),
            "ref_sorted": torch.from_numpy(p.ref_sorted),
            "ref_sorted_offsets": torch.from_numpy(p.ref_sorted_offsets),
            "ref_cat_offsets": torch.from_numpy(p.ref_cat_offsets),
            "cls_kind": int(getattr(p, "cls_kind", 0)),
            "cls_bias": float(getattr(p, "cls_bias", 0.0)),
            "if_denom": float(p.if_denom),
            "if_offset": float(p.if_offset),
            "if_threshold": float(p.if_threshold),
        }
        sess = ext.ScoreSession(model, capacity, self.device_index)
        self._gpu = {"ext": ext, "model": model}
        self._set_session(sess)

    def _set_session(self, sess):
        g = self._gpu
        g["sess"] = sess
        # zero-copy numpy views of the session's pinned staging/outputs;
        # first index = slot (double-buffered for pipelined callers)
        g["np_codes"] = sess.pin_codes.numpy()
        g["np_nums"] = sess.pin_nums.numpy()
        g["np_outs"] = sess.pin_outs.numpy()
        g["np_hist"] = sess.pin_hist.numpy()
        g["np_ksd"] = sess.pin_ksd.numpy()

    def _ensure_capacity(self, b: int):
        g = self._gpu
        if b > g["sess"].capacity:
            cap = 1 << (b - 1).bit_length()
            self._set_session(g["ext"].ScoreSession(g["model"], cap, self.device_index))

    def _score_gpu(self, codes: np.ndarray, nums: np.ndarray, with_drift: bool = True) -> dict:
        g = self._gpu
        b = len(codes)
        self._ensure_capacity(b)
        # Drift is a batch-population statistic; cap its sample at the K-S
        # kernel's LDS sort capacity (predictions still cover every row).
        drift_now = with_drift and b <= self.DRIFT_MAX_ROWS
        # one C call: pinned staging memcpy (GIL released) + graph replay
        g["sess"].submit_arrays(codes, nums, 0, drift_now, True)
        flat = g["np_outs"][0].reshape(-1)  # b-packed: proba | iscore | outlier
        out = {
            "predictions": flat[:b].copy(),
            "instance_score": flat[b : 2 * b].copy(),
            "outliers": flat[2 * b : 3 * b].copy(),
        }
        if with_drift and not drift_now:
            # oversized batch: score a capped drift sample in a second pass
            db = self.DRIFT_MAX_ROWS
            g["sess"].score(db, True, True)
            drift_now, b = True, db
        if drift_now:
            hist = g["np_hist"][0].copy()
            ks_d = g["np_ksd"][0].copy()
            out["p_vals"] = cpu_ref.pvals_from_stats(self.packed, hist, ks_d, b)
            out["cat_hist"] = hist
            out["ks_d"] = ks_d
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

    def encode_json_body(self, body: bytes) -> tuple:
        """Parse a raw /score JSON body into (codes, nums) with the native C
        parser (GIL released during the parse); raises ValueError on
        malformed/ill-typed bodies (callers fall back to pydantic for proper
        422 semantics)."""
        from .ops import gpu
        from .pack import CATEGORICAL_FEATURES, MISSING_CATEGORY, NUMERIC_FEATURES

        if gpu.available():
            codes, nums = self._ensure_json_encoder().encode(body)
            return np.asarray(codes), np.asarray(nums)
        import json

        recs = json.loads(body)
        if not isinstance(recs, list) or not all(isinstance(r, dict) for r in recs):
            raise ValueError("body must be a JSON array of records")
        # Strictness parity with the native parser: nulls and ill-typed
        # values raise ValueError so callers fall back to pydantic for the
        # reference's 422/coercion semantics; absent fields take the schema
        # defaults (pydantic default semantics, reference app/model.py:8-34
        # — encode_batch alone would fill missing categoricals with
        # MISSING_CATEGORY instead).
        cat = set(CATEGORICAL_FEATURES)
        num = set(NUMERIC_FEATURES)
        for r in recs:
            for k, v in r.items():
                if k in cat:
                    if not isinstance(v, str):
                        raise ValueError(f"field {k}: expected string")
                elif k in num:
                    if isinstance(v, bool) or not isinstance(v, (int, float)):
                        raise ValueError(f"field {k}: expected number")
        from .schema import LoanApplicant

        defaults = LoanApplicant().__dict__
        recs = [{**defaults, **r} for r in recs]
        return encode_batch(recs, self.packed.vocabs)

    def _ensure_json_encoder(self):
        enc = getattr(self, "_json_encoder", None)
        if enc is None:
            from .ops import gpu
            from .pack import CATEGORICAL_FEATURES, MISSING_CATEGORY, NUMERIC_FEATURES

            dc, dn = self.default_rows()
            enc = gpu.ext().JsonEncoder(
                self.packed.vocabs,
                CATEGORICAL_FEATURES,
                NUMERIC_FEATURES,
                MISSING_CATEGORY,
                dc,
                dn,
            )
            self._json_encoder = enc
        return enc

    def score_json(self, body: bytes) -> dict:
        """Score a raw /score JSON request body (wire-format fast path)."""
        codes, nums = self.encode_json_body(body)
        return self._score_encoded(codes, nums)

    def score_json_full(self, body: bytes) -> dict:
        """Fully-native request: one C++ call parses the wire JSON, stages,
        replays the graph, converts drift p-values and serializes the
        response — no Python object pass. Falls back to the staged path for
        oversized batches or on CPU."""
        if self.device != "cuda":
            return self.score_json_bytes(body)
        g = self._gpu
        try:
            resp, rows = g["sess"].score_json_full(
                body,
                self._ensure_json_encoder(),
                self.packed.ref_cat_counts,
                self.packed.ref_cat_offsets,
                int(self.packed.ref_sorted_offsets[1] - self.packed.ref_sorted_offsets[0]),
                FEATURES,
                self.DRIFT_MAX_ROWS,
            )
        except RuntimeError:
            # batch larger than the resident session capacity: grow + retry
            # through the staged path
            return self.score_json_bytes(body)
        return {
            "response_bytes": resp,
            "rows": int(rows),
            "cat_hist": g["np_hist"][0].copy(),
        }

    def default_rows(self) -> tuple:
        """Encoded schema-default record (absent request fields take these
        values — pydantic default semantics, reference app/model.py:8-34)."""
        if getattr(self, "_default_rows", None) is None:
            from .schema import LoanApplicant

            dc, dn = encode_batch([LoanApplicant().__dict__], self.packed.vocabs)
            self._default_rows = (
                np.ascontiguousarray(dc[0]),
                np.ascontiguousarray(dn[0]),
            )
        return self._default_rows

    def score_json_bytes(self, body: bytes) -> dict:
        """Wire-format in, wire-format out: parse the JSON request with the
        C parser, score through the session graph, and serialize the
        response JSON in C — the full /score answer with no Python object
        pass. Returns {"response_bytes", "rows"}."""
        codes, nums = self.encode_json_body(body)
        return self.score_encoded_bytes(codes, nums)

    def score_encoded_bytes(self, codes: np.ndarray, nums: np.ndarray) -> dict:
        b = len(codes)
        if self.device != "cuda" or b == 0:
            import json

            out = self._score_encoded(codes, nums)
            return {
                "response_bytes": json.dumps(out["response"]).encode(),
                "rows": b,
            }
        g = self._gpu
        self._ensure_capacity(b)
        nb = b
        if b <= self.DRIFT_MAX_ROWS:
            g["sess"].submit_arrays(codes, nums, 0, True, True)
        else:
            g["np_codes"][0][:b] = codes
            g["np_nums"][0][:b] = nums
            # Oversized batch: drift is a batch-population statistic, so run
            # the capped-sample drift pass FIRST (only with_drift passes
            # write the pinned drift blob), then the full batch without
            # drift — pin_outs then holds the b-packed layout
            # build_response_json reads. The reverse order overwrote the
            # b-packed outputs with a cap-packed second pass and corrupted
            # rows >= cap (round-1 advisor finding).
            g["sess"].score(self.DRIFT_MAX_ROWS, True, True)
            g["sess"].score(b, False, True)
            nb = self.DRIFT_MAX_ROWS
        pvals = cpu_ref.pvals_from_stats(self.packed, g["np_hist"][0], g["np_ksd"][0], nb)
        resp = g["ext"].build_response_json(
            g["sess"].pin_outs, b, np.ascontiguousarray(pvals), FEATURES
        )
        return {
            "response_bytes": resp,
            "rows": b,
            # for node-global drift aggregation by the serving layer
            "cat_hist": g["np_hist"][0].copy(),
        }

    # -------------------------------------------------- pipelined slot API
    def submit_encoded_slot(self, codes: np.ndarray, nums: np.ndarray, slot: int) -> int:
        """Stage + launch (no sync) into a pinned slot; pair with
        finish_slot. Slot i%2 lets step i's epilogue overlap step i+1's
        graph. Falls back to capacity growth like the sync path."""
        g = self._gpu
        b = len(codes)
        self._ensure_capacity(b)
        assert b <= self.DRIFT_MAX_ROWS, "pipelined path caps at DRIFT_MAX_ROWS"
        g["sess"].submit_arrays(codes, nums, slot & 1, True, False)
        return b

    def finish_slot(self, slot: int, b: int) -> dict:
        """Wait for the slot's graph; C epilogue: drift p-values +
        response serialization from the slot's pinned buffers."""
        g = self._gpu
        s = slot & 1
        n_ref = int(self.packed.ref_sorted_offsets[1] - self.packed.ref_sorted_offsets[0])
        resp = g["sess"].response_epilogue(
            s, b, self.packed.ref_cat_counts, self.packed.ref_cat_offsets,
            n_ref, FEATURES,
        )
        return {
            "response_bytes": resp,
            "rows": b,
            "cat_hist": g["np_hist"][s].copy(),
        }

    def score_records(self, records) -> dict:
        """Score a request body (list of dicts / DataFrame); returns the
        reference response shape (02-register cell-9)."""
        codes, nums = encode_batch(records, self.packed.vocabs)
        return self._score_encoded(codes, nums)

    def _score_encoded(self, codes: np.ndarray, nums: np.ndarray) -> dict:
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


def load_engine(
    model_directory: str,
    device: str = "auto",
    device_index: int = 0,
    drift_max_rows: int | None = None,
) -> ScoringEngine:
    """Load a model into a ScoringEngine. Accepts a pyfunc model dir, a
    packed .npz, or a registry URI (``models:/<name>/<version|latest>`` —
    the reference's MLflow registry addressing, 02-register cell-15)."""
    import os

    from . import pack as packmod
    from .config import ServeConfig
    from .registry import resolve_model_uri

    if device == "auto":
        device = ServeConfig().resolve_device()
    model_directory = resolve_model_uri(model_directory)
    if model_directory.endswith(".npz") and os.path.isfile(model_directory):
        packed = PackedModel.load(model_directory)
    else:
        packed = packmod.pack_pyfunc_dir(model_directory)
    return ScoringEngine(
        packed,
        device=device,
        device_index=device_index,
        drift_max_rows=drift_max_rows,
    )
