"""Single typed config (env overrides + CLI flags).

Replaces the reference's five disjoint config mechanisms (env vars, notebook
widgets, bundle variables, GH Actions vars, Bicep params — SURVEY.md §5.6)
with one dataclass. Env vars use the CREDITCORE_ prefix; the reference's
MODEL_DIRECTORY / SERVICE_NAME names are honoured for drop-in compatibility
(reference app/main.py:27,36).
"""

from __future__ import annotations

import argparse
import os
from dataclasses import dataclass, field, fields


def _env(name: str, default, cast=None):
    for key in (f"CREDITCORE_{name.upper()}", name.upper()):
        if key in os.environ:
            v = os.environ[key]
            if cast is bool:
                return v.lower() in ("1", "true", "yes", "on")
            return (cast or type(default))(v) if default is not None else v
    return default


@dataclass
class ServeConfig:
    # model + service (reference-compatible env names)
    model_directory: str = field(default_factory=lambda: _env("model_directory", "./model"))
    # optional dense wide-tabular model dir (enables POST /predict_dense)
    dense_model_dir: str = field(default_factory=lambda: _env("dense_model_dir", ""))
    service_name: str = field(default_factory=lambda: _env("service_name", "credit-default-api"))
    host: str = field(default_factory=lambda: _env("host", "0.0.0.0"))
    port: int = field(default_factory=lambda: _env("port", 5000, int))
    workers: int = field(default_factory=lambda: _env("workers", 1, int))
    raw_http: bool = field(default_factory=lambda: _env("raw_http", False, bool))

    # device
    device: str = field(default_factory=lambda: _env("device", "auto"))  # auto|cuda|cpu
    n_gpus: int = field(default_factory=lambda: _env("n_gpus", 0, int))  # 0 = all visible

    # micro-batching
    max_batch_rows: int = field(default_factory=lambda: _env("max_batch_rows", 8192, int))
    batch_wait_us: int = field(default_factory=lambda: _env("batch_wait_us", 300, int))
    # request-body cap for the raw frontend (413 above this)
    max_body_bytes: int = field(default_factory=lambda: _env("max_body_bytes", 64 << 20, int))

    # drift
    drift_sync_period: int = field(default_factory=lambda: _env("drift_sync_period", 64, int))
    # replica health: period of the probation loop that re-probes dead
    # replicas and re-admits the ones that answer (SURVEY §5.3 recovery)
    replica_probe_period_s: float = field(
        default_factory=lambda: _env("replica_probe_period_s", 5.0, float)
    )

    # drift-sample cap per request batch (rows beyond it are still scored;
    # drift statistics use the first drift_max_batch rows). Hardware ceiling
    # is 16384 (the K-S kernel's LDS sort capacity); values above clamp.
    drift_max_batch: int = field(default_factory=lambda: _env("drift_max_batch", 16384, int))
    # non-empty => drift histograms persist across restarts at this path
    drift_state_path: str = field(default_factory=lambda: _env("drift_state_path", ""))

    # observability
    log_inference_data: bool = field(default_factory=lambda: _env("log_inference_data", True, bool))
    log_responses: bool = field(default_factory=lambda: _env("log_responses", True, bool))

    # admin endpoints (/admin/reload). The reload body names an arbitrary
    # on-disk path that gets deserialized, so it must not be reachable by
    # any client of the public listener: with no token set, only loopback
    # clients may call it; with a token set, any client presenting it
    # (Authorization: Bearer <t> or X-Admin-Token: <t>) may.
    admin_token: str = field(default_factory=lambda: _env("admin_token", ""))

    @classmethod
    def from_args(cls, argv: list[str] | None = None) -> "ServeConfig":
        cfg = cls()
        p = argparse.ArgumentParser(prog="creditcore", description="creditcore serving")
        for f in fields(cls):
            flag = "--" + f.name.replace("_", "-")
            cur = getattr(cfg, f.name)
            if isinstance(cur, bool):
                p.add_argument(
                    flag,
                    type=lambda s: s.lower() in ("1", "true", "yes"),
                    nargs="?",
                    const=True,
                    default=cur,
                )
            else:
                p.add_argument(flag, type=type(cur), default=cur)
        ns = p.parse_args(argv)
        for f in fields(cls):
            setattr(cfg, f.name, getattr(ns, f.name))
        return cfg

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        try:
            import torch

            return "cuda" if torch.cuda.is_available() else "cpu"
        except Exception:
            return "cpu"
