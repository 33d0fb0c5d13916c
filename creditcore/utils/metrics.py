"""In-process service metrics (the local /metrics replacement for the
reference's Log Analytics + KQL pipeline — SURVEY.md §5.5)."""

from __future__ import annotations

import threading
import time
from collections import deque


class Metrics:
    """Lock-protected counters + a sliding latency window for percentiles."""

    def __init__(self, window: int = 4096):
        self._lock = threading.Lock()
        self._lat = deque(maxlen=window)
        self.requests_total = 0
        self.rows_total = 0
        self.errors_total = 0
        self.drift_syncs_total = 0
        self.started_at = time.time()

    def observe_request(self, rows: int, latency_ms: float) -> None:
        with self._lock:
            self.requests_total += 1
            self.rows_total += rows
            self._lat.append(latency_ms)

    def observe_error(self) -> None:
        with self._lock:
            self.errors_total += 1

    def observe_drift_sync(self) -> None:
        with self._lock:
            self.drift_syncs_total += 1

    def snapshot(self) -> dict:
        with self._lock:
            lat = sorted(self._lat)
            n = len(lat)

            def pct(p: float) -> float | None:
                return round(lat[min(n - 1, int(p * n))], 3) if n else None

            return {
                "uptime_s": round(time.time() - self.started_at, 1),
                "requests_total": self.requests_total,
                "rows_total": self.rows_total,
                "errors_total": self.errors_total,
                "drift_syncs_total": self.drift_syncs_total,
                "latency_ms_p50": pct(0.50),
                "latency_ms_p90": pct(0.90),
                "latency_ms_p99": pct(0.99),
            }

    def prometheus(self, prefix: str = "creditcore") -> str:
        """Render the snapshot in the Prometheus text exposition format
        (what a K8s scrape expects — the reference shipped logs to Log
        Analytics instead and had no scrape endpoint)."""
        s = self.snapshot()
        lines = []
        for name, kind in (
            ("requests_total", "counter"),
            ("rows_total", "counter"),
            ("errors_total", "counter"),
            ("drift_syncs_total", "counter"),
            ("uptime_s", "gauge"),
        ):
            lines.append(f"# TYPE {prefix}_{name} {kind}")
            lines.append(f"{prefix}_{name} {s[name]}")
        lines.append(f"# TYPE {prefix}_latency_ms summary")
        for q, key in (("0.5", "latency_ms_p50"), ("0.9", "latency_ms_p90"),
                       ("0.99", "latency_ms_p99")):
            if s[key] is not None:
                lines.append(f'{prefix}_latency_ms{{quantile="{q}"}} {s[key]}')
        return "\n".join(lines) + "\n"
