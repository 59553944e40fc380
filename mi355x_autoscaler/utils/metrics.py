"""Per-cycle phase timers + Prometheus export.

The reference's only observability was log lines and Slack
[R: main.py, notification.py — reconstructed]. SURVEY §5 requires
structured per-phase timers (ingest/classify/fit/pack/allreduce/
actuate), the benchmark metrics (scale-up decision latency p50/p99,
pods-scheduled/sec) and per-kernel timings, exported via
prometheus_client when available.
"""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Optional

try:
    from prometheus_client import Counter, Gauge, Histogram, start_http_server

    _PROM = True
except Exception:  # pragma: no cover
    _PROM = False

PHASES = ("ingest", "encode", "classify", "fit", "pack", "allreduce", "decide", "actuate", "maintain")


class PhaseTimer:
    """Collects wall-time per named phase per cycle."""

    def __init__(self):
        self.current: Dict[str, float] = {}
        self.history: List[Dict[str, float]] = []

    @contextmanager
    def phase(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self.current[name] = self.current.get(name, 0.0) + (time.perf_counter() - t0)

    def end_cycle(self) -> Dict[str, float]:
        snap = dict(self.current)
        snap["total"] = sum(self.current.values())
        self.history.append(snap)
        self.current = {}
        return snap

    def percentile(self, phase: str, q: float) -> Optional[float]:
        vals = sorted(h.get(phase, 0.0) for h in self.history)
        if not vals:
            return None
        idx = min(len(vals) - 1, int(q * len(vals)))
        return vals[idx]


class CycleMetrics:
    """Counters + histograms for the control loop, optionally exported."""

    def __init__(self, prometheus_port: Optional[int] = None):
        self.timer = PhaseTimer()
        self.counters: Dict[str, float] = defaultdict(float)
        self._prom = None
        if prometheus_port is not None and _PROM:
            start_http_server(prometheus_port)
            self._prom = {
                "cycles": Counter("autoscaler_cycles_total", "reconcile cycles"),
                "pods_pending": Gauge("autoscaler_pods_pending", "pending pods last cycle"),
                "pods_scheduled": Counter("autoscaler_pods_scheduled_total", "pods placed by simulation"),
                "scale_ups": Counter("autoscaler_scale_ups_total", "scale-up actuations"),
                "scale_downs": Counter("autoscaler_scale_downs_total", "nodes removed"),
                "watch_events": Counter("autoscaler_watch_events_total",
                                        "pod watch events applied to the delta cache"),
                "decision_latency": Histogram(
                    "autoscaler_decision_latency_seconds",
                    "pod-pending -> decision latency",
                    buckets=(0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1, 5, 30),
                ),
            }

    def count(self, name: str, value: float = 1.0):
        self.counters[name] += value
        if self._prom and name in self._prom:
            obj = self._prom[name]
            if hasattr(obj, "inc"):
                obj.inc(value)

    def gauge(self, name: str, value: float):
        self.counters[name] = value
        if self._prom and name in self._prom:
            obj = self._prom[name]
            if hasattr(obj, "set"):
                obj.set(value)

    def observe_latency(self, seconds: float):
        self.counters["last_decision_latency_s"] = seconds
        if self._prom:
            self._prom["decision_latency"].observe(seconds)
