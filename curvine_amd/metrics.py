"""Metrics registry.

Analog of the reference's `curvine-metrics` crate
(/root/reference/crates/core/curvine-metrics/src/lib.rs:97-160 prometheus
counter/gauge/histogram constructors) plus the per-service metric structs
(master_metrics.rs, worker_metrics.rs, fuse_metrics.rs).

Uses prometheus_client when present; hot paths keep plain dict counters
(OpStats) and only export to prometheus on scrape.
"""
from __future__ import annotations

import threading
import time

try:
    import prometheus_client as prom
except ImportError:  # pragma: no cover
    prom = None

_registry = None


def registry():
    global _registry
    if _registry is None and prom is not None:
        _registry = prom.CollectorRegistry()
    return _registry


class OpStats:
    """Lock-free-ish per-op counters: one instance per thread/channel,
    merged on scrape (fuse_metrics.rs per-op histograms analog)."""

    def __init__(self):
        self.count: dict[str, int] = {}
        self.time_s: dict[str, float] = {}
        self.bytes: dict[str, int] = {}
        self.errors: dict[str, int] = {}

    def record(self, op: str, dt: float, nbytes: int = 0, error: bool = False):
        self.count[op] = self.count.get(op, 0) + 1
        self.time_s[op] = self.time_s.get(op, 0.0) + dt
        if nbytes:
            self.bytes[op] = self.bytes.get(op, 0) + nbytes
        if error:
            self.errors[op] = self.errors.get(op, 0) + 1

    @staticmethod
    def merge(stats: list["OpStats"]) -> dict:
        out: dict[str, dict] = {}
        for s in stats:
            for op, n in s.count.items():
                d = out.setdefault(op, {"count": 0, "time_s": 0.0,
                                        "bytes": 0, "errors": 0})
                d["count"] += n
                d["time_s"] += s.time_s.get(op, 0.0)
                d["bytes"] += s.bytes.get(op, 0)
                d["errors"] += s.errors.get(op, 0)
        for op, d in out.items():
            if d["count"]:
                d["avg_us"] = round(d["time_s"] / d["count"] * 1e6, 1)
            d["time_s"] = round(d["time_s"], 3)
        return out


class SpeedCounter:
    """Windowed throughput meter (runtime SpeedCounter analog)."""

    def __init__(self, window_s: float = 10.0):
        self.window = window_s
        self._events: list[tuple[float, int]] = []
        self._lock = threading.Lock()

    def add(self, nbytes: int) -> None:
        now = time.monotonic()
        with self._lock:
            self._events.append((now, nbytes))
            cutoff = now - self.window
            while self._events and self._events[0][0] < cutoff:
                self._events.pop(0)

    def bytes_per_sec(self) -> float:
        now = time.monotonic()
        with self._lock:
            recent = [(t, n) for t, n in self._events if t > now - self.window]
            if not recent:
                return 0.0
            span = max(1e-6, now - recent[0][0])
            return sum(n for _, n in recent) / span


class TimeSpent:
    """Micro-timer with slow-op warning (runtime TimeSpent +
    io_slow_us analog, read_handler.rs:200-207)."""

    def __init__(self):
        self.t0 = time.perf_counter()

    def used_us(self) -> int:
        return int((time.perf_counter() - self.t0) * 1e6)
