"""Performance statistics engine.

Capability parity with the reference's PerfStats singleton
(/root/reference/pkg/utils/perf.go:13-335): named start/stop timers, direct
metric recording, min/max/avg/p50/p95/p99 aggregation, a table printer, a
`trace` decorator/context-manager (ref `TraceFunc`, perf.go:288), and
`get_stats`/`reset` for the HTTP API (ref perf.go:296-335).

Additions for the MI355X engine: p50 (the headline metric is agent-turn p50
latency) and engine-phase conventions (`engine_prefill`, `engine_decode_token`,
`engine_collective`, `engine_tokenize`) recorded by opsagent_amd.engine.
"""

from __future__ import annotations

import math
import threading
import time
from contextlib import contextmanager
from typing import Dict, List, Optional


class _Metric:
    __slots__ = ("values",)

    def __init__(self) -> None:
        self.values: List[float] = []


def _percentile(sorted_vals: List[float], pct: float) -> float:
    """Nearest-rank percentile on a pre-sorted list (ref perf.go:190-205)."""
    if not sorted_vals:
        return 0.0
    k = max(0, min(len(sorted_vals) - 1, math.ceil(pct / 100.0 * len(sorted_vals)) - 1))
    return sorted_vals[k]


class PerfStats:
    """Thread-safe named-metric aggregator (values in milliseconds for timers)."""

    def __init__(self, enabled: bool = True, max_samples_per_metric: int = 100_000) -> None:
        self._lock = threading.Lock()
        self._metrics: Dict[str, _Metric] = {}
        self._active: Dict[str, float] = {}
        self.enabled = enabled
        self.max_samples = max_samples_per_metric

    # -- recording ---------------------------------------------------------
    def start_timer(self, name: str) -> None:
        if not self.enabled:
            return
        with self._lock:
            self._active[name] = time.perf_counter()

    def stop_timer(self, name: str) -> float:
        """Stop a named timer and record elapsed ms. Returns elapsed ms (0 if never started)."""
        if not self.enabled:
            return 0.0
        now = time.perf_counter()
        with self._lock:
            t0 = self._active.pop(name, None)
            if t0 is None:
                return 0.0
            ms = (now - t0) * 1000.0
            self._record_locked(name, ms)
            return ms

    def record_metric(self, name: str, value: float) -> None:
        if not self.enabled:
            return
        with self._lock:
            self._record_locked(name, value)

    def _record_locked(self, name: str, value: float) -> None:
        m = self._metrics.get(name)
        if m is None:
            m = _Metric()
            self._metrics[name] = m
        if len(m.values) < self.max_samples:
            m.values.append(value)

    @contextmanager
    def trace(self, name: str):
        """Context manager form of ref TraceFunc (perf.go:288)."""
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.enabled:
                self.record_metric(name, (time.perf_counter() - t0) * 1000.0)

    # -- querying ----------------------------------------------------------
    def get_metric_stats(self, name: str) -> Optional[dict]:
        with self._lock:
            m = self._metrics.get(name)
            if m is None or not m.values:
                return None
            vals = sorted(m.values)
            n = len(vals)
            return {
                "count": n,
                "min": vals[0],
                "max": vals[-1],
                "avg": sum(vals) / n,
                "p50": _percentile(vals, 50.0),
                "p95": _percentile(vals, 95.0),
                "p99": _percentile(vals, 99.0),
            }

    def get_stats(self) -> Dict[str, dict]:
        with self._lock:
            names = list(self._metrics.keys())
        out = {}
        for name in sorted(names):
            s = self.get_metric_stats(name)
            if s is not None:
                out[name] = s
        return out

    def reset(self) -> None:
        with self._lock:
            self._metrics.clear()
            self._active.clear()

    def format_table(self) -> str:
        stats = self.get_stats()
        if not stats:
            return "(no metrics recorded)"
        hdr = f"{'metric':<40} {'count':>7} {'min':>10} {'avg':>10} {'p50':>10} {'p95':>10} {'p99':>10} {'max':>10}"
        lines = [hdr, "-" * len(hdr)]
        for name, s in stats.items():
            lines.append(
                f"{name:<40} {s['count']:>7} {s['min']:>10.2f} {s['avg']:>10.2f} "
                f"{s['p50']:>10.2f} {s['p95']:>10.2f} {s['p99']:>10.2f} {s['max']:>10.2f}"
            )
        return "\n".join(lines)


def start_auto_reset(stats: "PerfStats", interval_s: float) -> threading.Event:
    """Reset `stats` every interval_s seconds (ref perf.reset_interval config,
    configs/config.yaml:19). Returns a stop Event."""
    stop = threading.Event()

    def _loop():
        while not stop.wait(interval_s):
            stats.reset()

    t = threading.Thread(target=_loop, name="perf-auto-reset", daemon=True)
    t.start()
    return stop


_global_stats: Optional[PerfStats] = None
_global_lock = threading.Lock()


def get_perf_stats() -> PerfStats:
    """Process-wide singleton (ref perf.go:33-45)."""
    global _global_stats
    if _global_stats is None:
        with _global_lock:
            if _global_stats is None:
                _global_stats = PerfStats()
    return _global_stats
