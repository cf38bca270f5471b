"""HIP-event region timers feeding the metrics histograms.

SURVEY §5 (tracing): the reference instruments only the HTTP layer
(reference src/observability/tracing.py:181-265, decorator unused); the
MI355X equivalent adds device-side timers.  `KernelTimer` brackets a code
region with `torch.cuda.Event(enable_timing=True)` pairs so the measured
span is GPU time, not host wall-clock, and — critically for the serving hot
path — it NEVER synchronizes inside the region: event pairs are queued and
resolved lazily on later calls (only once `event.query()` reports the pair
complete) or explicitly via `flush()`.

On CPU (tests, non-GPU deployments) it degrades to `perf_counter`.

Usage:
    timer = KernelTimer("decode_step")
    with timer.measure():
        ... launches ...
    timer.flush()     # optional: force-resolve pending pairs (syncs)

Resolved durations go to ``metrics_collector.observe("gpu_region_seconds",
t, region=name)`` and are visible in /metrics.
"""

from __future__ import annotations

import time
from collections import deque
from contextlib import contextmanager
from threading import Lock

import torch

from sentio_amd.observability.metrics import metrics_collector


class KernelTimer:
    """Non-blocking GPU region timer (lazy HIP-event resolution)."""

    def __init__(self, region: str, max_pending: int = 256):
        self.region = region
        self._pending: deque = deque()
        self._lock = Lock()
        self._max_pending = max_pending
        self.count = 0
        self.total_s = 0.0
        self.last_s = 0.0

    @contextmanager
    def measure(self):
        if torch.cuda.is_available():
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            try:
                yield
            finally:
                end.record()
                with self._lock:
                    self._pending.append((start, end))
                self._drain(block=False)
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._record(time.perf_counter() - t0)

    def _drain(self, block: bool) -> None:
        """Resolve completed event pairs; only syncs when over capacity or
        when `block` is requested."""
        with self._lock:
            while self._pending:
                start, end = self._pending[0]
                over = len(self._pending) > self._max_pending
                if not (block or over) and not end.query():
                    break
                end.synchronize()
                self._pending.popleft()
                self._record(start.elapsed_time(end) / 1e3)

    def _record(self, seconds: float) -> None:
        self.count += 1
        self.total_s += seconds
        self.last_s = seconds
        metrics_collector.observe("gpu_region_seconds", seconds,
                                  region=self.region)

    def flush(self) -> None:
        """Force-resolve all pending pairs (synchronizes on them)."""
        self._drain(block=True)

    @property
    def mean_s(self) -> float:
        return self.total_s / self.count if self.count else 0.0


_timers: dict[str, KernelTimer] = {}
_timers_lock = Lock()


def get_timer(region: str) -> KernelTimer:
    """Process-wide named timer registry (one histogram series per region)."""
    with _timers_lock:
        t = _timers.get(region)
        if t is None:
            t = _timers[region] = KernelTimer(region)
        return t


def timer_snapshot() -> dict[str, dict[str, float]]:
    """Resolved stats for every registered region (for /metrics/performance)."""
    with _timers_lock:
        items = list(_timers.items())
    out = {}
    for name, t in items:
        t.flush()
        out[name] = {"count": t.count, "total_s": t.total_s,
                     "mean_s": t.mean_s, "last_s": t.last_s}
    return out
