"""Periodic component health checker
(reference src/core/resilience/patterns.py:252-306: HealthChecker with a
background loop, per-component check callables, global singleton
`health_checker` at patterns.py:466).

Here the registered checks are GPU-engine probes (HIP device health, index
residency, breaker states) instead of HTTP pings; the loop/threshold/
status-aggregation semantics match the reference."""

from __future__ import annotations

import threading
import time
from typing import Any, Callable


class HealthChecker:
    def __init__(self, interval_s: float = 30.0, unhealthy_threshold: int = 3):
        self.interval_s = interval_s
        self.unhealthy_threshold = unhealthy_threshold
        self._checks: dict[str, Callable[[], bool]] = {}
        self._failures: dict[str, int] = {}
        self._last_result: dict[str, dict[str, Any]] = {}
        self._lock = threading.Lock()
        self._thread: threading.Thread | None = None
        self._stop = threading.Event()

    def register(self, name: str, check: Callable[[], bool]) -> None:
        with self._lock:
            self._checks[name] = check
            self._failures.setdefault(name, 0)

    def unregister(self, name: str) -> None:
        with self._lock:
            self._checks.pop(name, None)
            self._failures.pop(name, None)
            self._last_result.pop(name, None)

    def run_checks(self) -> dict[str, dict[str, Any]]:
        """Run every registered check once (also called by the loop)."""
        with self._lock:
            checks = dict(self._checks)
        results: dict[str, dict[str, Any]] = {}
        for name, fn in checks.items():
            t0 = time.perf_counter()
            try:
                ok = bool(fn())
                err = None
            except Exception as e:  # a failing probe is a failed check
                ok, err = False, str(e)
            with self._lock:
                if ok:
                    self._failures[name] = 0
                else:
                    self._failures[name] = self._failures.get(name, 0) + 1
                results[name] = {
                    "healthy": ok,
                    "consecutive_failures": self._failures[name],
                    "unhealthy": self._failures[name] >= self.unhealthy_threshold,
                    "latency_ms": round((time.perf_counter() - t0) * 1e3, 2),
                    "error": err,
                    "checked_at": time.time(),
                }
                self._last_result[name] = results[name]
        return results

    def status(self) -> dict[str, Any]:
        with self._lock:
            unhealthy = [n for n, r in self._last_result.items() if r["unhealthy"]]
            return {
                "healthy": not unhealthy,
                "unhealthy_components": unhealthy,
                "components": dict(self._last_result),
                "running": self._thread is not None and self._thread.is_alive(),
            }

    # ----- background loop -----
    def start(self) -> None:
        if self._thread is not None and self._thread.is_alive():
            return
        self._stop.clear()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="sentio-health")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None

    def _loop(self) -> None:
        while not self._stop.wait(self.interval_s):
            self.run_checks()


health_checker = HealthChecker()
