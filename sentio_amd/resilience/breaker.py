"""Circuit breaker (reference src/core/resilience/patterns.py:65-142,310-400:
CLOSED/OPEN/HALF_OPEN, failure threshold, recovery timeout, stats).

In the reference these guarded remote HTTP APIs; here they guard the GPU
engines — a HIP error or kernel timeout opens the breaker and the pipeline
degrades (skip rerank / skip verify / BM25-only) exactly like the
reference's per-node fallbacks."""

from __future__ import annotations

import enum
import threading
import time
from typing import Any, Callable


class CircuitState(enum.Enum):
    CLOSED = "closed"
    OPEN = "open"
    HALF_OPEN = "half_open"


class CircuitOpenError(RuntimeError):
    pass


class CircuitBreaker:
    def __init__(self, name: str = "breaker", failure_threshold: int = 5,
                 recovery_timeout: float = 30.0, success_threshold: int = 2):
        self.name = name
        self.failure_threshold = failure_threshold
        self.recovery_timeout = recovery_timeout
        self.success_threshold = success_threshold
        self._state = CircuitState.CLOSED
        self._failures = 0
        self._successes = 0
        self._opened_at = 0.0
        self._lock = threading.RLock()
        self.stats = {"calls": 0, "failures": 0, "rejections": 0, "state_changes": 0}

    @property
    def state(self) -> CircuitState:
        with self._lock:
            if (
                self._state == CircuitState.OPEN
                and time.time() - self._opened_at >= self.recovery_timeout
            ):
                self._transition(CircuitState.HALF_OPEN)
            return self._state

    def _transition(self, new: CircuitState) -> None:
        if new != self._state:
            self._state = new
            self.stats["state_changes"] += 1
            if new == CircuitState.OPEN:
                self._opened_at = time.time()
            if new == CircuitState.HALF_OPEN:
                self._successes = 0
            if new == CircuitState.CLOSED:
                self._failures = 0

    def call(self, fn: Callable, *args, **kwargs) -> Any:
        st = self.state
        with self._lock:
            self.stats["calls"] += 1
            if st == CircuitState.OPEN:
                self.stats["rejections"] += 1
                raise CircuitOpenError(f"circuit '{self.name}' is open")
        try:
            result = fn(*args, **kwargs)
        except Exception:
            self.record_failure()
            raise
        self.record_success()
        return result

    def record_success(self) -> None:
        with self._lock:
            if self._state == CircuitState.HALF_OPEN:
                self._successes += 1
                if self._successes >= self.success_threshold:
                    self._transition(CircuitState.CLOSED)
            else:
                self._failures = 0

    def record_failure(self) -> None:
        with self._lock:
            self.stats["failures"] += 1
            self._failures += 1
            if self._state == CircuitState.HALF_OPEN:
                self._transition(CircuitState.OPEN)
            elif self._failures >= self.failure_threshold:
                self._transition(CircuitState.OPEN)

    def health(self) -> dict[str, Any]:
        return {"name": self.name, "state": self.state.value, **self.stats}
