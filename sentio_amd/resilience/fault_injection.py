"""Fault injection for resilience testing.

The reference had rich failure HANDLING (circuit breakers, fallbacks —
reference src/core/resilience/patterns.py:65-462) but no fault *injection*
framework.  This module provides one: deterministic, seedable
wrappers that make an engine fail on a schedule, so the degraded modes
(rerank passthrough, canned generation, cached-response fallback, breaker
opening) are testable without real GPU faults.

Usage:
    inj = FaultInjector(fail_every=3)           # every 3rd call raises
    flaky = inj.wrap(reranker, methods=("rerank",))
    # or probabilistic:
    inj = FaultInjector(fail_rate=0.5, seed=7)
"""

from __future__ import annotations

import random
import threading
from typing import Any, Iterable


class InjectedFault(RuntimeError):
    """The exception raised by injected failures (distinguishable from real
    errors in assertions)."""


class FaultInjector:
    def __init__(self, fail_every: int | None = None,
                 fail_rate: float | None = None, seed: int = 0,
                 exception: type[Exception] = InjectedFault):
        if (fail_every is None) == (fail_rate is None):
            raise ValueError("specify exactly one of fail_every / fail_rate")
        self.fail_every = fail_every
        self.fail_rate = fail_rate
        self.exception = exception
        self._rng = random.Random(seed)
        self._calls = 0
        self._injected = 0
        self._lock = threading.Lock()
        self.enabled = True

    def should_fail(self) -> bool:
        with self._lock:
            self._calls += 1
            if not self.enabled:
                return False
            if self.fail_every is not None:
                hit = self._calls % self.fail_every == 0
            else:
                hit = self._rng.random() < self.fail_rate
            if hit:
                self._injected += 1
            return hit

    @property
    def stats(self) -> dict[str, Any]:
        with self._lock:
            return {"calls": self._calls, "injected": self._injected}

    def wrap(self, target: Any, methods: Iterable[str]) -> Any:
        """Return a proxy of `target` whose `methods` raise on schedule."""
        injector = self
        method_set = set(methods)

        class _Proxy:
            def __getattr__(self, name):
                attr = getattr(target, name)
                if name in method_set and callable(attr):
                    def flaky(*args, **kwargs):
                        if injector.should_fail():
                            raise injector.exception(
                                f"injected fault in {name}()")
                        return attr(*args, **kwargs)

                    return flaky
                return attr

        return _Proxy()
