"""Pluggable eviction/admission strategies
(reference src/core/caching/strategies.py:16-343: TTLStrategy, LRUStrategy,
SizeBasedStrategy, AdaptiveStrategy — advisory policy objects).  Unlike the
reference (where the module was never wired in), CacheManager here accepts a
strategy for its L1 tier via `CacheManager(strategy=...)`."""

from __future__ import annotations

import sys
import time
from abc import ABC, abstractmethod
from typing import Any


class CacheStrategy(ABC):
    """Decides admission (should_cache), per-entry TTL, and eviction priority."""

    @abstractmethod
    def should_cache(self, key: str, value: Any) -> bool: ...

    @abstractmethod
    def ttl_for(self, key: str, value: Any) -> float | None: ...

    def on_hit(self, key: str) -> None:  # noqa: B027 - optional hook
        pass

    def on_evict(self, key: str) -> None:  # noqa: B027 - optional hook
        pass


class TTLStrategy(CacheStrategy):
    def __init__(self, ttl: float = 300.0):
        self.ttl = ttl

    def should_cache(self, key: str, value: Any) -> bool:
        return True

    def ttl_for(self, key: str, value: Any) -> float | None:
        return self.ttl


class LRUStrategy(CacheStrategy):
    """Pure LRU: cache everything, never expire by time (size bound evicts)."""

    def should_cache(self, key: str, value: Any) -> bool:
        return True

    def ttl_for(self, key: str, value: Any) -> float | None:
        return None


class SizeBasedStrategy(CacheStrategy):
    """Skip values larger than max_value_bytes (measured via sys.getsizeof
    recursively one level deep — cheap approximation, same as reference)."""

    def __init__(self, max_value_bytes: int = 1 << 20, ttl: float = 600.0):
        self.max_value_bytes = max_value_bytes
        self.ttl = ttl

    def _size(self, value: Any) -> int:
        size = sys.getsizeof(value)
        if isinstance(value, (list, tuple, set)):
            size += sum(sys.getsizeof(v) for v in value)
        elif isinstance(value, dict):
            size += sum(sys.getsizeof(k) + sys.getsizeof(v) for k, v in value.items())
        return size

    def should_cache(self, key: str, value: Any) -> bool:
        return self._size(value) <= self.max_value_bytes

    def ttl_for(self, key: str, value: Any) -> float | None:
        return self.ttl


class AdaptiveStrategy(CacheStrategy):
    """Hit-rate-driven TTL: keys that keep getting hit earn longer TTLs,
    cold keys decay to the base TTL (reference AdaptiveStrategy semantics)."""

    def __init__(self, base_ttl: float = 300.0, max_ttl: float = 3600.0,
                 window_s: float = 600.0):
        self.base_ttl = base_ttl
        self.max_ttl = max_ttl
        self.window_s = window_s
        self._hits: dict[str, list[float]] = {}

    def should_cache(self, key: str, value: Any) -> bool:
        return True

    def on_hit(self, key: str) -> None:
        now = time.time()
        h = self._hits.setdefault(key, [])
        h.append(now)
        cutoff = now - self.window_s
        while h and h[0] < cutoff:
            h.pop(0)

    def on_evict(self, key: str) -> None:
        self._hits.pop(key, None)

    def ttl_for(self, key: str, value: Any) -> float | None:
        hits = len(self._hits.get(key, ()))
        scale = min(1.0 + hits / 4.0, self.max_ttl / self.base_ttl)
        return self.base_ttl * scale
