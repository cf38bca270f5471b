"""Cache manager with pluggable backends
(reference src/core/caching/cache_manager.py:77-381: memory | multi_tier with
L1/L2 promotion, graceful degradation, global singleton via CACHE_BACKEND).
The reference's L2 was Redis; with no external services in the target
deployment the L2 tier is a larger, longer-TTL in-process cache — the
promotion semantics are preserved."""

from __future__ import annotations

import threading
from typing import Any

from sentio_amd.caching.memory import MemoryCache


class CacheManager:
    def __init__(self, backend: str = "memory", l1_size: int = 4096,
                 l2_size: int = 65536, l1_ttl: float = 300.0,
                 l2_ttl: float = 3600.0):
        self.backend = backend
        self.l1 = MemoryCache(max_size=l1_size, default_ttl=l1_ttl)
        self.l2 = MemoryCache(max_size=l2_size, default_ttl=l2_ttl) \
            if backend == "multi_tier" else None

    def get(self, key: str) -> Any | None:
        v = self.l1.get(key)
        if v is not None:
            return v
        if self.l2 is not None:
            v = self.l2.get(key)
            if v is not None:
                self.l1.set(key, v)  # promote L2 → L1
            return v
        return None

    def set(self, key: str, value: Any, ttl: float | None = None) -> None:
        self.l1.set(key, value, ttl)
        if self.l2 is not None:
            self.l2.set(key, value, ttl)

    def delete(self, key: str) -> None:
        self.l1.delete(key)
        if self.l2 is not None:
            self.l2.delete(key)

    def clear(self) -> None:
        self.l1.clear()
        if self.l2 is not None:
            self.l2.clear()

    def stats(self) -> dict[str, Any]:
        out = {"backend": self.backend, "l1": self.l1.stats()}
        if self.l2 is not None:
            out["l2"] = self.l2.stats()
        return out


_manager: CacheManager | None = None
_lock = threading.Lock()


def get_cache_manager(backend: str | None = None) -> CacheManager:
    global _manager
    with _lock:
        if _manager is None:
            from sentio_amd.config import settings

            _manager = CacheManager(backend or settings.cache_backend)
        return _manager


def reset_cache_manager() -> None:
    global _manager
    with _lock:
        _manager = None
