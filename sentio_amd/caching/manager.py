"""Cache manager with pluggable backends
(reference src/core/caching/cache_manager.py:77-381: memory | redis |
multi_tier with L1/L2 promotion, graceful degradation, global singleton via
CACHE_BACKEND).  Backends here:

* ``memory``     — L1 only (in-process LRU+TTL).
* ``multi_tier`` — L1 memory + L2 larger/longer-TTL memory tier with
  promotion (reference cache_manager.py:108-125 semantics).
* ``disk``       — L1 memory + L2 persistent on-disk tier (the reference's
  Redis L2 role: survives restarts; see caching/disk.py).

An optional eviction/admission ``strategy`` (caching/strategies.py) gates
what enters L1 and with which TTL — unlike the reference, where the
strategies module existed but was never wired in."""

from __future__ import annotations

import threading
from typing import Any

from sentio_amd.caching.disk import DiskCache
from sentio_amd.caching.memory import MemoryCache
from sentio_amd.caching.strategies import CacheStrategy


class CacheManager:
    def __init__(self, backend: str = "memory", l1_size: int = 4096,
                 l2_size: int = 65536, l1_ttl: float = 300.0,
                 l2_ttl: float = 3600.0, strategy: CacheStrategy | None = None,
                 disk_dir: str | None = None):
        self.backend = backend
        self.strategy = strategy
        self.l1 = MemoryCache(max_size=l1_size, default_ttl=l1_ttl)
        if backend == "multi_tier":
            self.l2: MemoryCache | DiskCache | None = MemoryCache(
                max_size=l2_size, default_ttl=l2_ttl)
        elif backend == "disk":
            try:
                self.l2 = DiskCache(directory=disk_dir, default_ttl=l2_ttl)
            except OSError:
                # graceful degradation (reference cache_manager.py:77-84:
                # Redis down → memory-only)
                self.backend = "memory"
                self.l2 = None
        else:
            self.l2 = None

    def get(self, key: str) -> Any | None:
        v = self.l1.get(key)
        if v is not None:
            if self.strategy is not None:
                self.strategy.on_hit(key)
            return v
        if self.l2 is not None:
            v = self.l2.get(key)
            if v is not None:
                self.l1.set(key, v)  # promote L2 → L1
            return v
        return None

    def set(self, key: str, value: Any, ttl: float | None = None) -> None:
        if self.strategy is not None:
            if not self.strategy.should_cache(key, value):
                return
            if ttl is None:
                ttl = self.strategy.ttl_for(key, value)
        self.l1.set(key, value, ttl)
        if self.l2 is not None:
            self.l2.set(key, value, ttl)

    def delete(self, key: str) -> None:
        self.l1.delete(key)
        if self.l2 is not None:
            self.l2.delete(key)
        if self.strategy is not None:
            self.strategy.on_evict(key)

    def clear(self) -> None:
        self.l1.clear()
        if self.l2 is not None:
            self.l2.clear()

    def cleanup_expired(self) -> int:
        n = self.l1.cleanup_expired()
        if self.l2 is not None:
            n += self.l2.cleanup_expired()
        return n

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
