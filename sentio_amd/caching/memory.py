"""Thread-safe LRU + TTL cache
(reference src/core/caching/memory_cache.py:36-364 semantics: OrderedDict
LRU, per-item TTL, periodic cleanup, hit/miss/eviction stats, glob
clear_pattern, typed embedding/query helpers with SHA keys)."""

from __future__ import annotations

import fnmatch
import hashlib
import threading
import time
from collections import OrderedDict
from dataclasses import dataclass
from typing import Any


@dataclass
class CacheItem:
    value: Any
    expires_at: float | None


class MemoryCache:
    def __init__(self, max_size: int = 10000, default_ttl: float | None = 300.0):
        self.max_size = max_size
        self.default_ttl = default_ttl
        self._data: OrderedDict[str, CacheItem] = OrderedDict()
        self._lock = threading.RLock()
        self.hits = 0
        self.misses = 0
        self.evictions = 0

    def get(self, key: str) -> Any | None:
        with self._lock:
            item = self._data.get(key)
            if item is None:
                self.misses += 1
                return None
            if item.expires_at is not None and time.time() > item.expires_at:
                del self._data[key]
                self.misses += 1
                return None
            self._data.move_to_end(key)
            self.hits += 1
            return item.value

    def set(self, key: str, value: Any, ttl: float | None = None) -> None:
        ttl = ttl if ttl is not None else self.default_ttl
        expires = time.time() + ttl if ttl is not None else None
        with self._lock:
            if key in self._data:
                self._data.move_to_end(key)
            self._data[key] = CacheItem(value, expires)
            while len(self._data) > self.max_size:
                self._data.popitem(last=False)
                self.evictions += 1

    def delete(self, key: str) -> bool:
        with self._lock:
            return self._data.pop(key, None) is not None

    def clear(self) -> None:
        with self._lock:
            self._data.clear()

    def clear_pattern(self, pattern: str) -> int:
        with self._lock:
            keys = [k for k in self._data if fnmatch.fnmatch(k, pattern)]
            for k in keys:
                del self._data[k]
            return len(keys)

    def cleanup_expired(self) -> int:
        now = time.time()
        with self._lock:
            expired = [
                k for k, it in self._data.items()
                if it.expires_at is not None and now > it.expires_at
            ]
            for k in expired:
                del self._data[k]
            return len(expired)

    def __len__(self) -> int:
        with self._lock:
            return len(self._data)

    def stats(self) -> dict[str, Any]:
        with self._lock:
            total = self.hits + self.misses
            return {
                "size": len(self._data),
                "max_size": self.max_size,
                "hits": self.hits,
                "misses": self.misses,
                "evictions": self.evictions,
                "hit_rate": self.hits / total if total else 0.0,
            }

    # typed helpers (reference memory_cache.py SHA-based keys)
    @staticmethod
    def _key(prefix: str, text: str) -> str:
        return f"{prefix}:{hashlib.sha256(text.encode()).hexdigest()}"

    def get_embedding(self, text: str):
        return self.get(self._key("emb", text))

    def set_embedding(self, text: str, vec, ttl: float | None = None) -> None:
        self.set(self._key("emb", text), vec, ttl)

    def get_query(self, query: str):
        return self.get(self._key("query", query))

    def set_query(self, query: str, response, ttl: float | None = None) -> None:
        self.set(self._key("query", query), response, ttl)
