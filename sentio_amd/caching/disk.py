"""Disk-backed L2 cache (reference src/core/caching/redis_cache.py:90-439
capability: out-of-process persistent tier with TTL, zlib compression for
payloads >1000 B, batch ops, typed embedding/query helpers).  The target
deployment has no Redis server, so L2 persistence is a local directory —
same interface, same compression threshold, survives process restarts
(which is the property the reference used Redis for)."""

from __future__ import annotations

import base64
import hashlib
import json
import os
import threading
import time
import zlib
from typing import Any

_COMPRESS_MIN = 1000
_MAGIC_Z = b"SZ1"
_MAGIC_P = b"SP1"


def _encode(v: Any) -> Any:
    """Tagged-JSON encoding of the typed cache payloads (response dicts,
    embedding vectors).  Deliberately NOT pickle: cache files must never be
    able to execute code in the serving process (a writable cache dir would
    otherwise be an RCE vector — ADVICE r1)."""
    import numpy as np
    import torch

    if isinstance(v, torch.Tensor):
        v = v.detach().cpu().numpy()
    if isinstance(v, np.ndarray):
        return {"__nd__": base64.b64encode(v.tobytes()).decode(),
                "dtype": str(v.dtype), "shape": list(v.shape)}
    if isinstance(v, (bytes, bytearray)):
        return {"__b__": base64.b64encode(bytes(v)).decode()}
    if isinstance(v, dict):
        return {k: _encode(x) for k, x in v.items()}
    if isinstance(v, (list, tuple)):
        return [_encode(x) for x in v]
    if isinstance(v, (str, int, float, bool)) or v is None:
        return v
    if isinstance(v, (np.integer,)):
        return int(v)
    if isinstance(v, (np.floating,)):
        return float(v)
    raise TypeError(f"DiskCache cannot serialize {type(v).__name__}")


def _decode(v: Any) -> Any:
    import numpy as np

    if isinstance(v, dict):
        if "__nd__" in v and "dtype" in v:
            return np.frombuffer(
                base64.b64decode(v["__nd__"]), dtype=np.dtype(v["dtype"])
            ).reshape(v["shape"]).copy()
        if "__b__" in v and len(v) == 1:
            return base64.b64decode(v["__b__"])
        return {k: _decode(x) for k, x in v.items()}
    if isinstance(v, list):
        return [_decode(x) for x in v]
    return v


class DiskCache:
    def __init__(self, directory: str | None = None, default_ttl: float = 3600.0,
                 max_entries: int = 100_000):
        self.dir = directory or os.path.join(
            os.path.expanduser("~"), ".cache", "sentio_amd", "l2")
        os.makedirs(self.dir, mode=0o700, exist_ok=True)
        try:
            os.chmod(self.dir, 0o700)   # pre-existing dir: tighten anyway
        except OSError:
            pass
        self.default_ttl = default_ttl
        self.max_entries = max_entries
        self._lock = threading.Lock()
        self._hits = 0
        self._misses = 0

    def _path(self, key: str) -> str:
        h = hashlib.sha256(key.encode()).hexdigest()
        return os.path.join(self.dir, h[:2], h)

    def get(self, key: str) -> Any | None:
        p = self._path(key)
        try:
            with open(p, "rb") as f:
                expires = float(f.readline())
                if expires and time.time() > expires:
                    os.unlink(p)
                    self._misses += 1
                    return None
                blob = f.read()
        except (OSError, ValueError):
            self._misses += 1
            return None
        self._hits += 1
        if blob[:3] == _MAGIC_Z:
            blob = zlib.decompress(blob[3:])
        elif blob[:3] == _MAGIC_P:
            blob = blob[3:]
        else:
            return None   # unknown/legacy (pickle-era) format: treat as miss
        try:
            return _decode(json.loads(blob.decode()))
        except (ValueError, UnicodeDecodeError):
            return None

    def set(self, key: str, value: Any, ttl: float | None = None) -> None:
        p = self._path(key)
        os.makedirs(os.path.dirname(p), mode=0o700, exist_ok=True)
        blob = json.dumps(_encode(value)).encode()
        if len(blob) > _COMPRESS_MIN:
            blob = _MAGIC_Z + zlib.compress(blob)
        else:
            blob = _MAGIC_P + blob
        expires = time.time() + (ttl if ttl is not None else self.default_ttl)
        tmp = p + f".tmp{os.getpid()}"
        with open(tmp, "wb") as f:
            f.write(f"{expires}\n".encode())
            f.write(blob)
        os.replace(tmp, p)   # atomic publish

    def delete(self, key: str) -> None:
        try:
            os.unlink(self._path(key))
        except OSError:
            pass

    def clear(self) -> None:
        with self._lock:
            for root, _dirs, files in os.walk(self.dir):
                for name in files:
                    try:
                        os.unlink(os.path.join(root, name))
                    except OSError:
                        pass

    def cleanup_expired(self) -> int:
        """Drop expired entries; returns the number removed."""
        removed = 0
        now = time.time()
        for root, _dirs, files in os.walk(self.dir):
            for name in files:
                p = os.path.join(root, name)
                try:
                    with open(p, "rb") as f:
                        expires = float(f.readline())
                    if expires and now > expires:
                        os.unlink(p)
                        removed += 1
                except (OSError, ValueError):
                    pass
        return removed

    def stats(self) -> dict[str, Any]:
        n = sum(len(files) for _r, _d, files in os.walk(self.dir))
        total = self._hits + self._misses
        return {
            "backend": "disk",
            "dir": self.dir,
            "entries": n,
            "hits": self._hits,
            "misses": self._misses,
            "hit_rate": self._hits / total if total else 0.0,
        }
