"""Lightweight span tracing
(reference src/observability/tracing.py:87-265 capability: operation spans
via context manager / decorator; OpenTelemetry is optional there and absent
here, so spans record into an in-process ring buffer that /info and tests
can inspect; HIP kernel timing feeds the same spans via ops timers)."""

from __future__ import annotations

import functools
import threading
import time
from collections import deque
from contextlib import contextmanager
from typing import Any

_spans: deque = deque(maxlen=2048)
_lock = threading.Lock()
_enabled = True


def set_enabled(v: bool) -> None:
    global _enabled
    _enabled = v


@contextmanager
def trace_operation(name: str, **attrs: Any):
    if not _enabled:
        yield None
        return
    t0 = time.perf_counter()
    err = None
    try:
        yield None
    except Exception as exc:
        err = repr(exc)
        raise
    finally:
        with _lock:
            _spans.append({
                "name": name,
                "start": t0,
                "duration_ms": (time.perf_counter() - t0) * 1e3,
                "error": err,
                **attrs,
            })


def trace_function(fn):
    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        with trace_operation(fn.__qualname__):
            return fn(*args, **kwargs)

    return wrapper


def recent_spans(limit: int = 100) -> list[dict[str, Any]]:
    with _lock:
        return list(_spans)[-limit:]


def clear_spans() -> None:
    with _lock:
        _spans.clear()
