"""Lightweight span tracing with OTLP export
(reference src/observability/tracing.py:87-265 capability: operation spans
via context manager / decorator, shipped to a Jaeger/OTLP collector).  The
opentelemetry SDK is not installed in this image, so spans record into an
in-process ring buffer that /info and tests can inspect, and an OTLP/HTTP
**JSON** exporter (the OTLP spec's official JSON encoding — no SDK needed)
ships them to any collector at `OTLP_ENDPOINT` (e.g.
http://collector:4318/v1/traces), either on demand (flush_otlp) or via the
background flusher start_otlp_exporter() the app starts when the env var is
set.  HIP kernel timing feeds the same spans via ops timers."""

from __future__ import annotations

import functools
import json
import os
import threading
import time
import urllib.request
from collections import deque
from contextlib import contextmanager
from typing import Any

_spans: deque = deque(maxlen=2048)
_lock = threading.Lock()
_enabled = True
_exported_until = 0.0     # start timestamp watermark for the OTLP flusher
_epoch_delta = time.time() - time.perf_counter()   # perf_counter → unix ns


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


# ---------------- OTLP/HTTP JSON export ----------------

def _otlp_payload(spans: list[dict[str, Any]]) -> dict[str, Any]:
    """OTLP ExportTraceServiceRequest in the spec's JSON encoding."""
    out = []
    for i, s in enumerate(spans):
        start_ns = int((s["start"] + _epoch_delta) * 1e9)
        end_ns = start_ns + int(s["duration_ms"] * 1e6)
        attrs = [{"key": k, "value": {"stringValue": str(v)}}
                 for k, v in s.items()
                 if k not in ("name", "start", "duration_ms", "error")]
        span = {
            "traceId": f"{(start_ns ^ 0x5e9710) & ((1 << 128) - 1):032x}",
            "spanId": f"{(start_ns + i) & ((1 << 64) - 1):016x}",
            "name": s["name"],
            "kind": 1,  # SPAN_KIND_INTERNAL
            "startTimeUnixNano": str(start_ns),
            "endTimeUnixNano": str(end_ns),
            "attributes": attrs,
            "status": ({"code": 2, "message": s["error"]}
                       if s.get("error") else {"code": 1}),
        }
        out.append(span)
    return {
        "resourceSpans": [{
            "resource": {"attributes": [
                {"key": "service.name",
                 "value": {"stringValue": "sentio-amd"}}]},
            "scopeSpans": [{
                "scope": {"name": "sentio_amd.tracing"},
                "spans": out,
            }],
        }]
    }


def flush_otlp(endpoint: str | None = None, timeout_s: float = 5.0) -> int:
    """POST spans recorded since the last flush to an OTLP/HTTP collector
    (…/v1/traces).  Returns the number of spans shipped (0 = nothing new
    or no endpoint configured)."""
    global _exported_until
    endpoint = endpoint or os.environ.get("OTLP_ENDPOINT", "")
    if not endpoint:
        return 0
    with _lock:
        pending = [s for s in _spans if s["start"] > _exported_until]
        if not pending:
            return 0
        watermark = max(s["start"] for s in pending)
    body = json.dumps(_otlp_payload(pending)).encode()
    req = urllib.request.Request(
        endpoint, data=body,
        headers={"Content-Type": "application/json"}, method="POST")
    with urllib.request.urlopen(req, timeout=timeout_s) as resp:
        resp.read()
        if resp.status >= 300:
            raise RuntimeError(f"OTLP export failed: HTTP {resp.status}")
    _exported_until = watermark
    return len(pending)


_flusher: threading.Thread | None = None
_flusher_stop = threading.Event()


def start_otlp_exporter(endpoint: str | None = None,
                        interval_s: float = 10.0) -> bool:
    """Background periodic flusher (the reference's BatchSpanProcessor
    role, tracing.py:141-146).  No-op without an endpoint."""
    global _flusher
    endpoint = endpoint or os.environ.get("OTLP_ENDPOINT", "")
    if not endpoint or (_flusher is not None and _flusher.is_alive()):
        return False

    def loop():
        while not _flusher_stop.wait(interval_s):
            try:
                flush_otlp(endpoint)
            except Exception:
                pass   # collector down: keep buffering, retry next tick

    _flusher_stop.clear()
    _flusher = threading.Thread(target=loop, daemon=True,
                                name="sentio-otlp-exporter")
    _flusher.start()
    return True


def stop_otlp_exporter() -> None:
    global _flusher
    _flusher_stop.set()
    if _flusher is not None:
        _flusher.join(timeout=2.0)
        _flusher = None
