"""Metrics collection with Prometheus exposition
(reference src/observability/metrics.py:99-514: request/embedding/retrieval/
LLM counters + duration histograms, gauges, in-memory fallback, text
exposition for /metrics).  Adds the GPU gauges the reference had no hardware
for: HBM bytes in use, kernel-time histograms, RCCL bytes."""

from __future__ import annotations

import threading
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Any

try:
    from prometheus_client import (
        CollectorRegistry,
        Counter,
        Gauge,
        Histogram,
        generate_latest,
    )

    _HAS_PROM = True
except ImportError:  # pragma: no cover
    _HAS_PROM = False


_DURATION_BUCKETS = (0.01, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0, 30.0, 100.0)


class MetricsCollector:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._counters: dict[str, float] = defaultdict(float)
        self._durations: dict[str, list[float]] = defaultdict(list)
        self._gauges: dict[str, float] = {}
        if _HAS_PROM:
            self.registry = CollectorRegistry()
            self.p_requests = Counter(
                "rag_requests_total", "RAG requests", ["endpoint", "status"],
                registry=self.registry)
            self.p_duration = Histogram(
                "rag_request_duration_seconds", "request durations", ["endpoint"],
                buckets=_DURATION_BUCKETS, registry=self.registry)
            self.p_stage = Histogram(
                "rag_stage_duration_seconds", "pipeline stage durations", ["stage"],
                buckets=_DURATION_BUCKETS, registry=self.registry)
            self.p_tokens = Counter(
                "rag_llm_tokens_total", "generated tokens", ["kind"],
                registry=self.registry)
            self.p_gauge = Gauge(
                "sentio_gauge", "generic gauges", ["name"], registry=self.registry)

    # counters / durations
    def inc(self, name: str, value: float = 1.0, **labels) -> None:
        key = name + "".join(f"|{k}={v}" for k, v in sorted(labels.items()))
        with self._lock:
            self._counters[key] += value
        if _HAS_PROM and name == "rag_requests_total":
            self.p_requests.labels(
                labels.get("endpoint", "?"), labels.get("status", "ok")).inc(value)
        elif _HAS_PROM and name == "rag_llm_tokens_total":
            self.p_tokens.labels(labels.get("kind", "completion")).inc(value)

    def observe(self, name: str, seconds: float, **labels) -> None:
        key = name + "".join(f"|{k}={v}" for k, v in sorted(labels.items()))
        with self._lock:
            lst = self._durations[key]
            lst.append(seconds)
            if len(lst) > 10000:
                del lst[:5000]
        if _HAS_PROM and name == "rag_request_duration_seconds":
            self.p_duration.labels(labels.get("endpoint", "?")).observe(seconds)
        elif _HAS_PROM and name == "rag_stage_duration_seconds":
            self.p_stage.labels(labels.get("stage", "?")).observe(seconds)

    def set_gauge(self, name: str, value: float) -> None:
        with self._lock:
            self._gauges[name] = value
        if _HAS_PROM:
            self.p_gauge.labels(name).set(value)

    @contextmanager
    def track_request(self, endpoint: str):
        t0 = time.perf_counter()
        status = "ok"
        try:
            yield
        except Exception:
            status = "error"
            raise
        finally:
            self.inc("rag_requests_total", endpoint=endpoint, status=status)
            self.observe("rag_request_duration_seconds",
                         time.perf_counter() - t0, endpoint=endpoint)

    # GPU gauges (new vs reference)
    def record_gpu(self) -> None:
        try:
            import torch

            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                self.set_gauge("gpu_hbm_used_bytes", float(total - free))
                self.set_gauge("gpu_hbm_total_bytes", float(total))
        except Exception:
            pass

    # exposition
    def prometheus_text(self) -> str:
        if _HAS_PROM:
            return generate_latest(self.registry).decode()
        return self._fallback_text()

    def _fallback_text(self) -> str:
        lines = []
        with self._lock:
            for k, v in sorted(self._counters.items()):
                lines.append(f"# counter {k} {v}")
            for k, v in sorted(self._gauges.items()):
                lines.append(f"# gauge {k} {v}")
        return "\n".join(lines) + "\n"

    def snapshot(self) -> dict[str, Any]:
        with self._lock:
            summary = {}
            for k, vals in self._durations.items():
                if not vals:
                    continue
                s = sorted(vals)
                summary[k] = {
                    "count": len(s),
                    "p50_ms": s[len(s) // 2] * 1e3,
                    "p95_ms": s[min(len(s) - 1, int(len(s) * 0.95))] * 1e3,
                    "mean_ms": sum(s) / len(s) * 1e3,
                }
            return {
                "counters": dict(self._counters),
                "durations": summary,
                "gauges": dict(self._gauges),
            }


metrics_collector = MetricsCollector()
