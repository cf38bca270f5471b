"""Request handlers (reference src/api/handlers/chat.py:25-274 and
health.py:20-344 capability): chat orchestration with the fallback chain
cached-response → template → apology, plus health probes."""

from __future__ import annotations

import logging
import time
import uuid
from typing import Any

from sentio_amd.caching.manager import CacheManager
from sentio_amd.models.document import Document
from sentio_amd.pipeline.state import create_initial_state
from sentio_amd.resilience.fallbacks import fallback_manager, llm_fallback
from sentio_amd.serving.container import ServiceContainer

logger = logging.getLogger(__name__)


class ChatHandler:
    def __init__(self, container: ServiceContainer):
        self.container = container
        self.cache: CacheManager = container.cache_manager()

    def process(self, question: str, top_k: int | None = None,
                temperature: float | None = None,
                history: list[dict[str, str]] | None = None) -> dict[str, Any]:
        query_id = str(uuid.uuid4())
        meta: dict[str, Any] = {"query_id": query_id}
        if top_k is not None:
            meta["user_top_k"] = int(top_k)
        if temperature is not None:
            meta["temperature"] = float(temperature)
        if history:
            meta["history"] = history

        cached = self.cache.l1.get_query(question)
        try:
            state = create_initial_state(question, meta)
            state = self.container.pipeline().invoke(state)
            answer = state.get("response", "")
            if not answer:
                raise RuntimeError("empty response from pipeline")
            sources = self._shape_sources(state.get("selected_documents") or
                                          state.get("retrieved_documents") or [])
            result = {
                "answer": answer,
                "sources": sources,
                "metadata": {
                    "query_id": query_id,
                    "pipeline_ms": state.get("metadata", {}).get("pipeline_ms"),
                    "retrieved_count": state.get("metadata", {}).get("retrieved_count"),
                    "selected_count": state.get("metadata", {}).get("selected_count"),
                    "verification": state.get("evaluation", {}).get("verification"),
                },
            }
            self.cache.l1.set_query(question, result)
            fallback_manager.cache_response(question, answer)
            self.container.heartbeat.beat()
            return result
        except Exception as exc:
            logger.error("chat pipeline failed: %s", exc)
            # fallback chain (reference chat.py:195-239)
            if cached is not None:
                cached = dict(cached)
                cached.setdefault("metadata", {})["fallback"] = "cached"
                return cached
            disk_cached = fallback_manager.get_cached_response(question)
            if disk_cached:
                return {"answer": disk_cached, "sources": [],
                        "metadata": {"query_id": query_id, "fallback": "disk_cache"}}
            template = llm_fallback.generate_fallback_response(question, "error")
            return {"answer": template, "sources": [],
                    "metadata": {"query_id": query_id, "fallback": "template",
                                 "error": str(exc)}}

    @staticmethod
    def _shape_sources(docs: list[Document]) -> list[dict[str, Any]]:
        out = []
        for d in docs:
            raw = d.metadata.get("score", 0.0)
            try:
                score = max(0.0, min(1.0, float(raw)))
            except (TypeError, ValueError):
                score = 0.0
            out.append({
                "text": (d.text or "")[:1000],
                "source": str(d.metadata.get("source", d.id)),
                "score": score,
                "metadata": {k: v for k, v in d.metadata.items()
                             if k not in ("content",)},
            })
        return out

    def probe(self, timeout_s: float = 5.0) -> bool:
        """Health probe: run the retriever stage only (reference chat.py:241-274)."""
        try:
            t0 = time.time()
            self.container.retriever().retrieve("health probe", top_k=1)
            return (time.time() - t0) <= timeout_s
        except Exception:
            return False


class HealthHandler:
    def __init__(self, container: ServiceContainer):
        self.container = container
        self._cache: tuple[float, dict] | None = None
        self._cache_ttl = 10.0  # reference health.py:28-30

    def basic(self) -> dict[str, Any]:
        return {
            "status": "healthy",
            "timestamp": time.time(),
            "version": __import__("sentio_amd").__version__,
            "services": {"engine": "ok", "device": self.container.device},
        }

    def detailed(self) -> dict[str, Any]:
        now = time.time()
        if self._cache and now - self._cache[0] < self._cache_ttl:
            return self._cache[1]
        from sentio_amd.observability.monitoring import resource_monitor

        checks: dict[str, Any] = {}
        try:
            self.container.encoder().embed(["health"])
            checks["encoder"] = "ok"
        except Exception as exc:
            checks["encoder"] = f"error: {exc}"
        checks["index"] = {"dense": len(self.container.dense_index()),
                           "bm25": self.container.bm25_index().n_docs}
        checks["breakers"] = {k: b.health()["state"]
                              for k, b in self.container.breakers.items()}
        checks["heartbeat"] = self.container.heartbeat.snapshot()
        # request/engine coalescing evidence (dynamic + micro batchers)
        gen = self.container._cache.get("generator_frontend")
        if gen is not None and hasattr(gen, "batcher"):
            checks["batcher"] = gen.batcher.health()
        checks["micro_batchers"] = {
            k: m.health() for k, m in (
                (k, getattr(self.container._cache.get(k), "micro", None))
                for k in ("encoder", "reranker"))
            if m is not None
        }
        status = "healthy" if checks.get("encoder") == "ok" else "degraded"
        result = {
            "status": status,
            "timestamp": now,
            "checks": checks,
            "periodic": self.container.health_checker().status(),
            "resources": resource_monitor.snapshot(),
        }
        self._cache = (now, result)
        return result

    def ready(self) -> bool:
        try:
            self.container.pipeline()
            return True
        except Exception:
            return False

    def live(self) -> bool:
        return True
