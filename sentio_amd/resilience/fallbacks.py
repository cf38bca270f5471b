"""Degraded-mode fallbacks
(reference src/core/resilience/fallbacks.py:24-265: SHA-keyed cached
responses, fallback chain primary → fallback fn → cache → default, canned
template responses)."""

from __future__ import annotations

import hashlib
import json
import logging
import os
import threading
from pathlib import Path
from typing import Any, Callable

from sentio_amd.pipeline.prompt_builder import PromptBuilder

logger = logging.getLogger(__name__)


class FallbackManager:
    """Disk-persisted response cache keyed by SHA-256 of the query."""

    def __init__(self, cache_dir: str | None = None):
        self.cache_dir = Path(
            cache_dir or os.path.join(os.path.expanduser("~"), ".cache", "sentio_amd")
        )
        self._lock = threading.Lock()
        self._responses: dict[str, str] = {}
        self._path = self.cache_dir / "responses.json"
        self._load()

    def _load(self) -> None:
        try:
            if self._path.exists():
                self._responses = json.loads(self._path.read_text())
        except Exception:
            self._responses = {}

    def _persist(self) -> None:
        try:
            self.cache_dir.mkdir(parents=True, exist_ok=True)
            self._path.write_text(json.dumps(self._responses))
        except Exception as exc:
            logger.debug("fallback persist failed: %s", exc)

    @staticmethod
    def _key(query: str) -> str:
        return hashlib.sha256(query.encode()).hexdigest()

    def cache_response(self, query: str, response: str) -> None:
        with self._lock:
            self._responses[self._key(query)] = response
            if len(self._responses) > 1000:
                # drop oldest half (insertion order)
                keys = list(self._responses)[: len(self._responses) // 2]
                for k in keys:
                    del self._responses[k]
            self._persist()

    def get_cached_response(self, query: str) -> str | None:
        with self._lock:
            return self._responses.get(self._key(query))

    def with_fallback(self, primary: Callable[[], Any],
                      fallback: Callable[[], Any] | None,
                      query: str, default: str) -> Any:
        """primary → fallback fn → cached response → default
        (reference fallbacks.py:100-159)."""
        try:
            return primary()
        except Exception as exc:
            logger.warning("primary failed (%s); trying fallbacks", exc)
        if fallback is not None:
            try:
                return fallback()
            except Exception:
                pass
        cached = self.get_cached_response(query)
        if cached is not None:
            return cached
        return default


class LLMFallback:
    """Canned template responses (reference fallbacks.py:205-259)."""

    def __init__(self):
        self._builder = PromptBuilder()

    def generate_fallback_response(self, query: str, kind: str = "default") -> str:
        return self._builder.fallback_text(kind)


fallback_manager = FallbackManager()
llm_fallback = LLMFallback()
