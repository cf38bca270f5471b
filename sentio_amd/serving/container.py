"""Service container — lazy singletons wiring engines, indexes, retriever and
pipeline (reference src/core/dependencies.py:24-223 capability).  The
reference wired HTTP clients; this wires GPU engines, so initialization
order is: device → encoder → indexes → retriever → reranker → generator →
pipeline → ingestor → handlers."""

from __future__ import annotations

import logging
import threading
from typing import Any

import torch

from sentio_amd.caching.manager import CacheManager
from sentio_amd.config import Settings, settings as global_settings
from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.ingest.chunker import TextChunker
from sentio_amd.ingest.ingestor import DocumentIngestor
from sentio_amd.pipeline.graph import GraphConfig, RagPipeline, build_basic_graph
from sentio_amd.pipeline.verifier import AnswerVerifier
from sentio_amd.resilience.breaker import CircuitBreaker
from sentio_amd.retrieval.factory import create_retriever
from sentio_amd.utils.auth import AuthManager

logger = logging.getLogger(__name__)


class ServiceContainer:
    def __init__(self, settings: Settings | None = None):
        self.settings = settings or global_settings
        self.device = self.settings.resolve_device()
        self._lock = threading.RLock()
        self._cache: dict[str, Any] = {}
        self.breakers = {
            "encoder": CircuitBreaker("encoder"),
            "reranker": CircuitBreaker("reranker"),
            "generator": CircuitBreaker("generator"),
        }
        from sentio_amd.resilience.gpu_health import RankHeartbeat
        self.heartbeat = RankHeartbeat()

    def _get(self, name: str, factory) -> Any:
        with self._lock:
            if name not in self._cache:
                self._cache[name] = factory()
            return self._cache[name]

    # --- engines ---
    def encoder(self):
        def make():
            if self.settings.mock_compute or self.device == "cpu":
                from sentio_amd.engines.mock import MockEncoderEngine

                return MockEncoderEngine(dim=self.settings.embedding_dim,
                                         device=self.device)
            from sentio_amd.engines.encoder import EncoderEngine

            eng = EncoderEngine(self.settings.encoder_model, device=self.device,
                                dtype=self.settings.compute_dtype,
                                cache_size=self.settings.embedding_cache_size,
                                cache_ttl=self.settings.embedding_cache_ttl_s)
            if self.settings.dynamic_batching:
                # concurrent requests' single-query embeds coalesce into one
                # encoder forward (the GPU load test showed per-request
                # retrieval serializing between generation batches)
                from sentio_amd.serving.batcher import BatchedEncoder

                return BatchedEncoder(eng)
            return eng

        return self._get("encoder", make)

    def reranker(self):
        def make():
            if self.settings.mock_compute or self.device == "cpu":
                from sentio_amd.engines.mock import MockRerankerEngine

                return MockRerankerEngine()
            from sentio_amd.engines.reranker import RerankerEngine

            eng = RerankerEngine(self.settings.reranker_model,
                                 device=self.device,
                                 dtype=self.settings.compute_dtype)
            if self.settings.dynamic_batching:
                from sentio_amd.serving.batcher import BatchedReranker

                return BatchedReranker(eng)
            return eng

        return self._get("reranker", make)

    def generator(self):
        def make():
            if self.settings.mock_compute or self.device == "cpu":
                from sentio_amd.engines.mock import MockGeneratorEngine

                return MockGeneratorEngine()
            from sentio_amd.engines.generator import GeneratorEngine

            return GeneratorEngine(self.settings.generator_model, device=self.device,
                                   dtype=self.settings.compute_dtype,
                                   max_seq=self.settings.kv_cache_max_tokens)

        return self._get("generator", make)

    def generator_frontend(self):
        """The generator the pipeline sees: wraps the raw engine in the
        dynamic batcher so concurrent /chat requests share decode batches
        (weight-bandwidth amortization; no reference counterpart — the
        reference's provider batched server-side)."""
        def make():
            raw = self.generator()
            if (self.settings.continuous_batching
                    and self.settings.dynamic_batching
                    and hasattr(raw, "make_slot_session")):
                # real engine only: mock engines take the wave batcher
                from sentio_amd.serving.batcher import ContinuousGenerator

                return ContinuousGenerator(
                    raw, slots=self.settings.max_batch_size)
            if not self.settings.dynamic_batching:
                return raw
            from sentio_amd.serving.batcher import BatchedGenerator

            return BatchedGenerator(raw,
                                    max_batch=self.settings.max_batch_size,
                                    max_wait_ms=self.settings.batch_wait_ms)

        return self._get("generator_frontend", make)

    # --- indexes ---
    def dense_index(self) -> DenseIndex:
        return self._get("dense_index", lambda: DenseIndex(
            dim=self.settings.embedding_dim, device=self.device,
            dtype=torch.float16))

    def bm25_index(self) -> BM25Index:
        return self._get("bm25_index", lambda: BM25Index(
            k1=self.settings.bm25_k1, b=self.settings.bm25_b,
            variant=self.settings.bm25_variant))

    # --- composition ---
    def retriever(self):
        return self._get("retriever", lambda: create_retriever(
            self.settings, self.encoder(), self.dense_index(),
            self.bm25_index(), device=self.device))

    def verifier(self):
        return self._get("verifier", lambda: AnswerVerifier(
            self.generator(), max_tokens=self.settings.verifier_max_tokens))

    def pipeline(self) -> RagPipeline:
        def make():
            cfg = GraphConfig.from_settings(
                self.settings,
                retriever=self.retriever(),
                reranker=self.reranker() if self.settings.use_reranker else None,
                generator=self.generator_frontend(),
                verifier=self.verifier() if self.settings.use_verifier else None,
            )
            return build_basic_graph(cfg)

        return self._get("pipeline", make)

    def ingestor(self) -> DocumentIngestor:
        return self._get("ingestor", lambda: DocumentIngestor(
            self.encoder(), self.dense_index(), self.bm25_index(),
            TextChunker(self.settings.chunk_size, self.settings.chunk_overlap)))

    def cache_manager(self) -> CacheManager:
        return self._get("cache_manager",
                         lambda: CacheManager(self.settings.cache_backend))

    def auth_manager(self) -> AuthManager:
        return self._get("auth", lambda: AuthManager(
            secret=self.settings.auth_secret,
            token_ttl_s=self.settings.auth_token_ttl_s))

    def health_checker(self):
        """Periodic probes over the GPU engines and indexes
        (reference patterns.py:252-306 HealthChecker role)."""
        def make():
            from sentio_amd.resilience.health import HealthChecker

            hc = HealthChecker(interval_s=self.settings.health_interval_s)

            def check_encoder() -> bool:
                v = self.encoder().embed(["health probe"])
                return v.shape[-1] == self.settings.embedding_dim

            def check_device() -> bool:
                if self.device == "cpu":
                    return True
                import torch

                free, total = torch.cuda.mem_get_info()
                return free > 0

            hc.register("encoder", check_encoder)
            hc.register("device", check_device)
            from sentio_amd.resilience.gpu_health import gpu_health_check
            hc.register("gpu_roundtrip",
                        lambda: gpu_health_check(self.device))
            hc.register("dense_index", lambda: len(self.dense_index()) >= 0)
            for name, br in self.breakers.items():
                hc.register(f"breaker:{name}",
                            lambda b=br: b.state.value != "open")
            return hc

        return self._get("health_checker", make)

    # --- index snapshots (HBM→disk; reference persisted only a BM25 pickle,
    # sparse.py:102-157 — here the whole index state checkpoints) ---
    def save_indexes(self, directory: str) -> dict[str, Any]:
        import os

        os.makedirs(directory, exist_ok=True)
        dense_path = os.path.join(directory, "dense.pt")
        bm25_path = os.path.join(directory, "bm25.npz")
        self.dense_index().save(dense_path)
        self.bm25_index().save(bm25_path)
        return {"dense": dense_path, "bm25": bm25_path,
                "docs": len(self.dense_index())}

    def load_indexes(self, directory: str) -> dict[str, Any]:
        import os

        with self._lock:
            self._cache["dense_index"] = DenseIndex.load(
                os.path.join(directory, "dense.pt"), device=self.device)
            self._cache["bm25_index"] = BM25Index.load(
                os.path.join(directory, "bm25.npz"))
            for k in ("retriever", "pipeline", "ingestor"):
                self._cache.pop(k, None)
        return {"docs": len(self.dense_index())}

    _WARM_BUCKETS = (1, 2, 4, 8, 16, 32, 64)

    def initialize_all(self) -> None:
        """Eager startup init in dependency order."""
        self.encoder()
        on_gpu = self.device != "cpu" and not self.settings.mock_compute
        try:
            # warm-up embeds (reference warm_up_embeddings startup hook,
            # embeddings/factory.py:122-137).  On GPU: capture the hipGraph
            # for EVERY batch bucket now, serially — a capture taken
            # mid-traffic runs concurrently with decode-graph replays on
            # other threads, which can wedge the HIP context (observed as a
            # hung /chat/stream + every later CUDA call blocking)
            enc = self.encoder()
            raw = getattr(enc, "raw", enc)
            for b in (self._WARM_BUCKETS if on_gpu else (1,)):
                raw.embed([f"warm up {b}.{i}" for i in range(b)])
        except Exception as exc:
            logger.warning("encoder warm-up failed: %s", exc)
        self.dense_index()
        self.bm25_index()
        self.retriever()
        if self.settings.use_reranker:
            rr = self.reranker()
            if on_gpu:
                try:
                    raw = getattr(rr, "raw", rr)
                    for b in self._WARM_BUCKETS:
                        raw.score_packed([f"warm\ndoc {b}.{i}"
                                          for i in range(b)])
                except Exception as exc:
                    logger.warning("reranker warm-up failed: %s", exc)
        self.generator()
        if self.device != "cpu" and not self.settings.mock_compute:
            # pre-capture the decode hipGraphs for the serving batch
            # buckets: a first-occurrence capture costs ~1 s, which the GPU
            # load test otherwise pays mid-traffic (p95 blowout)
            try:
                gen = self.generator()
                # continuous batching serves through ONE slot session;
                # the wave-batcher's bucket pool would cost ~1 GB of KV
                # per row (kv_cache_max_tokens) for sessions that mode
                # never uses (64-slot config OOM'd a 288 GB GPU on the
                # full pool) — pre-capture only what the active mode runs
                buckets = ((1,) if self.settings.continuous_batching
                           else (1, 2, 4, 8, 16, 24, 32, 48, 64))
                for b in buckets:
                    if b <= self.settings.max_batch_size:
                        gen.generate([f"warm {i}" for i in range(b)],
                                     max_new_tokens=2, temperature=0.0)
            except Exception as exc:
                logger.warning("generator warm-up failed: %s", exc)
            try:
                # capture the continuous batcher's slot-session decode
                # graph before traffic (same mid-traffic-capture hazard)
                self.generator_frontend().generate(["warm frontend"],
                                                   max_new_tokens=2,
                                                   temperature=0.0)
            except Exception as exc:
                logger.warning("frontend warm-up failed: %s", exc)
        self.pipeline()
        self.ingestor()
        self.auth_manager()
        logger.info("container initialized on device=%s", self.device)

    def clear_indexes(self) -> None:
        self.dense_index().clear()
        with self._lock:
            self._cache["bm25_index"] = BM25Index(
                k1=self.settings.bm25_k1, b=self.settings.bm25_b,
                variant=self.settings.bm25_variant)
            # retriever/pipeline/ingestor hold the old bm25; rebuild them
            for k in ("retriever", "pipeline", "ingestor"):
                self._cache.pop(k, None)

    def health(self) -> dict[str, Any]:
        gen = self._cache.get("generator")
        batcher = getattr(gen, "batcher", None)
        return {
            "device": self.device,
            "index_size": len(self.dense_index()),
            "bm25_docs": self.bm25_index().n_docs,
            "breakers": {k: b.health() for k, b in self.breakers.items()},
            # dynamic-batcher coalescing stats (requests/batches/max seen)
            "batcher": batcher.health() if batcher is not None else None,
            "micro_batchers": {
                k: getattr(self._cache.get(k), "micro", None).health()
                for k in ("encoder", "reranker")
                if getattr(self._cache.get(k), "micro", None) is not None
            },
            "initialized": sorted(self._cache.keys()),
        }


_container: ServiceContainer | None = None
_container_lock = threading.Lock()


def get_container(settings: Settings | None = None) -> ServiceContainer:
    global _container
    with _container_lock:
        if _container is None:
            _container = ServiceContainer(settings)
        return _container


def reset_container() -> None:
    global _container
    with _container_lock:
        _container = None
