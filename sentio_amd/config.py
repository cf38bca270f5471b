"""Typed configuration for the whole engine.

Replaces the reference's pydantic-settings singleton plus its scattered
``os.getenv`` escapes (reference src/utils/settings.py:27-191,
src/core/graph/factory.py:76-91, src/core/retrievers/factory.py:35-50) with a
single dataclass resolved once from the environment.  The flag *names* keep
the reference's spelling so a user of the reference can switch without
relearning the env surface; GPU-topology flags (shards, TP degree, dtype) are
new and MI355X-specific.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field, fields


def _env_bool(name: str, default: bool) -> bool:
    v = os.getenv(name)
    if v is None:
        return default
    return v.strip().lower() in ("1", "true", "yes", "on")


def _env_int(name: str, default: int) -> int:
    v = os.getenv(name)
    try:
        return int(v) if v is not None else default
    except ValueError:
        return default


def _env_float(name: str, default: float) -> float:
    v = os.getenv(name)
    try:
        return float(v) if v is not None else default
    except ValueError:
        return default


def _env_str(name: str, default: str) -> str:
    v = os.getenv(name)
    return v if v is not None else default


@dataclass
class Settings:
    """All engine flags.  Defaults mirror the reference's defaults where a
    counterpart exists (cited per field group)."""

    # --- retrieval (reference settings.py + retrievers/factory.py) ---
    retrieval_strategy: str = "hybrid"      # dense | bm25 | hybrid
    fusion_method: str = "rrf"              # rrf | weighted_rrf | comb_sum
    dense_weight: float = 0.7
    sparse_weight: float = 0.3
    rrf_k: int = 60                         # reference hybrid.py RRF constant
    retrieval_top_k: int = 10
    reranking_top_k: int = 5
    selection_top_k: int = 3
    use_reranker: bool = True
    use_verifier: bool = False
    selector_max_tokens: int = 2000         # reference graph/factory.py:90
    bm25_variant: str = "okapi"             # okapi | plus
    bm25_k1: float = 1.5
    bm25_b: float = 0.75

    # --- generation (reference llm/factory.py, generator.py) ---
    generation_mode: str = "balanced"       # fast | balanced | quality | creative
    llm_max_tokens: int = 1024
    verifier_max_tokens: int = 512

    # --- chunking / ingest (reference settings.py:89-90) ---
    chunk_size: int = 512
    chunk_overlap: int = 64

    # --- serving (reference app.py) ---
    api_host: str = "0.0.0.0"
    api_port: int = 8000
    disable_auth: bool = True
    rate_limit_chat_per_min: int = 100      # reference app.py:259-271
    rate_limit_embed_per_min: int = 10
    max_query_len: int = 2000               # reference security.py InputValidator
    max_document_len: int = 50000
    cache_backend: str = "memory"           # memory | multi_tier | disk
    cache_ttl_s: float = 300.0
    embedding_cache_size: int = 2048        # reference base.py:23-106 LFU+TTL
    embedding_cache_ttl_s: float = 3600.0
    health_interval_s: float = 30.0         # reference patterns.py:252-306 loop

    # --- models / engines (MI355X-native; sizes are the bench ladder's) ---
    embedding_dim: int = 1024               # jina-v3 class (reference jina.py:23-27)
    encoder_model: str = "sentio-encoder-base"
    reranker_model: str = "sentio-reranker-base"
    generator_model: str = "llama3-8b"
    compute_dtype: str = "bf16"             # bf16 | fp16
    mock_compute: bool = False              # deterministic hash engines (tests/CPU)

    # --- GPU topology (new; no reference counterpart) ---
    tp_degree: int = 1
    index_shards: int = 1                   # = world size when distributed
    kv_cache_max_tokens: int = 8192
    max_batch_size: int = 64          # continuous-batching slots (64 rows of KV ~ 68 GB at 8k ctx)
    dynamic_batching: bool = True           # batch concurrent /chat generations
    continuous_batching: bool = True        # requests join decode mid-flight
    batch_wait_ms: float = 8.0
    device: str = "auto"                    # auto | cuda | cpu

    # --- observability ---
    enable_metrics: bool = True
    enable_tracing: bool = False

    # --- auth (stdlib HMAC tokens; reference used JWT via python-jose) ---
    auth_secret: str = "sentio-dev-secret"
    auth_token_ttl_s: int = 3600

    extra: dict = field(default_factory=dict)

    _ENV_MAP = {
        "retrieval_strategy": "RETRIEVAL_STRATEGY",
        "fusion_method": "FUSION_METHOD",
        "dense_weight": "DENSE_WEIGHT",
        "sparse_weight": "SPARSE_WEIGHT",
        "rrf_k": "RRF_K",
        "retrieval_top_k": "RETRIEVAL_TOP_K",
        "reranking_top_k": "RERANKING_TOP_K",
        "selection_top_k": "SELECTION_TOP_K",
        "use_reranker": "USE_RERANKER",
        "use_verifier": "USE_VERIFIER",
        "selector_max_tokens": "SELECTOR_MAX_TOKENS",
        "bm25_variant": "BM25_VARIANT",
        "bm25_k1": "BM25_K1",
        "bm25_b": "BM25_B",
        "generation_mode": "GENERATION_MODE",
        "llm_max_tokens": "LLM_MAX_TOKENS",
        "chunk_size": "CHUNK_SIZE",
        "chunk_overlap": "CHUNK_OVERLAP",
        "api_host": "API_HOST",
        "api_port": "API_PORT",
        "disable_auth": "DISABLE_AUTH",
        "cache_backend": "CACHE_BACKEND",
        "embedding_dim": "EMBEDDING_DIM",
        "encoder_model": "ENCODER_MODEL",
        "reranker_model": "RERANKER_MODEL",
        "generator_model": "GENERATOR_MODEL",
        "compute_dtype": "COMPUTE_DTYPE",
        "mock_compute": "MOCK_COMPUTE",
        "tp_degree": "TP_DEGREE",
        "index_shards": "INDEX_SHARDS",
        "kv_cache_max_tokens": "KV_CACHE_MAX_TOKENS",
        "max_batch_size": "MAX_BATCH_SIZE",
        "dynamic_batching": "DYNAMIC_BATCHING",
        "continuous_batching": "CONTINUOUS_BATCHING",
        "batch_wait_ms": "BATCH_WAIT_MS",
        "device": "SENTIO_DEVICE",
        "enable_metrics": "ENABLE_METRICS",
        "enable_tracing": "ENABLE_TRACING",
        "auth_secret": "AUTH_SECRET",
        "rate_limit_chat_per_min": "RATE_LIMIT_CHAT_PER_MIN",
        "rate_limit_embed_per_min": "RATE_LIMIT_EMBED_PER_MIN",
    }

    @classmethod
    def from_env(cls) -> "Settings":
        s = cls()
        for f in fields(cls):
            env = cls._ENV_MAP.get(f.name)
            if env is None or os.getenv(env) is None:
                continue
            cur = getattr(s, f.name)
            if isinstance(cur, bool):
                setattr(s, f.name, _env_bool(env, cur))
            elif isinstance(cur, int):
                setattr(s, f.name, _env_int(env, cur))
            elif isinstance(cur, float):
                setattr(s, f.name, _env_float(env, cur))
            else:
                setattr(s, f.name, _env_str(env, cur))
        return s

    @property
    def auth_enabled(self) -> bool:
        return not self.disable_auth

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        try:
            import torch

            return "cuda" if torch.cuda.is_available() else "cpu"
        except Exception:
            return "cpu"


settings = Settings.from_env()


def reload_settings() -> Settings:
    """Re-read the environment (used by tests)."""
    global settings
    settings = Settings.from_env()
    return settings
