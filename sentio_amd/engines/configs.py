"""Model family configurations.

Sizes follow the BASELINE.json config ladder: jina-v3-class encoder,
bge-reranker-base-class cross-encoder, Llama-3-8B / 70B-class generators.
Weights are always random-init (no network for checkpoints — BASELINE.md).
Tiny variants exist for CPU tests.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class ModelConfig:
    name: str
    dim: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    ffn_dim: int
    vocab_size: int
    max_seq: int = 8192
    rope_base: float = 500000.0
    norm_eps: float = 1e-5
    causal: bool = True          # decoder vs encoder
    pooled_head: int = 0         # >0: classification head width (reranker)

    @property
    def head_dim(self) -> int:
        return self.dim // self.n_heads

    @property
    def n_params(self) -> int:
        d, L, f, v = self.dim, self.n_layers, self.ffn_dim, self.vocab_size
        hd = self.head_dim
        attn = d * (self.n_heads * hd) + 2 * d * (self.n_kv_heads * hd) + (self.n_heads * hd) * d
        ffn = 3 * d * f
        per_layer = attn + ffn + 2 * d
        emb = v * d
        head = 0 if self.causal is False else v * d
        return L * per_layer + emb + head + d


MODEL_CONFIGS: dict[str, ModelConfig] = {
    # --- generators (Llama-3 family shapes) ---
    "llama3-8b": ModelConfig(
        name="llama3-8b", dim=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        ffn_dim=14336, vocab_size=128256, max_seq=8192,
    ),
    "llama3-70b": ModelConfig(
        name="llama3-70b", dim=8192, n_layers=80, n_heads=64, n_kv_heads=8,
        ffn_dim=28672, vocab_size=128256, max_seq=8192,
    ),
    "qwen2-7b": ModelConfig(   # Qwen-2 family shape (GQA 28/4, 3584 dim)
        name="qwen2-7b", dim=3584, n_layers=28, n_heads=28, n_kv_heads=4,
        ffn_dim=18944, vocab_size=152064, max_seq=8192, rope_base=1000000.0,
    ),
    "mistral-7b": ModelConfig(  # Mistral-7B family shape (GQA 32/8)
        name="mistral-7b", dim=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        ffn_dim=14336, vocab_size=32768, max_seq=8192, rope_base=1000000.0,
    ),
    "llama3-1b": ModelConfig(  # small real decoder for quick GPU checks
        name="llama3-1b", dim=2048, n_layers=16, n_heads=32, n_kv_heads=8,
        ffn_dim=8192, vocab_size=128256, max_seq=8192,
    ),
    "tiny-decoder": ModelConfig(
        name="tiny-decoder", dim=64, n_layers=2, n_heads=4, n_kv_heads=2,
        ffn_dim=128, vocab_size=4608, max_seq=1024, rope_base=10000.0,
    ),
    "tiny-decoder64": ModelConfig(  # head_dim 64 — GPU-kernel-compatible tiny
        name="tiny-decoder64", dim=256, n_layers=2, n_heads=4, n_kv_heads=2,
        ffn_dim=512, vocab_size=4608, max_seq=1024, rope_base=10000.0,
    ),
    # --- encoders (jina-v3 class: 1024-dim output, reference jina.py:23-27) ---
    "sentio-encoder-base": ModelConfig(
        name="sentio-encoder-base", dim=1024, n_layers=24, n_heads=16,
        n_kv_heads=16, ffn_dim=4096, vocab_size=4608, max_seq=2048,
        rope_base=10000.0, causal=False,
    ),
    "sentio-encoder-small": ModelConfig(
        name="sentio-encoder-small", dim=1024, n_layers=6, n_heads=16,
        n_kv_heads=16, ffn_dim=2048, vocab_size=4608, max_seq=2048,
        rope_base=10000.0, causal=False,
    ),
    "tiny-encoder": ModelConfig(
        name="tiny-encoder", dim=64, n_layers=2, n_heads=4, n_kv_heads=4,
        ffn_dim=128, vocab_size=4608, max_seq=512, rope_base=10000.0, causal=False,
    ),
    # --- rerankers (bge-reranker-base class) ---
    "sentio-reranker-base": ModelConfig(
        name="sentio-reranker-base", dim=768, n_layers=12, n_heads=12,
        n_kv_heads=12, ffn_dim=3072, vocab_size=4608, max_seq=1024,
        rope_base=10000.0, causal=False, pooled_head=1,
    ),
    "tiny-reranker": ModelConfig(
        name="tiny-reranker", dim=64, n_layers=2, n_heads=4, n_kv_heads=4,
        ffn_dim=128, vocab_size=4608, max_seq=512, rope_base=10000.0,
        causal=False, pooled_head=1,
    ),
}


def get_model_config(name: str) -> ModelConfig:
    if name not in MODEL_CONFIGS:
        raise KeyError(f"unknown model config '{name}' (have: {sorted(MODEL_CONFIGS)})")
    return MODEL_CONFIGS[name]
