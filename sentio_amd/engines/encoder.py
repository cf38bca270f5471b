"""On-device text-embedding engine (replaces the reference's Jina API calls,
reference src/core/embeddings/providers/jina.py:165-172; K1/K8 in SURVEY §2.3).

Encoder transformer forward → masked mean-pool + L2-normalize → [B, 1024]
fp32 vectors.  Batched (reference batched 100 texts per request; here the
batch rides one kernel launch sequence).  Includes the reference's
embedding-cache semantics (LFU+TTL, reference embeddings/base.py:23-106) in
simplified LRU+TTL form via caching.MemoryCache at the call site.
"""

from __future__ import annotations

import torch

from sentio_amd import ops
from sentio_amd.engines.configs import get_model_config
from sentio_amd.engines.tokenizer import ByteTokenizer
from sentio_amd.engines.transformer import Transformer


class EncoderEngine:
    def __init__(self, model: str = "sentio-encoder-base", device: str = "cpu",
                 dtype: str = "bf16", max_seq: int = 512, seed: int = 101):
        self.cfg = get_model_config(model)
        self.device = device
        self.max_seq = min(max_seq, self.cfg.max_seq)
        self.tokenizer = ByteTokenizer()
        self.model = Transformer(self.cfg, device=device, dtype=dtype, seed=seed)
        self.dim = self.cfg.dim
        self.calls = 0

    @torch.inference_mode()
    def embed(self, texts: list[str], batch_size: int = 64) -> torch.Tensor:
        """texts → [N, dim] fp32 L2-normalized embeddings (on self.device)."""
        if not texts:
            return torch.empty(0, self.dim, device=self.device)
        outs = []
        for i in range(0, len(texts), batch_size):
            outs.append(self._embed_batch(texts[i : i + batch_size]))
        self.calls += 1
        return torch.cat(outs, dim=0)

    def _embed_batch(self, texts: list[str]) -> torch.Tensor:
        padded, lens = self.tokenizer.encode_batch(texts, self.max_seq)
        tokens = torch.tensor(padded, dtype=torch.int64, device=self.device)
        B, S = tokens.shape
        kv_lens = torch.tensor(lens, dtype=torch.int32, device=self.device)
        mask = torch.arange(S, device=self.device).unsqueeze(0) < kv_lens.unsqueeze(1)
        hidden = self.model.forward_hidden(tokens, kv_lens=kv_lens)
        return ops.mean_pool_l2norm(hidden, mask)

    def embed_one(self, text: str) -> torch.Tensor:
        return self.embed([text])[0]
