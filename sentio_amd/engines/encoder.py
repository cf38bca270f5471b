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
from sentio_amd.engines.bpe import get_tokenizer
from sentio_amd.engines.transformer import Transformer


class EncoderEngine:
    def __init__(self, model: str = "sentio-encoder-base", device: str = "cpu",
                 dtype: str = "bf16", max_seq: int = 512, seed: int = 101,
                 cache_size: int = 0, cache_ttl: float = 3600.0):
        self.cfg = get_model_config(model)
        self.device = device
        self.max_seq = min(max_seq, self.cfg.max_seq)
        self.tokenizer = get_tokenizer()
        self.model = Transformer(self.cfg, device=device, dtype=dtype, seed=seed)
        self.dim = self.cfg.dim
        self.calls = 0
        self.cache_hits = 0
        self.texts_embedded = 0
        self.errors = 0
        self.total_time_s = 0.0
        self.cache = None
        if cache_size > 0:
            from sentio_amd.caching.memory import MemoryCache

            self.cache = MemoryCache(max_size=cache_size, default_ttl=cache_ttl)

    @torch.inference_mode()
    def embed(self, texts: list[str], batch_size: int = 64) -> torch.Tensor:
        """texts → [N, dim] fp32 L2-normalized embeddings (on self.device).

        With a cache configured, per-text lookup + merge semantics mirror the
        reference's embedding cache (reference embeddings/base.py:23-106,
        jina.py:217-248: hit texts skip the forward, misses are embedded in
        one batch and written back)."""
        if not texts:
            return torch.empty(0, self.dim, device=self.device)
        self.calls += 1
        if self.cache is None:
            outs = []
            for i in range(0, len(texts), batch_size):
                outs.append(self._embed_batch(texts[i : i + batch_size]))
            return torch.cat(outs, dim=0)

        out = torch.empty(len(texts), self.dim, device=self.device)
        miss_idx: list[int] = []
        for i, t in enumerate(texts):
            hit = self.cache.get_embedding(t)
            if hit is not None:
                out[i] = hit.to(self.device)
                self.cache_hits += 1
            else:
                miss_idx.append(i)
        for i0 in range(0, len(miss_idx), batch_size):
            idxs = miss_idx[i0 : i0 + batch_size]
            vecs = self._embed_batch([texts[i] for i in idxs])
            for j, i in enumerate(idxs):
                out[i] = vecs[j]
                self.cache.set_embedding(texts[i], vecs[j])
        return out

    def _graph_pool(self):
        pool = getattr(self, "_graphs", None)
        if pool is None:
            from sentio_amd.engines.graphed import GraphedEnginePool

            def fwd(tokens, kv_lens):
                S = tokens.shape[1]
                mask = (torch.arange(S, device=self.device).unsqueeze(0)
                        < kv_lens.unsqueeze(1))
                hidden = self.model.forward_hidden(tokens, kv_lens=kv_lens)
                return ops.mean_pool_l2norm(hidden, mask)

            pool = self._graphs = GraphedEnginePool(fwd, self.max_seq,
                                                    self.device)
        return pool

    def stats(self) -> dict:
        """Usage counters (reference embeddings/base.py:245-284 stats role:
        calls, cache hits, errors, average embed time)."""
        n = max(self.texts_embedded, 1)
        return {
            "calls": self.calls,
            "texts_embedded": self.texts_embedded,
            "cache_hits": self.cache_hits,
            "errors": self.errors,
            "avg_time_ms_per_text": round(1e3 * self.total_time_s / n, 3),
        }

    def _embed_batch(self, texts: list[str]) -> torch.Tensor:
        import time as _time

        t0 = _time.perf_counter()
        try:
            v = self._embed_batch_inner(texts)
        except Exception:
            self.errors += 1
            raise
        self.texts_embedded += len(texts)
        self.total_time_s += _time.perf_counter() - t0
        return v

    def _embed_batch_inner(self, texts: list[str]) -> torch.Tensor:
        padded, lens = self.tokenizer.encode_batch(texts, self.max_seq)
        tokens = torch.tensor(padded, dtype=torch.int64, device=self.device)
        B, S = tokens.shape
        kv_lens = torch.tensor(lens, dtype=torch.int32, device=self.device)
        pool = self._graph_pool()
        if pool.active:
            # fixed-shape hipGraph replay (launch-bound at these sizes)
            return pool.run(tokens, kv_lens)
        mask = torch.arange(S, device=self.device).unsqueeze(0) < kv_lens.unsqueeze(1)
        hidden = self.model.forward_hidden(tokens, kv_lens=kv_lens)
        return ops.mean_pool_l2norm(hidden, mask)

    def embed_one(self, text: str) -> torch.Tensor:
        return self.embed([text])[0]
