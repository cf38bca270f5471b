"""Deterministic offline mock engines.

Models the reference's built-in fake backends: hash-seeded deterministic
embedding vectors (reference jina.py:141-159), order-preserving fallback
rerank scores 1.0 − 0.1·idx (reference jina_reranker.py:297-322), and canned
template generation (reference resilience/fallbacks.py:205-259).  Used by
CPU plumbing tests and as degraded-mode fallbacks; they share the engine
interfaces exactly.
"""

from __future__ import annotations

import hashlib

import numpy as np
import torch

from sentio_amd.models.document import Document


class MockEncoderEngine:
    def __init__(self, dim: int = 1024, device: str = "cpu", **_kw):
        self.dim = dim
        self.device = device
        self.calls = 0

    def _vec(self, text: str) -> np.ndarray:
        seed = int.from_bytes(hashlib.md5(text.encode()).digest()[:4], "little")
        rng = np.random.RandomState(seed)
        v = rng.standard_normal(self.dim).astype(np.float32)
        return v / max(np.linalg.norm(v), 1e-12)

    def embed(self, texts: list[str], batch_size: int = 64) -> torch.Tensor:
        self.calls += 1
        if not texts:
            return torch.empty(0, self.dim, device=self.device)
        arr = np.stack([self._vec(t) for t in texts])
        return torch.from_numpy(arr).to(self.device)

    def embed_one(self, text: str) -> torch.Tensor:
        return self.embed([text])[0]


class MockRerankerEngine:
    def __init__(self, **_kw):
        pass

    def score_pairs(self, query: str, texts: list[str], batch_size: int = 32) -> list[float]:
        # deterministic overlap-based score so tests can assert ordering
        q = set(query.lower().split())
        out = []
        for t in texts:
            toks = set(t.lower().split())
            inter = len(q & toks) / max(len(q), 1)
            out.append(min(1.0, 0.1 + inter))
        return out

    def rerank(self, query: str, docs: list[Document], top_k: int) -> list[Document]:
        if not docs:
            return []
        top_n = min(len(docs), 2 * top_k)
        cand = docs[:top_n]
        scores = self.score_pairs(query, [d.text for d in cand])
        order = sorted(range(len(cand)), key=lambda i: scores[i], reverse=True)
        out = []
        for i in order[:top_k]:
            d = cand[i]
            d.metadata["rerank_score"] = float(scores[i])
            d.metadata["score"] = float(scores[i])
            out.append(d)
        return out


class MockGeneratorEngine:
    """Template answers citing the numbered context — shaped like the
    reference's fallback responses so verifier/selector paths exercise."""

    def __init__(self, **_kw):
        self.tokenizer = None

    def generate(self, prompts: list[str], max_new_tokens: int = 128,
                 temperature: float = 0.3, stop_on_eos: bool = True,
                 on_token=None) -> list[str]:
        outs = []
        for p in prompts:
            h = hashlib.sha256(p.encode()).hexdigest()[:8]
            outs.append(
                f"Based on the provided context [1], the answer addresses the "
                f"query (trace {h}). Key supporting details appear in [1] and [2]."
            )
        return outs

    def stream(self, prompt: str, max_new_tokens: int = 128, temperature: float = 0.3):
        text = self.generate([prompt])[0]
        for i in range(0, len(text), 24):
            yield text[i : i + 24]
