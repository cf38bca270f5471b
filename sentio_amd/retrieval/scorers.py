"""Scorer plugins (reference src/core/retrievers/scorers.py:25-273 semantics):

* KeywordMatchScorer: |keywords ∩ doc words| / |keywords| · weight
* RecencyScorer: age-normalized score · weight
* SemanticSimilarityScorer: cosine(query_emb, doc_embs) · weight
* MMRScorer: greedy λ·relevance − (1−λ)·max-redundancy selection bonus
"""

from __future__ import annotations

import time


from sentio_amd.models.document import Document


class KeywordMatchScorer:
    def __init__(self, weight: float = 0.2):
        self.weight = weight

    def score(self, query: str, documents: list[Document]) -> list[float]:
        keywords = set(query.lower().split())
        if not keywords:
            return [0.0] * len(documents)
        out = []
        for doc in documents:
            words = set((doc.text or "").lower().split())
            out.append(self.weight * len(keywords & words) / len(keywords))
        return out


class RecencyScorer:
    def __init__(self, weight: float = 0.1, half_life_days: float = 30.0,
                 timestamp_key: str = "timestamp"):
        self.weight = weight
        self.half_life_s = half_life_days * 86400.0
        self.key = timestamp_key

    def score(self, query: str, documents: list[Document]) -> list[float]:
        now = time.time()
        out = []
        for doc in documents:
            ts = doc.metadata.get(self.key)
            if ts is None:
                out.append(0.0)
                continue
            age = max(0.0, now - float(ts))
            out.append(self.weight * (0.5 ** (age / self.half_life_s)))
        return out


class SemanticSimilarityScorer:
    def __init__(self, embedder, weight: float = 0.8):
        self.embedder = embedder
        self.weight = weight

    def score(self, query: str, documents: list[Document]) -> list[float]:
        if not documents:
            return []
        qv = self.embedder.embed([query])  # [1,D], normalized
        dv = self.embedder.embed([d.text or "" for d in documents])  # [N,D]
        sims = (qv.float() @ dv.float().T).squeeze(0)
        return (self.weight * sims).cpu().tolist()


class MMRScorer:
    """Maximal-marginal-relevance selection bonus: documents picked early by
    the greedy λ·rel − (1−λ)·max-redundancy loop get a decaying bonus."""

    def __init__(self, embedder, lambda_param: float = 0.5, weight: float = 0.5):
        self.embedder = embedder
        self.lam = lambda_param
        self.weight = weight

    def score(self, query: str, documents: list[Document]) -> list[float]:
        n = len(documents)
        if n == 0:
            return []
        qv = self.embedder.embed([query]).float()
        dv = self.embedder.embed([d.text or "" for d in documents]).float()
        rel = (qv @ dv.T).squeeze(0)  # [N]
        sim = dv @ dv.T               # [N,N]
        selected: list[int] = []
        remaining = set(range(n))
        scores = [0.0] * n
        rank = 0
        while remaining:
            best, best_val = None, None
            for i in remaining:
                red = max((float(sim[i, j]) for j in selected), default=0.0)
                val = self.lam * float(rel[i]) - (1.0 - self.lam) * red
                if best_val is None or val > best_val:
                    best, best_val = i, val
            selected.append(best)
            remaining.discard(best)
            # earlier MMR picks get a larger bonus
            scores[best] = self.weight * (1.0 - rank / max(n, 1))
            rank += 1
        return scores


def default_scorer_plugins(embedder) -> list:
    """Default plugin set (reference retrievers/factory.py:64-80):
    semantic w=0.8, keyword w=0.2, MMR λ=0.5 w=0.5."""
    return [
        SemanticSimilarityScorer(embedder, weight=0.8),
        KeywordMatchScorer(weight=0.2),
        MMRScorer(embedder, lambda_param=0.5, weight=0.5),
    ]
