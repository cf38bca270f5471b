"""On-device cross-encoder reranker (replaces the reference's Jina rerank API,
reference src/core/rerankers/jina_reranker.py:172-185; K5 in SURVEY §2.3).

Packs (query, doc) pairs into one batch, runs the encoder transformer, takes
the first-position hidden state through a scalar head → relevance score.
Mirrors the reference's call shape: top_n = min(len(docs), 2·top_k)
candidates scored, results mapped back by index, sorted, truncated.
"""

from __future__ import annotations

import torch

from sentio_amd.engines.configs import get_model_config
from sentio_amd.engines.bpe import get_tokenizer
from sentio_amd.engines.transformer import Transformer
from sentio_amd.models.document import Document


class RerankerEngine:
    def __init__(self, model: str = "sentio-reranker-base", device: str = "cpu",
                 dtype: str = "bf16", max_seq: int = 512, seed: int = 202):
        self.cfg = get_model_config(model)
        self.device = device
        self.max_seq = min(max_seq, self.cfg.max_seq)
        self.tokenizer = get_tokenizer()
        self.model = Transformer(self.cfg, device=device, dtype=dtype, seed=seed)

    def _graph_pool(self):
        pool = getattr(self, "_graphs", None)
        if pool is None:
            from sentio_amd.engines.graphed import GraphedEnginePool

            def fwd(tokens, kv_lens):
                hidden = self.model.forward_hidden(tokens, kv_lens=kv_lens)
                pooled = hidden[:, 0, :]   # first-token pooled representation
                s = torch.nn.functional.linear(pooled, self.model.w.head)
                return torch.sigmoid(s.float()).squeeze(-1)

            pool = self._graphs = GraphedEnginePool(fwd, self.max_seq,
                                                    self.device)
        return pool

    @torch.inference_mode()
    def score_pairs(self, query: str, texts: list[str], batch_size: int = 32) -> list[float]:
        return self.score_packed([f"{query}\n{t}" for t in texts], batch_size)

    @torch.inference_mode()
    def score_packed(self, pair_texts_all: list[str],
                     batch_size: int = 32) -> list[float]:
        """Score pre-packed "query\\ntext" pairs — the batched entry point
        (serving micro-batcher, bench) where each pair carries its own
        query; identical packing to score_pairs."""
        if not pair_texts_all:
            return []
        pool = self._graph_pool()
        scores: list[float] = []
        for i in range(0, len(pair_texts_all), batch_size):
            pair_texts = pair_texts_all[i : i + batch_size]
            padded, lens = self.tokenizer.encode_batch(pair_texts, self.max_seq)
            tokens = torch.tensor(padded, dtype=torch.int64, device=self.device)
            kv_lens = torch.tensor(lens, dtype=torch.int32, device=self.device)
            if pool.active:
                s = pool.run(tokens, kv_lens)
            else:
                hidden = self.model.forward_hidden(tokens, kv_lens=kv_lens)
                pooled = hidden[:, 0, :]  # first-token pooled representation
                s = torch.sigmoid(torch.nn.functional.linear(
                    pooled, self.model.w.head).float()).squeeze(-1)
            scores.extend(s.cpu().tolist())
        return scores

    @torch.inference_mode()
    def rerank(self, query: str, docs: list[Document], top_k: int) -> list[Document]:
        """Score min(len, 2·top_k) candidates (reference jina_reranker.py:172),
        sort desc, truncate to top_k, tag metadata['rerank_score']."""
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
