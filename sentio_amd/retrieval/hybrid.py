"""Hybrid dense+sparse retriever with rrf / weighted_rrf / comb_sum fusion
(reference src/core/retrievers/hybrid.py:48-323 semantics; the fusion math
itself lives in index/fusion.py and, on device, in the K4 fusion epilogue)."""

from __future__ import annotations

from sentio_amd.index import fusion
from sentio_amd.models.document import Document
from sentio_amd.retrieval.base import BaseRetriever, ScorerPlugin


class HybridRetriever(BaseRetriever):
    def __init__(
        self,
        dense=None,
        sparse=None,
        fusion_method: str = "rrf",
        rrf_k: int = 60,
        dense_weight: float = 0.7,
        sparse_weight: float = 0.3,
        scorer_plugins: list[ScorerPlugin] | None = None,
        cache_retriever=None,
        cache_score_threshold: float = 0.9,
    ):
        self.dense = dense
        self.sparse = sparse
        self.fusion_method = fusion_method
        self.rrf_k = rrf_k
        self.dense_weight = dense_weight
        self.sparse_weight = sparse_weight
        self.scorer_plugins = scorer_plugins or []
        # optional hot second collection probed before the main corpus
        # (reference hybrid.py:96-107,146-182 "web_cache" collection):
        # strong cache hits short-circuit the full hybrid search; weaker
        # ones join the dense candidate pool for fusion.
        self.cache_retriever = cache_retriever
        self.cache_score_threshold = cache_score_threshold

    # "host" | "device" — which fusion path the last retrieve() took
    # (GPU tests assert the K4 kernel really runs on the /chat path)
    last_fusion_path: str = "none"

    def _index_device(self) -> str:
        idx = getattr(self.dense, "index", None)
        return str(getattr(idx, "device", "cpu"))

    def _fuse(self, dense_hits, sparse_hits, top_k: int):
        """Candidate fusion.  When the dense index is GPU-resident the K4
        fuse_topk kernel runs on device (one launch, one sync — the /chat
        hot path); the host fusion.py oracle is the CPU path and the
        semantics reference."""
        import torch

        dev = self._index_device()
        n_union = len(dense_hits) + len(sparse_hits)
        if (dev != "cpu" and torch.cuda.is_available()
                and 0 < n_union <= 128 and top_k <= 128):
            from sentio_amd import ops

            interned: dict[str, int] = {}

            def ref(doc_id: str) -> int:
                return interned.setdefault(doc_id, len(interned))

            Kd, Ks = max(len(dense_hits), 1), max(len(sparse_hits), 1)
            d_i = torch.full((1, Kd), -1, dtype=torch.int64)
            d_s = torch.zeros(1, Kd)
            for j, (i, s) in enumerate(dense_hits):
                d_i[0, j] = ref(i)
                d_s[0, j] = s
            s_i = torch.full((1, Ks), -1, dtype=torch.int64)
            s_s = torch.zeros(1, Ks)
            for j, (i, s) in enumerate(sparse_hits):
                s_i[0, j] = ref(i)
                s_s[0, j] = s
            out_i, out_s = ops.fuse_topk(
                d_i.to(dev), d_s.to(dev), s_i.to(dev), s_s.to(dev),
                method=self.fusion_method, top_k=min(top_k, 128),
                rrf_k=self.rrf_k, dense_weight=self.dense_weight,
                sparse_weight=self.sparse_weight)
            back = {v: k for k, v in interned.items()}
            self.last_fusion_path = "device"
            return [(back[int(i)], float(s))
                    for i, s in zip(out_i[0].cpu().tolist(),
                                    out_s[0].cpu().tolist()) if int(i) >= 0]
        self.last_fusion_path = "host"
        return fusion.fuse(
            dense_hits, sparse_hits, method=self.fusion_method, top_k=top_k,
            rrf_k=self.rrf_k, dense_weight=self.dense_weight,
            sparse_weight=self.sparse_weight)

    def retrieve(self, query: str, top_k: int = 10) -> list[Document]:
        cache_docs: list[Document] = []
        if self.cache_retriever is not None:
            try:
                cache_docs = self.cache_retriever.retrieve(query, top_k)
            except Exception:
                cache_docs = []
            for d in cache_docs:
                d.metadata["from_cache_collection"] = True
            strong = [d for d in cache_docs
                      if float(d.metadata.get("score", 0.0))
                      >= self.cache_score_threshold]
            if len(strong) >= top_k:
                return strong[:top_k]

        dense_docs = self.dense.retrieve(query, top_k) if self.dense else []
        if cache_docs:
            seen = {d.id for d in dense_docs}
            dense_docs = dense_docs + [d for d in cache_docs
                                       if d.id not in seen]
            dense_docs.sort(key=lambda d: float(d.metadata.get("score", 0.0)),
                            reverse=True)
        sparse_docs = self.sparse.retrieve(query, top_k) if self.sparse else []

        dense_hits = [(d.id, float(d.metadata.get("score", 0.0))) for d in dense_docs]
        sparse_hits = [(d.id, float(d.metadata.get("bm25_score", 0.0))) for d in sparse_docs]

        fused = self._fuse(dense_hits, sparse_hits,
                           top_k=max(top_k, len(dense_hits) + len(sparse_hits)))

        id_to_doc: dict[str, Document] = {}
        for d in dense_docs:
            id_to_doc[d.id] = d
        for d in sparse_docs:
            id_to_doc.setdefault(d.id, d)

        # scorer plugins add directly to the fused score (reference hybrid.py:275-285)
        if self.scorer_plugins:
            merged_docs = [id_to_doc[i] for i, _ in fused if i in id_to_doc]
            plugin_total: dict[str, float] = {}
            for p_idx, scorer in enumerate(self.scorer_plugins):
                try:
                    scores = scorer.score(query, merged_docs)
                except Exception:
                    continue
                for doc, s in zip(merged_docs, scores):
                    doc.metadata[f"plugin_{p_idx}_score"] = float(s)
                    plugin_total[doc.id] = plugin_total.get(doc.id, 0.0) + float(s)
            fused = fusion.add_plugin_scores(fused, plugin_total)

        out: list[Document] = []
        for doc_id, score in fused[:top_k]:
            doc = id_to_doc.get(doc_id)
            if doc is None:
                continue
            doc.metadata["hybrid_score"] = float(score)
            doc.metadata["score"] = float(score)
            out.append(doc)
        return out
