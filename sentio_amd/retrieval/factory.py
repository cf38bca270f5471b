"""Retriever construction from settings
(reference src/core/retrievers/factory.py:21-214 strategy switch:
dense | bm25 | hybrid).  Unlike the reference — which scrolled the whole
corpus out of Qdrant over HTTP at query-service start (factory.py:94-128,
flagged in SURVEY §3.5) — the BM25 postings are built at ingest time and
live next to the dense index, so there is no cold-start corpus scan."""

from __future__ import annotations

from sentio_amd.config import Settings
from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.retrieval.dense import DenseRetriever
from sentio_amd.retrieval.hybrid import HybridRetriever
from sentio_amd.retrieval.sparse import BM25Retriever


def create_retriever(
    settings: Settings,
    embedder,
    dense_index: DenseIndex,
    bm25_index: BM25Index,
    device: str = "cpu",
    scorer_plugins: list | None = None,
):
    strategy = settings.retrieval_strategy
    dense = DenseRetriever(embedder, dense_index)
    sparse = BM25Retriever(bm25_index, doc_lookup=dense_index.get_document,
                           device=device)
    if strategy == "dense":
        return dense
    if strategy in ("bm25", "sparse", "pyserini"):
        # "pyserini" (reference factory.py Lucene backend) maps to the same
        # CSR BM25 — the GPU index IS the Lucene-class engine here, so a
        # reference deployment's RETRIEVAL_STRATEGY keeps working.
        return sparse
    if strategy == "hybrid":
        return HybridRetriever(
            dense=dense,
            sparse=sparse,
            fusion_method=settings.fusion_method,
            rrf_k=settings.rrf_k,
            dense_weight=settings.dense_weight,
            sparse_weight=settings.sparse_weight,
            scorer_plugins=scorer_plugins or [],
        )
    raise ValueError(f"unknown RETRIEVAL_STRATEGY: {strategy}")
