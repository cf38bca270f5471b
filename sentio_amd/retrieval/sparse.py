"""BM25 retriever over the CSR postings index
(reference src/core/retrievers/sparse.py:33-204 semantics: score>0 filter,
bm25_score into metadata, okapi/plus variants)."""

from __future__ import annotations

from sentio_amd.index.bm25 import BM25Index
from sentio_amd.models.document import Document
from sentio_amd.retrieval.base import BaseRetriever


class BM25Retriever(BaseRetriever):
    def __init__(self, index: BM25Index, doc_lookup=None, device: str = "cpu"):
        """doc_lookup: callable doc_id → Document (usually DenseIndex.get_document
        or the store's payload map)."""
        self.index = index
        self.doc_lookup = doc_lookup or (lambda _id: None)
        self.device = device

    def retrieve(self, query: str, top_k: int = 10) -> list[Document]:
        hits = self.index.search(query, top_k, device=self.device)
        out: list[Document] = []
        for doc_id, score in hits:
            src = self.doc_lookup(doc_id)
            if src is None:
                doc = Document(text="", id=doc_id)
            else:
                doc = Document(text=src.text, metadata=dict(src.metadata), id=src.id)
            doc.metadata["bm25_score"] = float(score)
            doc.metadata["score"] = float(score)
            doc.metadata["retrieval_method"] = "bm25"
            out.append(doc)
        return out
