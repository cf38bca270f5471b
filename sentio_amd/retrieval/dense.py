"""Dense retriever over the in-HBM index
(reference src/core/retrievers/dense.py:21-119 semantics: embed the query,
cosine top-k, text-key fallbacks, score into metadata)."""

from __future__ import annotations

from sentio_amd.index.dense import DenseIndex
from sentio_amd.models.document import Document
from sentio_amd.retrieval.base import BaseRetriever


def _doc_text(doc: Document) -> str:
    """Text-key fallback chain text → metadata.content
    (reference dense.py:80-86)."""
    if doc.text:
        return doc.text
    return str(doc.metadata.get("content", ""))


class DenseRetriever(BaseRetriever):
    def __init__(self, embedder, index: DenseIndex):
        self.embedder = embedder
        self.index = index

    def retrieve(self, query: str, top_k: int = 10) -> list[Document]:
        qv = self.embedder.embed([query])
        hits = self.index.search(qv, top_k)[0]
        out: list[Document] = []
        for doc_id, score in hits:
            src = self.index.get_document(doc_id)
            if src is None:
                continue
            doc = Document(text=_doc_text(src), metadata=dict(src.metadata), id=src.id)
            doc.metadata["score"] = float(score)
            doc.metadata["dense_score"] = float(score)
            doc.metadata["retrieval_method"] = "dense"
            out.append(doc)
        return out
