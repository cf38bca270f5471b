"""Document ingestion pipeline: load → chunk → embed → index
(reference src/core/ingest/ingest.py:20-529 capability).  The reference
batched HTTP calls to Jina and upserted to remote Qdrant; here chunks embed
on-device in one batch and append to the in-HBM dense index + BM25 postings
in the same process."""

from __future__ import annotations

import logging
import time
from typing import Any

from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.ingest.chunker import TextChunker
from sentio_amd.ingest.readers import read_directory
from sentio_amd.models.document import Document

logger = logging.getLogger(__name__)


class DocumentIngestor:
    def __init__(
        self,
        embedder,
        dense_index: DenseIndex,
        bm25_index: BM25Index | None = None,
        chunker: TextChunker | None = None,
    ):
        self.embedder = embedder
        self.dense_index = dense_index
        self.bm25_index = bm25_index
        self.chunker = chunker or TextChunker()
        self.stats: dict[str, Any] = {
            "documents": 0, "chunks": 0, "errors": 0, "total_time_s": 0.0,
        }

    def ingest_document(self, doc: Document) -> dict[str, Any]:
        """Single-document path used by /embed (reference ingest.py:460-488)."""
        return self.ingest_documents([doc])

    def ingest_documents(self, docs: list[Document]) -> dict[str, Any]:
        t0 = time.time()
        chunks = self.chunker.split(docs)
        chunks = [c for c in chunks if c.text.strip()]
        if not chunks:
            return {"documents": len(docs), "chunks": 0, "status": "empty"}
        embeddings = self.embedder.embed([c.text for c in chunks])
        # keep content in metadata for payload-style lookups
        for c in chunks:
            c.metadata.setdefault("content", c.text)
        self.dense_index.add(chunks, embeddings)
        if self.bm25_index is not None:
            self.bm25_index.add([c.id for c in chunks], [c.text for c in chunks])
        dt = time.time() - t0
        self.stats["documents"] += len(docs)
        self.stats["chunks"] += len(chunks)
        self.stats["total_time_s"] += dt
        return {
            "documents": len(docs),
            "chunks": len(chunks),
            "status": "ok",
            "time_s": dt,
        }


def ingest_directory(directory: str, ingestor: DocumentIngestor,
                     recursive: bool = True) -> dict[str, Any]:
    """Bulk CLI ingest (reference ingest.py:491-529)."""
    docs = read_directory(directory, recursive)
    if not docs:
        return {"documents": 0, "chunks": 0, "status": "no_documents"}
    return ingestor.ingest_documents(docs)
