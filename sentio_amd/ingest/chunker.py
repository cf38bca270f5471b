"""Text chunking (reference src/core/chunking/text_splitter.py:34-196:
recursive separator splitting with size/overlap, 'fixed' strategy,
parent_id preserved in metadata).  Own implementation of the recursive
strategy — split on the coarsest separator that yields pieces under the
budget, then merge with overlap."""

from __future__ import annotations

import uuid
from dataclasses import dataclass

from sentio_amd.models.document import Document

SEPARATORS = ["\n\n", "\n", ". ", " ", ""]


class ChunkingError(ValueError):
    pass


@dataclass
class TextChunker:
    chunk_size: int = 512
    chunk_overlap: int = 64
    strategy: str = "recursive"  # recursive | fixed

    def __post_init__(self):
        if self.chunk_overlap >= self.chunk_size:
            raise ChunkingError("overlap must be smaller than chunk size")
        self.stats = {"documents": 0, "chunks": 0}

    def split_text(self, text: str) -> list[str]:
        if not text:
            return []
        if self.strategy == "fixed":
            return self._fixed(text)
        return self._merge(self._recursive(text, 0))

    def _fixed(self, text: str) -> list[str]:
        step = self.chunk_size - self.chunk_overlap
        return [text[i : i + self.chunk_size] for i in range(0, len(text), step)
                if text[i : i + self.chunk_size].strip()]

    def _recursive(self, text: str, level: int) -> list[str]:
        if len(text) <= self.chunk_size:
            return [text] if text.strip() else []
        if level >= len(SEPARATORS):
            return self._fixed(text)
        sep = SEPARATORS[level]
        if sep == "":
            return self._fixed(text)
        parts = text.split(sep)
        out: list[str] = []
        for p in parts:
            piece = p + sep if p is not parts[-1] else p
            if len(piece) <= self.chunk_size:
                if piece.strip():
                    out.append(piece)
            else:
                out.extend(self._recursive(piece, level + 1))
        return out

    def _merge(self, pieces: list[str]) -> list[str]:
        """Greedily merge small pieces up to chunk_size with overlap carry."""
        chunks: list[str] = []
        cur = ""
        for p in pieces:
            if len(cur) + len(p) <= self.chunk_size:
                cur += p
            else:
                if cur.strip():
                    chunks.append(cur.strip())
                tail = cur[-self.chunk_overlap:] if self.chunk_overlap else ""
                cur = tail + p
                if len(cur) > self.chunk_size:
                    # overlap is best-effort: trim carried-over tail chars
                    # from the FRONT, never the new piece's content
                    cur = cur[-self.chunk_size:]
        if cur.strip():
            chunks.append(cur.strip())
        return chunks

    def split(self, docs: list[Document]) -> list[Document]:
        out: list[Document] = []
        for doc in docs:
            for i, chunk in enumerate(self.split_text(doc.text)):
                meta = dict(doc.metadata)
                meta["parent_id"] = doc.id
                meta["chunk_index"] = i
                out.append(Document(text=chunk, metadata=meta,
                                    id=f"{doc.id}:{i}" if doc.id else str(uuid.uuid4())))
            self.stats["documents"] += 1
        self.stats["chunks"] += len(out)
        return out
