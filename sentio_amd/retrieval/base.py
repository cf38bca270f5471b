"""Retriever and scorer-plugin interfaces
(reference src/core/retrievers/base.py:13-42)."""

from __future__ import annotations

import asyncio
from abc import ABC, abstractmethod
from typing import Protocol, runtime_checkable

from sentio_amd.models.document import Document


@runtime_checkable
class ScorerPlugin(Protocol):
    def score(self, query: str, documents: list[Document]) -> list[float]:
        ...


class BaseRetriever(ABC):
    @abstractmethod
    def retrieve(self, query: str, top_k: int = 10) -> list[Document]:
        ...

    async def retrieve_async(self, query: str, top_k: int = 10) -> list[Document]:
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(None, self.retrieve, query, top_k)
