"""Pipeline state container.

Same keys and mutator semantics as the reference's ``RAGState`` TypedDict
(reference src/core/graph/state.py:10-139): query, retrieved/reranked/
selected_documents, response, metadata, evaluation.
"""

from __future__ import annotations

import time
from typing import Any, TypedDict

from sentio_amd.models.document import Document


class RAGState(TypedDict, total=False):
    query: str
    retrieved_documents: list[Document]
    reranked_documents: list[Document]
    selected_documents: list[Document]
    response: str
    metadata: dict[str, Any]
    evaluation: dict[str, Any]


def create_initial_state(query: str, metadata: dict[str, Any] | None = None) -> RAGState:
    return RAGState(
        query=query,
        retrieved_documents=[],
        reranked_documents=[],
        selected_documents=[],
        response="",
        metadata=dict(metadata or {}),
        evaluation={},
    )


def add_retrieved_documents(state: RAGState, docs: list[Document]) -> RAGState:
    state["retrieved_documents"] = list(docs)
    return state


def add_reranked_documents(state: RAGState, docs: list[Document]) -> RAGState:
    state["reranked_documents"] = list(docs)
    return state


def add_selected_documents(state: RAGState, docs: list[Document]) -> RAGState:
    state["selected_documents"] = list(docs)
    return state


def set_response(state: RAGState, response: str) -> RAGState:
    state["response"] = response
    state.setdefault("metadata", {})["response_time"] = time.time()
    return state


def add_metadata(state: RAGState, key: str, value: Any) -> RAGState:
    state.setdefault("metadata", {})[key] = value
    return state


def add_evaluation(state: RAGState, key: str, value: Any) -> RAGState:
    state.setdefault("evaluation", {})[key] = value
    return state
