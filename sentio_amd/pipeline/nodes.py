"""Pipeline stage factories — the five nodes of the RAG graph
(reference src/core/graph/nodes.py:37-478 semantics, rebuilt as plain
closures over the on-device engines; no LangGraph dependency)."""

from __future__ import annotations

import logging
import time
from typing import Callable

from sentio_amd.models.document import Document
from sentio_amd.observability.metrics import metrics_collector
from sentio_amd.pipeline.context import numbered_context, prepare_context
from sentio_amd.pipeline.prompt_builder import PromptBuilder
from sentio_amd.pipeline.state import (
    RAGState,
    add_metadata,
    add_reranked_documents,
    add_retrieved_documents,
    add_selected_documents,
    set_response,
)

logger = logging.getLogger(__name__)

Node = Callable[[RAGState], RAGState]


def create_retriever_node(retriever, top_k: int = 10) -> Node:
    """Retrieve stage.  Per-request user_top_k override
    (reference nodes.py:63-68)."""

    def retrieve_node(state: RAGState) -> RAGState:
        t0 = time.perf_counter()
        user_top_k = state.get("metadata", {}).get("user_top_k")
        k = int(user_top_k) if isinstance(user_top_k, (int, float)) else top_k
        try:
            docs = retriever.retrieve(state["query"], top_k=k)
        except Exception as exc:
            logger.error("retriever failed: %s", exc)
            add_metadata(state, "retriever_error", str(exc))
            docs = []
        add_retrieved_documents(state, docs)
        add_metadata(state, "retrieved_count", len(docs))
        elapsed = time.perf_counter() - t0
        add_metadata(state, "retrieve_ms", elapsed * 1e3)
        metrics_collector.observe("rag_stage_duration_seconds", elapsed,
                                  stage="retrieve")
        return state

    return retrieve_node


def create_reranker_node(reranker, top_k: int = 5) -> Node:
    """Rerank stage; on any error falls back to the retrieved order
    truncated to top_k (reference nodes.py:208-226)."""

    def rerank_node(state: RAGState) -> RAGState:
        t0 = time.perf_counter()
        docs = state.get("retrieved_documents", [])
        if not docs:
            add_reranked_documents(state, [])
            return state
        try:
            reranked = reranker.rerank(state["query"], docs, top_k=top_k)
        except Exception as exc:
            logger.error("reranker failed, passing retrieved through: %s", exc)
            add_metadata(state, "reranker_error", str(exc))
            reranked = docs[:top_k]
        add_reranked_documents(state, reranked)
        add_metadata(state, "reranked_count", len(reranked))
        elapsed = time.perf_counter() - t0
        add_metadata(state, "rerank_ms", elapsed * 1e3)
        metrics_collector.observe("rag_stage_duration_seconds", elapsed,
                                  stage="rerank")
        return state

    return rerank_node


def create_selector_node(top_k: int = 3, max_tokens: int = 2000) -> Node:
    """Select stage (reference nodes.py:249-372): sort by score desc, dedup
    by id, greedy token-budget pack at 4 chars ≈ 1 token, honor user_top_k."""

    def select_node(state: RAGState) -> RAGState:
        candidates = state.get("reranked_documents") or state.get("retrieved_documents") or []
        if not candidates:
            return state
        user_top_k = state.get("metadata", {}).get("user_top_k")
        k = int(user_top_k) if isinstance(user_top_k, (int, float)) else top_k

        ranked = sorted(
            candidates,
            key=lambda d: float(d.metadata.get("score", 0.0) or 0.0),
            reverse=True,
        )
        seen: set[str] = set()
        unique: list[Document] = []
        for d in ranked:
            if d.id and d.id in seen:
                continue
            if d.id:
                seen.add(d.id)
            unique.append(d)

        selected: list[Document] = []
        total_tokens = 0
        for doc in unique[:k]:
            text = doc.text or str(doc.metadata.get("content", "") or "")
            if not text.strip():
                continue
            doc_tokens = len(text) // 4
            if total_tokens + doc_tokens > max_tokens:
                break
            selected.append(Document(text=text, metadata=dict(doc.metadata), id=doc.id))
            total_tokens += doc_tokens

        add_selected_documents(state, selected)
        add_metadata(state, "selected_count", len(selected))
        add_metadata(state, "selected_tokens", total_tokens)
        return state

    return select_node


def create_generator_node(generator, mode: str = "balanced",
                          max_tokens: int = 1024) -> Node:
    """Generate stage (reference nodes.py:377-434 + generator.py:52-135)."""
    builder = PromptBuilder(mode)

    def generate_node(state: RAGState) -> RAGState:
        t0 = time.perf_counter()
        docs = state.get("selected_documents") or state.get("retrieved_documents") or []
        context = prepare_context(docs)
        temperature = state.get("metadata", {}).get("temperature")
        if not isinstance(temperature, (int, float)):
            from sentio_amd.engines.generator import MODE_TEMPERATURE

            temperature = MODE_TEMPERATURE.get(mode, 0.3)
        history = state.get("metadata", {}).get("history") or []
        hist_txt = "".join(
            f"{turn.get('role', 'user')}: {turn.get('content', '')}\n"
            for turn in history[-6:] if isinstance(turn, dict)
        )
        prompt = builder.system_prompt() + "\n\n"
        if hist_txt:
            prompt += "Conversation so far:\n" + hist_txt + "\n"
        prompt += builder.build_qa_prompt(state["query"], context, mode)
        try:
            answer = generator.generate(
                [prompt], max_new_tokens=max_tokens, temperature=float(temperature)
            )[0]
        except Exception as exc:
            logger.error("generation failed: %s", exc)
            add_metadata(state, "generator_error", str(exc))
            answer = "I could not generate an answer for this query."
        set_response(state, answer)
        elapsed = time.perf_counter() - t0
        add_metadata(state, "generate_ms", elapsed * 1e3)
        metrics_collector.observe("rag_stage_duration_seconds", elapsed,
                                  stage="generate")
        return state

    return generate_node


def create_verifier_node(verifier) -> Node:
    """Verify stage (reference nodes.py:437-476): numbered context, replace
    answer with revised_answer on verdict=fail."""

    def verify_node(state: RAGState) -> RAGState:
        docs = state.get("selected_documents") or []
        answer = state.get("response", "")
        if not answer:
            return state
        result = verifier.verify(
            query=state["query"], context=numbered_context(docs), answer=answer
        )
        state.setdefault("evaluation", {})["verification"] = dict(result)
        if result.get("verdict") == "fail" and result.get("revised_answer"):
            set_response(state, result["revised_answer"])
            add_metadata(state, "answer_revised", True)
        return state

    return verify_node
