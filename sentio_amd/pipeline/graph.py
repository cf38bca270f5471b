"""The RAG pipeline executor.

Reproduces the reference's LangGraph wiring (src/core/graph/factory.py:94-188:
entry=retriever, conditional reranker edge, selector, generator, conditional
verifier, compile) as a plain typed stage list — a 5-stage conditional
pipeline needs no graph framework, and the node bodies stay hot-path-free of
framework overhead.
"""

from __future__ import annotations

import asyncio
import logging
import time
from dataclasses import dataclass, field
from typing import Callable

from sentio_amd.config import Settings
from sentio_amd.pipeline import nodes as N
from sentio_amd.observability.tracing import trace_operation
from sentio_amd.pipeline.state import RAGState, add_metadata

logger = logging.getLogger(__name__)


@dataclass
class GraphConfig:
    """Component wiring (reference graph/factory.py:28-91)."""

    retriever: object = None
    reranker: object = None
    generator: object = None
    verifier: object = None
    use_reranker: bool = True
    use_verifier: bool = False
    retrieval_top_k: int = 10
    reranking_top_k: int = 5
    selection_top_k: int = 3
    selector_max_tokens: int = 2000
    generation_mode: str = "balanced"
    llm_max_tokens: int = 1024

    @classmethod
    def from_settings(cls, s: Settings, *, retriever, reranker=None,
                      generator=None, verifier=None) -> "GraphConfig":
        return cls(
            retriever=retriever, reranker=reranker, generator=generator,
            verifier=verifier,
            use_reranker=s.use_reranker and reranker is not None,
            use_verifier=s.use_verifier and verifier is not None,
            retrieval_top_k=s.retrieval_top_k,
            reranking_top_k=s.reranking_top_k,
            selection_top_k=s.selection_top_k,
            selector_max_tokens=s.selector_max_tokens,
            generation_mode=s.generation_mode,
            llm_max_tokens=s.llm_max_tokens,
        )


@dataclass
class RagPipeline:
    stages: list[tuple[str, Callable[[RAGState], RAGState]]] = field(default_factory=list)

    def invoke(self, state: RAGState) -> RAGState:
        t0 = time.perf_counter()
        with trace_operation("pipeline.invoke",
                             query_id=state.get("metadata", {}).get("query_id")):
            for name, fn in self.stages:
                try:
                    with trace_operation(f"stage.{name}"):
                        state = fn(state)
                except Exception as exc:
                    logger.error("stage %s raised: %s", name, exc)
                    add_metadata(state, f"{name}_error", str(exc))
        add_metadata(state, "pipeline_ms", (time.perf_counter() - t0) * 1e3)
        return state

    async def ainvoke(self, state: RAGState) -> RAGState:
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(None, self.invoke, state)


def build_basic_graph(cfg: GraphConfig) -> RagPipeline:
    stages: list[tuple[str, Callable]] = [
        ("retriever", N.create_retriever_node(cfg.retriever, cfg.retrieval_top_k)),
    ]
    if cfg.use_reranker and cfg.reranker is not None:
        stages.append(("reranker", N.create_reranker_node(cfg.reranker, cfg.reranking_top_k)))
    stages.append(("selector", N.create_selector_node(cfg.selection_top_k,
                                                      cfg.selector_max_tokens)))
    stages.append(("generator", N.create_generator_node(cfg.generator,
                                                        cfg.generation_mode,
                                                        cfg.llm_max_tokens)))
    if cfg.use_verifier and cfg.verifier is not None:
        stages.append(("verifier", N.create_verifier_node(cfg.verifier)))
    return RagPipeline(stages)


def build_streaming_graph(cfg: GraphConfig) -> RagPipeline:
    """Alias of the basic graph (reference factory.py:191-208 does the same);
    token streaming happens inside the generator engine."""
    return build_basic_graph(cfg)
