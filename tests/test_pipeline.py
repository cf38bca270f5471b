"""Pipeline graph: stage flow, conditional stages, selector policy,
fallbacks — mirrors the reference's graph tests
(reference src/tests/graph/test_basic_graph.py approach)."""

import pytest

from sentio_amd.models.document import Document
from sentio_amd.pipeline.graph import GraphConfig, build_basic_graph
from sentio_amd.pipeline.nodes import create_selector_node
from sentio_amd.pipeline.state import create_initial_state


class StubRetriever:
    def __init__(self, docs):
        self.docs = docs
        self.last_top_k = None

    def retrieve(self, query, top_k=10):
        self.last_top_k = top_k
        return self.docs[:top_k]


class StubReranker:
    def rerank(self, query, docs, top_k):
        return list(reversed(docs))[:top_k]


class FailingReranker:
    def rerank(self, query, docs, top_k):
        raise RuntimeError("boom")


class StubGenerator:
    def __init__(self):
        self.prompts = []

    def generate(self, prompts, max_new_tokens=128, temperature=0.3, **kw):
        self.prompts.extend(prompts)
        return ["stub answer [1]"] * len(prompts)


def _docs(n, score_desc=True):
    return [
        Document(text=f"document number {i} content", id=f"doc{i}",
                 metadata={"score": (n - i) / n if score_desc else 0.5})
        for i in range(n)
    ]


def _cfg(**kw):
    defaults = dict(retriever=StubRetriever(_docs(8)), reranker=StubReranker(),
                    generator=StubGenerator(), use_reranker=True,
                    use_verifier=False)
    defaults.update(kw)
    return GraphConfig(**defaults)


def test_full_flow_populates_state():
    pipe = build_basic_graph(_cfg())
    state = pipe.invoke(create_initial_state("what is doc 1?"))
    assert state["retrieved_documents"]
    assert state["reranked_documents"]
    assert state["selected_documents"]
    assert state["response"] == "stub answer [1]"
    assert "pipeline_ms" in state["metadata"]


def test_user_top_k_override_reaches_retriever():
    r = StubRetriever(_docs(8))
    pipe = build_basic_graph(_cfg(retriever=r))
    pipe.invoke(create_initial_state("q", {"user_top_k": 2}))
    assert r.last_top_k == 2


def test_reranker_disabled_skips_stage():
    pipe = build_basic_graph(_cfg(use_reranker=False))
    names = [n for n, _ in pipe.stages]
    assert "reranker" not in names
    state = pipe.invoke(create_initial_state("q"))
    assert state["response"]


def test_reranker_failure_falls_back_to_retrieved():
    pipe = build_basic_graph(_cfg(reranker=FailingReranker(), reranking_top_k=3))
    state = pipe.invoke(create_initial_state("q"))
    # fallback: retrieved order truncated to top_k (reference nodes.py:208-226)
    assert [d.id for d in state["reranked_documents"]] == ["doc0", "doc1", "doc2"]
    assert "reranker_error" in state["metadata"]


def test_selector_sorts_dedups_and_budgets():
    node = create_selector_node(top_k=5, max_tokens=20)
    docs = [
        Document(text="x" * 40, id="a", metadata={"score": 0.1}),
        Document(text="y" * 40, id="b", metadata={"score": 0.9}),
        Document(text="z" * 40, id="b", metadata={"score": 0.9}),  # dup id
        Document(text="w" * 200, id="c", metadata={"score": 0.5}),
    ]
    state = create_initial_state("q")
    state["retrieved_documents"] = docs
    state = node(state)
    sel = state["selected_documents"]
    # b first (score), dedup removes second b; c (50 tokens) busts the 20-token
    # budget after b (10 tokens) → selection stops at the break
    assert [d.id for d in sel] == ["b", "a"] or [d.id for d in sel] == ["b"]
    assert state["metadata"]["selected_tokens"] <= 20


def test_selector_empty_text_fallback_to_metadata_content():
    node = create_selector_node(top_k=2, max_tokens=100)
    docs = [Document(text="", id="m", metadata={"score": 1.0, "content": "meta text"})]
    state = create_initial_state("q")
    state["retrieved_documents"] = docs
    state = node(state)
    assert state["selected_documents"][0].text == "meta text"


def test_generation_error_yields_apology_not_crash():
    class Boom:
        def generate(self, *a, **k):
            raise RuntimeError("gpu fell over")

    pipe = build_basic_graph(_cfg(generator=Boom()))
    state = pipe.invoke(create_initial_state("q"))
    assert "could not generate" in state["response"]
    assert "generator_error" in state["metadata"]


def test_verifier_fail_replaces_answer():
    class StubVerifier:
        def verify(self, query, context, answer):
            return {"verdict": "fail", "citations_ok": False,
                    "notes": [], "revised_answer": "revised [1]"}

    cfg = _cfg(use_verifier=True, verifier=StubVerifier())
    pipe = build_basic_graph(cfg)
    state = pipe.invoke(create_initial_state("q"))
    assert state["response"] == "revised [1]"
    assert state["metadata"].get("answer_revised") is True


def test_ainvoke():
    import asyncio

    pipe = build_basic_graph(_cfg())
    state = asyncio.run(pipe.ainvoke(create_initial_state("async q")))
    assert state["response"]
