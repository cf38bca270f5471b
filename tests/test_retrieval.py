"""Retrievers over the indexes: dense, sparse, hybrid, scorers, factory."""

import torch

from sentio_amd.config import Settings
from sentio_amd.engines.mock import MockEncoderEngine
from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.models.document import Document
from sentio_amd.retrieval.dense import DenseRetriever
from sentio_amd.retrieval.factory import create_retriever
from sentio_amd.retrieval.hybrid import HybridRetriever
from sentio_amd.retrieval.scorers import (
    KeywordMatchScorer,
    MMRScorer,
    SemanticSimilarityScorer,
)
from sentio_amd.retrieval.sparse import BM25Retriever


def _build(n=20):
    enc = MockEncoderEngine(dim=64)
    dense = DenseIndex(dim=64, device="cpu")
    bm25 = BM25Index()
    docs = [
        Document(text=f"topic {i % 5} document body number {i}", id=f"d{i}",
                 metadata={"source": f"s{i}"})
        for i in range(n)
    ]
    dense.add(docs, enc.embed([d.text for d in docs]))
    bm25.build([d.id for d in docs], [d.text for d in docs])
    return enc, dense, bm25, docs


def test_dense_retriever_self_retrieval():
    enc, dense, _, docs = _build()
    r = DenseRetriever(enc, dense)
    hits = r.retrieve(docs[7].text, top_k=3)
    assert hits[0].id == "d7"  # identical text → identical mock vector
    assert hits[0].metadata["score"] > 0.99
    assert hits[0].metadata["retrieval_method"] == "dense"


def test_sparse_retriever_finds_term():
    enc, dense, bm25, docs = _build()
    r = BM25Retriever(bm25, doc_lookup=dense.get_document)
    hits = r.retrieve("number 13", top_k=5)
    assert any(h.id == "d13" for h in hits)
    assert all("bm25_score" in h.metadata for h in hits)


def test_hybrid_fuses_and_tags_scores():
    enc, dense, bm25, docs = _build()
    hy = HybridRetriever(
        dense=DenseRetriever(enc, dense),
        sparse=BM25Retriever(bm25, doc_lookup=dense.get_document),
        fusion_method="rrf",
    )
    hits = hy.retrieve(docs[3].text, top_k=5)
    assert hits
    assert all("hybrid_score" in h.metadata for h in hits)
    assert hits[0].id == "d3"


def test_factory_strategies():
    enc, dense, bm25, docs = _build()
    for strategy, cls in (("dense", DenseRetriever), ("bm25", BM25Retriever),
                          ("hybrid", HybridRetriever)):
        s = Settings()
        s.retrieval_strategy = strategy
        r = create_retriever(s, enc, dense, bm25)
        assert isinstance(r, cls)


def test_keyword_scorer():
    sc = KeywordMatchScorer(weight=1.0)
    docs = [Document(text="alpha beta gamma"), Document(text="delta")]
    scores = sc.score("alpha beta", docs)
    assert scores[0] == 1.0 and scores[1] == 0.0


def test_semantic_scorer_prefers_same_text():
    enc = MockEncoderEngine(dim=64)
    sc = SemanticSimilarityScorer(enc, weight=1.0)
    docs = [Document(text="identical query text"), Document(text="something else")]
    scores = sc.score("identical query text", docs)
    assert scores[0] > scores[1]


def test_mmr_scorer_rewards_diversity():
    enc = MockEncoderEngine(dim=64)
    sc = MMRScorer(enc, lambda_param=0.5, weight=1.0)
    docs = [Document(text="aaa"), Document(text="aaa"), Document(text="zzz")]
    scores = sc.score("aaa", docs)
    assert len(scores) == 3 and max(scores) <= 1.0


def test_dense_index_save_load(tmp_path):
    enc, dense, _, docs = _build(8)
    p = str(tmp_path / "idx.pt")
    dense.save(p)
    loaded = DenseIndex.load(p)
    assert len(loaded) == len(dense)
    q = enc.embed([docs[2].text])
    got = loaded.search(q, 1)[0][0][0]
    assert got == "d2"


def test_dense_search_metadata_filter():
    import torch

    from sentio_amd.index.dense import DenseIndex
    from sentio_amd.models.document import Document

    idx = DenseIndex(dim=8, device="cpu")
    docs = [Document(text=f"d{i}", id=f"d{i}",
                     metadata={"lang": "en" if i % 2 == 0 else "de"})
            for i in range(10)]
    vecs = torch.randn(10, 8)
    idx.add(docs, vecs)
    hits = idx.search(vecs[1], top_k=5, metadata_filter={"lang": "de"})[0]
    assert hits and all(h[0] in {"d1", "d3", "d5", "d7", "d9"} for h in hits)
    assert hits[0][0] == "d1"   # self-match still ranks first among de docs


# ---- web-cache second collection (reference hybrid.py:96-107,146-182) ----

class _CannedRetriever:
    def __init__(self, docs):
        self.docs = docs
        self.calls = 0

    def retrieve(self, query, top_k=10):
        self.calls += 1
        return self.docs[:top_k]


def _mk(i, score, cache=False):
    from sentio_amd.models.document import Document
    d = Document(text=f"doc {i}", metadata={"score": score}, id=f"{'c' if cache else 'm'}{i}")
    return d


def test_web_cache_short_circuits_on_strong_hits():
    from sentio_amd.retrieval.hybrid import HybridRetriever

    cache = _CannedRetriever([_mk(i, 0.95, cache=True) for i in range(3)])
    main = _CannedRetriever([_mk(i, 0.5) for i in range(3)])
    h = HybridRetriever(dense=main, cache_retriever=cache,
                        cache_score_threshold=0.9)
    out = h.retrieve("q", top_k=2)
    assert [d.id for d in out] == ["c0", "c1"]
    assert all(d.metadata["from_cache_collection"] for d in out)
    assert main.calls == 0          # main corpus never touched


def test_web_cache_weak_hits_join_fusion():
    from sentio_amd.retrieval.hybrid import HybridRetriever

    cache = _CannedRetriever([_mk(0, 0.4, cache=True)])
    main = _CannedRetriever([_mk(i, 0.8 - 0.1 * i) for i in range(3)])
    h = HybridRetriever(dense=main, cache_retriever=cache,
                        cache_score_threshold=0.9)
    out = h.retrieve("q", top_k=4)
    assert main.calls == 1
    ids = [d.id for d in out]
    assert "c0" in ids and "m0" in ids
    # cache doc carries its marker through fusion
    assert next(d for d in out if d.id == "c0").metadata["from_cache_collection"]


def test_web_cache_errors_are_soft():
    from sentio_amd.retrieval.hybrid import HybridRetriever

    class _Boom:
        def retrieve(self, q, top_k=10):
            raise RuntimeError("cache down")

    main = _CannedRetriever([_mk(i, 0.8) for i in range(2)])
    h = HybridRetriever(dense=main, cache_retriever=_Boom())
    out = h.retrieve("q", top_k=2)
    assert len(out) == 2 and main.calls == 1


def test_retrieval_quality_harness_sane():
    """The quality harness (scripts/retrieval_quality.py) is deterministic
    and orders the retrievers sensibly on the lexical topic task: BM25
    near-perfect, hybrid >= dense (fusion must not destroy the lexical
    signal)."""
    import importlib.util
    import os

    spec = importlib.util.spec_from_file_location(
        "retrieval_quality",
        os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "scripts", "retrieval_quality.py"))
    rq = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(rq)

    import numpy as np

    from sentio_amd.engines.encoder import EncoderEngine
    from sentio_amd.index.bm25 import BM25Index
    from sentio_amd.index.dense import DenseIndex
    from sentio_amd.models.document import Document
    from sentio_amd.retrieval.dense import DenseRetriever
    from sentio_amd.retrieval.hybrid import HybridRetriever
    from sentio_amd.retrieval.sparse import BM25Retriever

    rng = np.random.RandomState(71)
    texts, labels = rq.build_corpus(180, rng)
    queries = rq.build_queries(24, rng)
    docs = [Document(text=t, metadata={"topic": lab}, id=f"d{i}")
            for i, (t, lab) in enumerate(zip(texts, labels))]
    by_id = {d.id: d.metadata["topic"] for d in docs}

    enc = EncoderEngine("tiny-encoder", device="cpu", max_seq=128)
    didx = DenseIndex(dim=enc.dim, device="cpu")
    didx.add(docs, enc.embed(texts))
    bidx = BM25Index()
    bidx.build([d.id for d in docs], texts)
    dense = DenseRetriever(enc, didx)
    sparse = BM25Retriever(bidx, doc_lookup=didx.get_document)

    r_bm25 = rq.evaluate(sparse, queries, by_id, 10)
    r_dense = rq.evaluate(dense, queries, by_id, 10)
    hyb = HybridRetriever(dense=dense, sparse=sparse, fusion_method="rrf")
    r_hyb = rq.evaluate(hyb, queries, by_id, 10)

    assert r_bm25["recall_at_k"] >= 0.95          # lexical task
    assert 0.0 <= r_dense["recall_at_k"] <= 1.0
    assert r_hyb["recall_at_k"] >= r_dense["recall_at_k"] - 0.05
    # determinism
    assert rq.evaluate(sparse, queries, by_id, 10) == r_bm25
