"""BM25 index: formula correctness, CSR build, top-k behavior, persistence."""

import math

import numpy as np
import pytest

from sentio_amd.index.bm25 import BM25Index, tokenize


CORPUS = [
    ("d0", "the quick brown fox jumps over the lazy dog"),
    ("d1", "a quick brown cat sat on the mat"),
    ("d2", "dogs and cats are friendly animals"),
    ("d3", "the stock market rallied on quick gains today"),
]


@pytest.fixture()
def index():
    idx = BM25Index()
    idx.build([d for d, _ in CORPUS], [t for _, t in CORPUS])
    return idx


def manual_okapi(index: BM25Index, query: str) -> np.ndarray:
    """Independent recomputation of Okapi BM25 from raw term counts."""
    toks = [tokenize(t) for _, t in CORPUS]
    n = len(toks)
    avgdl = sum(len(t) for t in toks) / n
    scores = np.zeros(n)
    for term in tokenize(query):
        df = sum(1 for t in toks if term in t)
        if df == 0:
            continue
        idf = math.log((n - df + 0.5) / (df + 0.5) + 1.0)
        for d, t in enumerate(toks):
            tf = t.count(term)
            if tf == 0:
                continue
            denom = tf + index.k1 * (1 - index.b + index.b * len(t) / avgdl)
            scores[d] += idf * tf * (index.k1 + 1) / denom
    return scores


def test_scores_match_manual_formula(index):
    for query in ("quick brown", "lazy dog", "stock market gains", "cats"):
        got = index.get_scores(query)
        want = manual_okapi(index, query)
        np.testing.assert_allclose(got, want, rtol=1e-5)


def test_search_filters_zero_scores(index):
    hits = index.search("quick", top_k=10)
    ids = [h[0] for h in hits]
    assert "d2" not in ids  # no 'quick' in d2
    assert all(s > 0 for _, s in hits)


def test_search_ranks_best_match_first(index):
    hits = index.search("lazy dog jumps", top_k=4)
    assert hits[0][0] == "d0"


def test_plus_variant_differs(index):
    plus = BM25Index(variant="plus")
    plus.build([d for d, _ in CORPUS], [t for _, t in CORPUS])
    s1 = index.get_scores("quick brown")
    s2 = plus.get_scores("quick brown")
    assert not np.allclose(s1, s2)


def test_incremental_add():
    idx = BM25Index()
    idx.build(["a"], ["hello world"])
    idx.add(["b"], ["hello hip kernels"])
    assert idx.n_docs == 2
    hits = idx.search("kernels", top_k=2)
    assert hits and hits[0][0] == "b"


def test_save_load_roundtrip(tmp_path, index):
    p = str(tmp_path / "bm25.pkl")
    index.save(p)
    loaded = BM25Index.load(p)
    np.testing.assert_allclose(
        loaded.get_scores("quick brown"), index.get_scores("quick brown"))


def test_empty_query_and_empty_index():
    idx = BM25Index()
    assert idx.search("anything", top_k=5) == []
    idx.build(["a"], ["some text"])
    assert idx.search("zzz unknown terms", top_k=5) == []


def test_add_after_load_keeps_old_postings(tmp_path):
    """Regression: adding documents after load() must not drop the loaded
    corpus's postings (token cache reconstructed from CSR)."""
    from sentio_amd.index.bm25 import BM25Index

    idx = BM25Index()
    idx.build(["a", "b"], ["gpu kernels stream data", "retrieval ranks docs"])
    p = str(tmp_path / "bm.pkl")
    idx.save(p)

    loaded = BM25Index.load(p)
    loaded.add(["c"], ["new doc about gpu retrieval"])
    assert loaded.n_docs == 3
    hits = dict(loaded.search("kernels", top_k=3))
    assert "a" in hits                     # old posting survived
    hits2 = [d for d, _ in loaded.search("retrieval", top_k=3)]
    assert set(hits2) >= {"b", "c"}
