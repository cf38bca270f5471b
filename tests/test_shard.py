"""ShardedIndex unit tests on a single process (the gloo multi-process tests
cover the collective paths; these cover the local API surface)."""

from __future__ import annotations

import numpy as np
import torch

from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.models.document import Document
from sentio_amd.parallel.shard import ShardedIndex


def _docs(n=20):
    rng = np.random.RandomState(3)
    out = []
    for i in range(n):
        words = rng.choice(["alpha", "beta", "gamma", "delta"], size=12)
        out.append(Document(text=" ".join(words), id=f"doc{i}"))
    return out


def test_sharded_index_single_rank_matches_local():
    dim = 32
    dense = DenseIndex(dim=dim, device="cpu")
    bm = BM25Index()
    docs = _docs()
    rng = np.random.RandomState(1)
    vecs = torch.tensor(rng.randn(len(docs), dim), dtype=torch.float32)
    dense.add(docs, vecs)
    bm.build([d.id for d in docs], [d.text for d in docs])
    sharded = ShardedIndex(dense, bm)
    q = vecs[3].unsqueeze(0)
    hits = sharded.search_dense(q, top_k=3)[0]
    assert hits[0][0] == "0:d3"          # shard-prefixed row ref, self-match first
    s_hits = sharded.search_sparse("alpha beta", top_k=5)
    assert s_hits and all(ref.startswith("0:") for ref, _ in s_hits)
    fetched = sharded.fetch_documents([hits[0][0]])
    assert fetched[hits[0][0]].id == "doc3"
    assert sharded.total_docs() == len(docs)
