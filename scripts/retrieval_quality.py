"""Retrieval-quality evaluation: recall@k / MRR for dense, BM25 and hybrid
over a deterministic synthetic labeled corpus.

The reference repo's retriever factory cites a quality study
(reference src/core/retrievers/factory.py:29-37, "best config: dense,
RRF_K=20, RETRIEVAL_TOP_K=10") whose results file is NOT in its tree.
This harness makes the equivalent measurement reproducible here: topic-
clustered documents (each doc draws most words from its topic's vocabulary,
the rest from a shared pool), queries that target one topic, ground truth =
same-topic docs.  Runs the REAL pipeline components (tiny encoder on CPU or
the full encoder on GPU, GPU BM25 when available, the fusion oracle /
device K4 kernel through HybridRetriever).

Usage:
    python scripts/retrieval_quality.py [--docs 600] [--queries 60] [--k 10]
Writes JSON to stdout (and profiles/retrieval_quality.json with --save).
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


TOPICS = {
    "gpu": "kernel wavefront occupancy register shader compute matrix tensor "
           "hbm bandwidth stream graph dispatch barrier lane".split(),
    "net": "packet router latency switch ethernet socket protocol gateway "
           "bandwidth firewall subnet routing congestion".split(),
    "bio": "protein enzyme cell membrane genome ribosome mutation organism "
           "chromosome bacteria molecule receptor".split(),
    "law": "contract statute liability plaintiff court verdict appeal "
           "clause jurisdiction precedent tort counsel".split(),
    "food": "recipe flavor roast simmer spice dough ferment butter "
            "seasoning skillet marinade garnish".split(),
    "music": "melody rhythm chord tempo harmony octave compose orchestra "
             "cadence timbre scale refrain".split(),
}
SHARED = ("the a of and to in for with on that is are was were it this "
          "process system result common case study note point item").split()


def build_corpus(n_docs: int, rng: np.random.RandomState):
    names = list(TOPICS)
    docs, labels = [], []
    for i in range(n_docs):
        t = names[i % len(names)]
        vocab = TOPICS[t]
        words = [vocab[rng.randint(len(vocab))] if rng.rand() < 0.55
                 else SHARED[rng.randint(len(SHARED))]
                 for _ in range(rng.randint(25, 60))]
        docs.append(" ".join(words))
        labels.append(t)
    return docs, labels


def build_queries(n_q: int, rng: np.random.RandomState):
    names = list(TOPICS)
    out = []
    for i in range(n_q):
        t = names[i % len(names)]
        vocab = TOPICS[t]
        picks = [vocab[rng.randint(len(vocab))] for _ in range(3)]
        out.append((f"what about {picks[0]} and {picks[1]} {picks[2]}?", t))
    return out


def evaluate(retriever, queries, labels_by_id, k: int):
    recalls, mrrs = [], []
    for q, topic in queries:
        docs = retriever.retrieve(q, top_k=k)
        got = [labels_by_id.get(d.id) for d in docs]
        n_rel_total = sum(1 for v in labels_by_id.values() if v == topic)
        n_rel = sum(1 for g in got if g == topic)
        recalls.append(n_rel / min(k, n_rel_total))
        rr = 0.0
        for rank, g in enumerate(got, 1):
            if g == topic:
                rr = 1.0 / rank
                break
        mrrs.append(rr)
    return {"recall_at_k": round(float(np.mean(recalls)), 4),
            "mrr": round(float(np.mean(mrrs)), 4)}


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=600)
    ap.add_argument("--queries", type=int, default=60)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--save", action="store_true",
                    help="also write profiles/retrieval_quality.json")
    args = ap.parse_args()

    import torch

    from sentio_amd.engines.encoder import EncoderEngine
    from sentio_amd.index.bm25 import BM25Index
    from sentio_amd.index.dense import DenseIndex
    from sentio_amd.models.document import Document
    from sentio_amd.retrieval.dense import DenseRetriever
    from sentio_amd.retrieval.hybrid import HybridRetriever
    from sentio_amd.retrieval.sparse import BM25Retriever

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    rng = np.random.RandomState(71)
    texts, labels = build_corpus(args.docs, rng)
    queries = build_queries(args.queries, rng)
    docs = [Document(text=t, metadata={"topic": lab}, id=f"d{i}")
            for i, (t, lab) in enumerate(zip(texts, labels))]
    labels_by_id = {d.id: d.metadata["topic"] for d in docs}

    enc = EncoderEngine(
        "sentio-encoder-small" if device != "cpu" else "tiny-encoder",
        device=device, max_seq=128)
    dense_idx = DenseIndex(dim=enc.dim, device=device)
    dense_idx.add(docs, enc.embed(texts))
    bm_idx = BM25Index()
    bm_idx.build([d.id for d in docs], texts)

    dense = DenseRetriever(enc, dense_idx)
    sparse = BM25Retriever(bm_idx, doc_lookup=dense_idx.get_document,
                           device=device)
    results = {
        "config": {"docs": args.docs, "queries": args.queries, "k": args.k,
                   "device": device, "encoder": enc.cfg.name
                   if hasattr(enc.cfg, "name") else "tiny",
                   "note": "synthetic topic-clustered corpus; random-init "
                           "encoder weights — dense quality reflects the "
                           "PIPELINE (lexical overlap via the trained BPE "
                           "token space), not a trained embedding model"},
        "dense": evaluate(dense, queries, labels_by_id, args.k),
        "bm25": evaluate(sparse, queries, labels_by_id, args.k),
    }
    for method in ("rrf", "weighted_rrf", "comb_sum"):
        hyb = HybridRetriever(dense=dense, sparse=sparse,
                              fusion_method=method, rrf_k=60)
        results[f"hybrid_{method}"] = evaluate(hyb, queries, labels_by_id,
                                               args.k)
    print(json.dumps(results, indent=1))
    if args.save:
        out = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "profiles",
            "retrieval_quality.json")
        with open(out, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
