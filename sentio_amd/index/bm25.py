"""BM25 lexical index with a CSR postings layout.

Capability parity with the reference's in-memory BM25 retriever
(reference src/core/retrievers/sparse.py:70-184: lower().split()
tokenization, Okapi and "plus" variants, top-k via argsort, score>0 filter,
pickle save/load) — but organised the MI355X way: the index is a set of flat
CSR arrays (term → postings of (doc, tf)) that score on-device with an
atomic-accumulate HIP kernel (ops.bm25_score) and a fused top-k, instead of
a per-query Python loop over the corpus.

Scoring math (Okapi BM25, matching rank_bm25's conventions):
  idf(t)   = ln((N - df + 0.5)/(df + 0.5) + 1)
  score(d) = Σ_t idf(t) · tf·(k1+1) / (tf + k1·(1 − b + b·|d|/avgdl))
BM25+ ("plus" variant): idf(t) = ln((N+1)/df), and the tf term gains +δ.
"""

from __future__ import annotations

import pickle
from dataclasses import dataclass, field
from typing import Iterable

import numpy as np


def tokenize(text: str) -> list[str]:
    """Whitespace-lowercase tokenization (reference sparse.py:88,175)."""
    return text.lower().split()


@dataclass
class BM25Index:
    k1: float = 1.5
    b: float = 0.75
    delta: float = 1.0           # BM25+ delta
    variant: str = "okapi"       # okapi | plus

    # corpus state
    doc_ids: list[str] = field(default_factory=list)
    doc_len: np.ndarray = field(default_factory=lambda: np.zeros(0, np.float32))
    vocab: dict[str, int] = field(default_factory=dict)

    # CSR postings: for term t, docs/tfs in [indptr[t], indptr[t+1])
    indptr: np.ndarray = field(default_factory=lambda: np.zeros(1, np.int64))
    post_doc: np.ndarray = field(default_factory=lambda: np.zeros(0, np.int32))
    post_tf: np.ndarray = field(default_factory=lambda: np.zeros(0, np.float32))
    idf: np.ndarray = field(default_factory=lambda: np.zeros(0, np.float32))

    _device_arrays: dict = field(default_factory=dict, repr=False)

    # ----- build -----
    def build(self, doc_ids: list[str], texts: Iterable[str]) -> None:
        tokenized = [tokenize(t) for t in texts]
        self.build_tokenized(doc_ids, tokenized)

    def build_tokenized(self, doc_ids: list[str], tokenized: list[list[str]]) -> None:
        self.doc_ids = list(doc_ids)
        n_docs = len(tokenized)
        self.doc_len = np.array([len(t) for t in tokenized], np.float32)

        vocab: dict[str, int] = {}
        # term -> list of (doc, tf)
        per_doc_counts: list[dict[int, int]] = []
        for toks in tokenized:
            counts: dict[int, int] = {}
            for tok in toks:
                tid = vocab.setdefault(tok, len(vocab))
                counts[tid] = counts.get(tid, 0) + 1
            per_doc_counts.append(counts)
        self.vocab = vocab
        n_terms = len(vocab)

        df = np.zeros(n_terms, np.int64)
        nnz = 0
        for counts in per_doc_counts:
            for tid in counts:
                df[tid] += 1
            nnz += len(counts)

        indptr = np.zeros(n_terms + 1, np.int64)
        np.cumsum(df, out=indptr[1:])
        post_doc = np.zeros(nnz, np.int32)
        post_tf = np.zeros(nnz, np.float32)
        cursor = indptr[:-1].copy()
        for d, counts in enumerate(per_doc_counts):
            for tid, tf in counts.items():
                p = cursor[tid]
                post_doc[p] = d
                post_tf[p] = tf
                cursor[tid] += 1

        self.indptr, self.post_doc, self.post_tf = indptr, post_doc, post_tf
        self.idf = self._compute_idf(df, n_docs)
        self._device_arrays.clear()

    def _compute_idf(self, df: np.ndarray, n_docs: int) -> np.ndarray:
        dff = df.astype(np.float64)
        if self.variant == "plus":
            return np.log((n_docs + 1.0) / np.maximum(dff, 1.0)).astype(np.float32)
        return np.log((n_docs - dff + 0.5) / (dff + 0.5) + 1.0).astype(np.float32)

    # ----- incremental add (used by /embed) -----
    def add(self, doc_ids: list[str], texts: list[str]) -> None:
        """Rebuild including new docs.  Postings arrays are immutable-by-design
        so the on-device copy stays flat; ingest batches amortize the rebuild.
        After a load() the token cache is reconstructed by inverting the CSR
        postings (token order is lost — irrelevant to BM25)."""
        all_ids = self.doc_ids + list(doc_ids)
        old_tok = getattr(self, "_tokenized_cache", None)
        new_tok = [tokenize(t) for t in texts]
        if old_tok is None:
            old_tok = self._retokenize_from_postings()
        tokenized = old_tok + new_tok
        self._tokenized_cache = tokenized
        self.build_tokenized(all_ids, tokenized)

    def _retokenize_from_postings(self) -> list[list[str]]:
        out: list[list[str]] = [[] for _ in self.doc_ids]
        if not self.doc_ids or len(self.post_doc) == 0:
            return out
        inv = {tid: tok for tok, tid in self.vocab.items()}
        for t in range(len(self.vocab)):
            for p in range(int(self.indptr[t]), int(self.indptr[t + 1])):
                out[int(self.post_doc[p])].extend(
                    [inv[t]] * int(self.post_tf[p]))
        return out

    @property
    def n_docs(self) -> int:
        return len(self.doc_ids)

    @property
    def avgdl(self) -> float:
        return float(self.doc_len.mean()) if len(self.doc_len) else 0.0

    # ----- query (CPU reference path) -----
    def query_term_ids(self, query: str) -> np.ndarray:
        toks = tokenize(query)
        ids = [self.vocab[t] for t in toks if t in self.vocab]
        return np.array(ids, np.int64)

    def get_scores(self, query: str) -> np.ndarray:
        """Dense score vector over all docs (CPU).  Mirrors the math the HIP
        kernel computes; used as the numerics reference in tests."""
        scores = np.zeros(self.n_docs, np.float32)
        if self.n_docs == 0:
            return scores
        avgdl = max(self.avgdl, 1e-9)
        norm_den = self.k1 * (1.0 - self.b + self.b * self.doc_len / avgdl)
        for tid in self.query_term_ids(query):
            lo, hi = self.indptr[tid], self.indptr[tid + 1]
            docs = self.post_doc[lo:hi]
            tf = self.post_tf[lo:hi]
            contrib = tf * (self.k1 + 1.0) / (tf + norm_den[docs])
            if self.variant == "plus":
                contrib = contrib + self.delta
            scores[docs] += self.idf[tid] * contrib
        return scores

    def search(self, query: str, top_k: int, device: str = "cpu") -> list[tuple[str, float]]:
        """Top-k (doc_id, score), score>0 only (reference sparse.py:183)."""
        if device != "cpu":
            return self._search_device(query, top_k, device)
        scores = self.get_scores(query)
        if self.n_docs == 0:
            return []
        k = min(top_k, self.n_docs)
        idx = np.argpartition(-scores, k - 1)[:k]
        idx = idx[np.argsort(-scores[idx], kind="stable")]
        return [(self.doc_ids[i], float(scores[i])) for i in idx if scores[i] > 0.0]

    # ----- device path -----
    def to_device(self, device: str) -> None:
        import torch

        self._device_arrays = {
            "indptr": torch.from_numpy(self.indptr).to(device),
            "post_doc": torch.from_numpy(self.post_doc).to(device),
            "post_tf": torch.from_numpy(self.post_tf).to(device),
            "idf": torch.from_numpy(self.idf).to(device),
            "doc_len": torch.from_numpy(self.doc_len).to(device),
        }

    def _search_device(self, query: str, top_k: int, device: str) -> list[tuple[str, float]]:
        import torch

        from sentio_amd import ops

        if not self._device_arrays:
            self.to_device(device)
        tids = torch.from_numpy(self.query_term_ids(query)).to(device)
        a = self._device_arrays
        scores = ops.bm25_score(
            tids, a["indptr"], a["post_doc"], a["post_tf"], a["idf"], a["doc_len"],
            n_docs=self.n_docs, k1=self.k1, b=self.b, avgdl=max(self.avgdl, 1e-9),
            plus_delta=self.delta if self.variant == "plus" else 0.0,
        )
        k = min(top_k, self.n_docs)
        vals, idx = torch.topk(scores, k)
        vals = vals.cpu().tolist()
        idx = idx.cpu().tolist()
        return [(self.doc_ids[i], float(v)) for i, v in zip(idx, vals) if v > 0.0]

    def search_rows(self, query: str, top_k: int, device: str = "cpu"):
        """Tensor-only top-k: (scores [k] f32, rows [k] i64) on `device`,
        padded with (-inf, -1); zero/negative scores are padded out
        (reference sparse.py:183 keeps score>0 only).  No host sync on the
        GPU path — the sharded merge all-gathers these tensors directly."""
        import torch

        if self.n_docs == 0:
            return (torch.full((top_k,), float("-inf"), device=device),
                    torch.full((top_k,), -1, dtype=torch.int64, device=device))
        if device != "cpu":
            from sentio_amd import ops

            if not self._device_arrays:
                self.to_device(device)
            tids = torch.from_numpy(self.query_term_ids(query)).to(device)
            a = self._device_arrays
            scores = ops.bm25_score(
                tids, a["indptr"], a["post_doc"], a["post_tf"], a["idf"],
                a["doc_len"], n_docs=self.n_docs, k1=self.k1, b=self.b,
                avgdl=max(self.avgdl, 1e-9),
                plus_delta=self.delta if self.variant == "plus" else 0.0)
        else:
            scores = torch.from_numpy(self.get_scores(query))
        k = min(top_k, self.n_docs)
        vals, idx = torch.topk(scores, k)
        idx = idx.long()
        dead = vals <= 0.0
        vals = vals.float().masked_fill(dead, float("-inf"))
        idx = idx.masked_fill(dead, -1)
        if k < top_k:
            vals = torch.cat([vals, torch.full((top_k - k,), float("-inf"),
                                               device=vals.device)])
            idx = torch.cat([idx, torch.full((top_k - k,), -1,
                                             dtype=torch.int64,
                                             device=idx.device)])
        return vals, idx

    # ----- persistence (reference sparse.py:102-157) -----
    def save(self, path: str) -> None:
        state = {k: getattr(self, k) for k in (
            "k1", "b", "delta", "variant", "doc_ids", "doc_len", "vocab",
            "indptr", "post_doc", "post_tf", "idf")}
        with open(path, "wb") as f:
            pickle.dump(state, f)

    @classmethod
    def load(cls, path: str) -> "BM25Index":
        with open(path, "rb") as f:
            state = pickle.load(f)
        idx = cls()
        for k, v in state.items():
            setattr(idx, k, v)
        return idx
