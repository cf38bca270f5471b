"""Corpus sharding across GPUs: per-shard top-k + RCCL all-gather merge.

Design (SURVEY §2.4 / BASELINE north star): the dense index and BM25
postings shard across the node's GPUs (288 GB HBM each).  A query batch is
SPMD: every rank embeds its own queries, the query vectors all-gather
(tiny, latency-bound), every rank scores ALL queries against its local
shard (one fused cosine-scan amortizes the HBM read over W·B queries),
per-shard top-k candidates all-gather back (k·(id,score) ≈ KBs), and each
query's owner rank merges.  Document payloads stay shard-local and are
fetched by id on demand (all-to-all object exchange).
"""

from __future__ import annotations

import torch

from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.models.document import Document
from sentio_amd.parallel import dist as D


class ShardedIndex:
    """Wraps the local shard (DenseIndex + BM25Index) with collective search."""

    def __init__(self, dense: DenseIndex, bm25: BM25Index | None = None,
                 device: str = "cpu"):
        self.dense = dense
        self.bm25 = bm25
        self.device = device
        self.rank = D.get_rank()
        self.world = D.get_world_size()

    # ---- dense path ----
    def search_dense(self, queries: torch.Tensor, top_k: int
                     ) -> list[list[tuple[str, float]]]:
        """queries: [B, dim] on self.device (this rank's own queries).
        Returns this rank's queries' merged global top-k as
        (doc_ref, score) where doc_ref = "shard:local_id"."""
        B = queries.shape[0]
        if self.world == 1:
            return [
                [(f"0:{doc_id}", s) for doc_id, s in hits]
                for hits in self.dense.search(queries, top_k)
            ]

        q_all = D.all_gather_tensor(queries)            # [W*B, dim]
        hits_all = self.dense.search(q_all, top_k)      # local shard, all queries

        # pack local candidates: scores [W*B, k], plus local doc ids by row
        k = max((len(h) for h in hits_all), default=0)
        k = max(k, 1)
        scores = torch.full((q_all.shape[0], k), -1e30, dtype=torch.float32,
                            device=queries.device)
        for i, hits in enumerate(hits_all):
            for j, (_id, s) in enumerate(hits):
                scores[i, j] = s
        local_ids = [[h[0] for h in hits] for hits in hits_all]

        # all-gather candidate scores (tensor) and ids (object — small)
        gathered_scores = D.all_gather_objects(scores.cpu())
        gathered_ids = D.all_gather_objects(local_ids)

        out: list[list[tuple[str, float]]] = []
        base = self.rank * B
        for qi in range(B):
            row = base + qi
            cands: list[tuple[str, float]] = []
            for shard in range(self.world):
                ids = gathered_ids[shard][row]
                sc = gathered_scores[shard][row]
                for j, doc_id in enumerate(ids):
                    cands.append((f"{shard}:{doc_id}", float(sc[j])))
            cands.sort(key=lambda x: x[1], reverse=True)
            out.append(cands[:top_k])
        return out

    # ---- sparse path ----
    def search_sparse(self, query: str, top_k: int) -> list[tuple[str, float]]:
        local = (
            self.bm25.search(query, top_k, device=self.device)
            if self.bm25 is not None else []
        )
        if self.world == 1:
            return [(f"0:{doc_id}", s) for doc_id, s in local]
        gathered = D.all_gather_objects(local)
        cands = [
            (f"{shard}:{doc_id}", float(s))
            for shard, hits in enumerate(gathered)
            for doc_id, s in hits
        ]
        cands.sort(key=lambda x: x[1], reverse=True)
        return cands[:top_k]

    # ---- payload fetch ----
    def fetch_documents(self, refs: list[str]) -> dict[str, Document]:
        """Resolve "shard:doc_id" refs to Documents.  Local refs resolve
        directly; remote refs go through an all-gather request/response."""
        local = {}
        remote_want: list[str] = []
        for ref in refs:
            shard_s, doc_id = ref.split(":", 1)
            if int(shard_s) == self.rank:
                doc = self.dense.get_document(doc_id)
                if doc is not None:
                    local[ref] = doc
            else:
                remote_want.append(ref)
        if self.world == 1 or not D.is_distributed():
            return local

        # every rank publishes its wants; every rank answers what it owns
        all_wants = D.all_gather_objects(remote_want)
        answers: dict[str, dict] = {}
        for wants in all_wants:
            for ref in wants:
                shard_s, doc_id = ref.split(":", 1)
                if int(shard_s) == self.rank:
                    doc = self.dense.get_document(doc_id)
                    if doc is not None:
                        answers[ref] = doc.to_dict()
        all_answers = D.all_gather_objects(answers)
        for ans in all_answers:
            for ref, dd in ans.items():
                if ref in remote_want:
                    local[ref] = Document.from_dict(dd)
        return local

    def total_docs(self) -> int:
        # NCCL reduces GPU tensors; gloo reduces CPU tensors
        dev = self.device if self.device != "cpu" else "cpu"
        n = torch.tensor([len(self.dense)], dtype=torch.int64, device=dev)
        if D.is_distributed():
            D.all_reduce_sum(n)
        return int(n.item())
