"""Corpus sharding across GPUs: per-shard top-k + RCCL all-gather merge.

Design (SURVEY §2.4 / BASELINE north star): the dense index and BM25
postings shard across the node's GPUs (288 GB HBM each).  A query batch is
SPMD: every rank embeds its own queries, the query vectors all-gather
(tiny, latency-bound), every rank scores ALL queries against its local
shard (one fused cosine-scan amortizes the HBM read over W·B queries),
per-shard top-k candidates all-gather back (k·(score,row) ≈ KBs), and each
query's owner rank merges — ALL of it as padded fp32/int64 TENSOR
collectives: no pickled objects anywhere on the query hot path (the r1
all_gather_object merge was a CPU/pickle round trip in a latency-bound
xGMI gather).  Document payloads stay shard-local; refs are
"{shard}:{d|s}{row}" row handles resolved on demand by the owning shard
through a targeted point-to-point byte exchange (xGMI is p2p — no O(W²)
broadcast of every rank's wants).
"""

from __future__ import annotations

import json

import torch

from sentio_amd.index.bm25 import BM25Index
from sentio_amd.index.dense import DenseIndex
from sentio_amd.models.document import Document
from sentio_amd.parallel import dist as D


def _code_to_int(code: str) -> int:
    # "d123" -> 246, "s123" -> 247 (kind in the low bit)
    return int(code[1:]) * 2 + (1 if code[0] == "s" else 0)


def _int_to_code(v: int) -> str:
    return ("s" if v & 1 else "d") + str(v >> 1)


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
        (doc_ref, score) with doc_ref = "shard:d<row>" (resolve payloads
        via fetch_documents)."""
        B = queries.shape[0]
        W = self.world
        if W == 1:
            vals, rows = self.dense.search_rows(queries, top_k)
            vl, rl = vals.cpu().tolist(), rows.cpu().tolist()
            return [[(f"0:d{r}", float(v)) for v, r in zip(vl[i], rl[i])
                     if r >= 0] for i in range(B)]

        q_all = D.all_gather_tensor(queries)            # [W*B, dim]
        # local shard scores ALL queries; merge stays tensor-resident
        vals, rows = self.dense.search_rows(q_all, top_k)   # [W*B, k]
        vals_all = D.all_gather_tensor(vals).view(W, W * B, top_k)
        rows_all = D.all_gather_tensor(rows).view(W, W * B, top_k)
        base = self.rank * B
        v = vals_all[:, base: base + B].permute(1, 0, 2).reshape(B, W * top_k)
        r = rows_all[:, base: base + B].permute(1, 0, 2).reshape(B, W * top_k)
        topv, topi = v.topk(min(top_k, W * top_k), dim=1)
        shard_of = topi // top_k
        row_of = r.gather(1, topi)
        tv, ts, tr = (topv.cpu().tolist(), shard_of.cpu().tolist(),
                      row_of.cpu().tolist())
        return [[(f"{s}:d{rr}", float(vv))
                 for vv, s, rr in zip(tv[qi], ts[qi], tr[qi]) if rr >= 0]
                for qi in range(B)]

    # ---- sparse path ----
    def search_sparse(self, query: str, top_k: int) -> list[tuple[str, float]]:
        dev = self.device
        if self.bm25 is not None:
            vals, rows = self.bm25.search_rows(query, top_k, device=dev)
        else:
            vals = torch.full((top_k,), float("-inf"), device=dev)
            rows = torch.full((top_k,), -1, dtype=torch.int64, device=dev)
        if self.world == 1:
            vl, rl = vals.cpu().tolist(), rows.cpu().tolist()
            return [(f"0:s{r}", float(v)) for v, r in zip(vl, rl) if r >= 0]
        W = self.world
        vals_all = D.all_gather_tensor(vals.view(-1))   # [W*k]
        rows_all = D.all_gather_tensor(rows.view(-1))
        topv, topi = vals_all.topk(min(top_k, W * top_k))
        shard_of = topi // top_k
        row_of = rows_all.gather(0, topi)
        tv, ts, tr = (topv.cpu().tolist(), shard_of.cpu().tolist(),
                      row_of.cpu().tolist())
        return [(f"{s}:s{r}", float(v))
                for v, s, r in zip(tv, ts, tr) if r >= 0]

    # ---- payload resolution ----
    def _resolve_local(self, code: str) -> Document | None:
        row = int(code[1:])
        if code[0] == "d":
            return self.dense.get_document_by_row(row)
        if self.bm25 is not None and 0 <= row < len(self.bm25.doc_ids):
            doc_id = self.bm25.doc_ids[row]
            doc = self.dense.get_document(doc_id)
            if doc is not None:
                return doc
            return Document(text="", metadata={}, id=doc_id)
        return None

    def fetch_documents(self, refs: list[str]) -> dict[str, Document]:
        """Resolve "shard:{d|s}row" refs to Documents.  Local refs resolve
        directly; remote refs go through a tiny int64 wants all-gather plus
        a TARGETED p2p byte exchange of the payloads (owner → requester
        only).  Collective: every rank must call it each search step."""
        local: dict[str, Document] = {}
        remote: list[tuple[int, int]] = []          # (owner, coded row)
        for ref in refs:
            shard_s, code = ref.split(":", 1)
            owner = int(shard_s)
            if owner == self.rank:
                doc = self._resolve_local(code)
                if doc is not None:
                    local[ref] = doc
            else:
                remote.append((owner, _code_to_int(code)))
        if self.world == 1 or not D.is_distributed():
            return local

        dev = D.collective_device()
        n = torch.tensor([len(remote)], dtype=torch.int64, device=dev)
        counts = D.all_gather_tensor(n)             # [W]
        mx = max(int(counts.max().item()), 1)
        buf = torch.full((mx, 2), -1, dtype=torch.int64)
        if remote:
            buf[: len(remote)] = torch.tensor(remote, dtype=torch.int64)
        wants_all = D.all_gather_tensor(buf.to(dev)).view(self.world, mx, 2)
        wants_all = wants_all.cpu()
        counts = counts.cpu()

        # answer only what this rank owns, addressed per requester
        to_send: dict[int, bytes] = {}
        for req in range(self.world):
            if req == self.rank:
                continue
            items = []
            for owner, coded in wants_all[req][: int(counts[req])].tolist():
                if owner == self.rank:
                    doc = self._resolve_local(_int_to_code(coded))
                    if doc is not None:
                        items.append((coded, doc.to_dict()))
            if items:
                to_send[req] = json.dumps(items).encode()
        answers = D.exchange_bytes(to_send)
        for src, blob in answers.items():
            for coded, dd in json.loads(blob.decode()):
                local[f"{src}:{_int_to_code(coded)}"] = Document.from_dict(dd)
        return local

    def total_docs(self) -> int:
        # NCCL reduces GPU tensors; gloo reduces CPU tensors
        dev = self.device if self.device != "cpu" else "cpu"
        n = torch.tensor([len(self.dense)], dtype=torch.int64, device=dev)
        if D.is_distributed():
            D.all_reduce_sum(n)
        return int(n.item())
