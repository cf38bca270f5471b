"""Retrieval at production scale on one MI355X: 10M-doc dense scan + GPU
BM25 (SURVEY hard part #3 — fused top-k over tens of millions of rows at
interactive latency; the full bench uses 1.25M docs/GPU so this probe
shows headroom)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from sentio_amd import ops


def main():
    if not torch.cuda.is_available():
        print(f"{__file__}: needs a GPU — skipping")
        return
    dev = "cuda:0"
    N, D, B = 10_000_000, 1024, 32
    print(f"dense index: {N} x {D} fp16 = {N*D*2/1e9:.1f} GB HBM")
    mat = torch.empty(N, D, dtype=torch.float16, device=dev)
    for i0 in range(0, N, 1_000_000):
        blk = torch.randn(1_000_000, D, device=dev)
        mat[i0:i0+1_000_000] = (blk / blk.norm(dim=1, keepdim=True)).half()
    q = torch.randn(B, D, device=dev)
    q = (q / q.norm(dim=1, keepdim=True)).half()
    for _ in range(3):
        ops.cosine_topk(q, mat, 10)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        vals, idx = ops.cosine_topk(q, mat, 10)
    torch.cuda.synchronize()
    t = (time.perf_counter() - t0) / iters
    print(f"cosine top-k: batch {B} over {N/1e6:.0f}M docs in {t*1e3:.1f} ms "
          f"({N*D*2/t/1e12:.2f} TB/s scan)")
    del mat
    torch.cuda.empty_cache()

    # BM25: 10M docs x 20 postings
    vocab, per_doc = 100_000, 20
    nnz = N * per_doc
    print(f"bm25 postings: {nnz/1e6:.0f}M entries "
          f"({(nnz*(4+4))/1e9:.1f} GB device)")
    rng = np.random.RandomState(0)
    terms = torch.from_numpy((rng.zipf(1.3, size=nnz) - 1) % vocab)
    order = torch.argsort(terms)
    post_doc = torch.repeat_interleave(
        torch.arange(N, dtype=torch.int32), per_doc)[order].to(dev)
    post_tf = torch.randint(1, 5, (nnz,), dtype=torch.float32)[order].to(dev)
    counts = torch.bincount(terms, minlength=vocab)
    indptr = torch.zeros(vocab + 1, dtype=torch.int64)
    indptr[1:] = torch.cumsum(counts, 0)
    indptr = indptr.to(dev)
    doc_len = torch.randint(40, 200, (N,), dtype=torch.float32, device=dev)
    idf = torch.log((N - counts.float() + 0.5) / (counts.float() + 0.5) + 1.0).to(dev)
    tids = torch.randint(0, vocab, (4,), dtype=torch.int64, device=dev)
    for _ in range(3):
        ops.bm25_score(tids, indptr, post_doc, post_tf, idf, doc_len,
                       n_docs=N, k1=1.5, b=0.75, avgdl=120.0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        s = ops.bm25_score(tids, indptr, post_doc, post_tf, idf, doc_len,
                           n_docs=N, k1=1.5, b=0.75, avgdl=120.0)
        torch.topk(s, 10)
    torch.cuda.synchronize()
    print(f"bm25 score+topk over {N/1e6:.0f}M docs: "
          f"{(time.perf_counter()-t0)/10*1e3:.1f} ms per query")


if __name__ == "__main__":
    main()
