"""hipGraph-captured fixed-shape forward sessions for the encoder and
reranker engines.

The encoder / reranker batches are small (a few thousand activation rows
through 12-24 layers), so their wall time on MI355X is dominated by
per-kernel launch gaps, not kernel time.  Padding every batch to a fixed
(B, S) bucket makes the whole forward shape-static, so it can be captured
once as a hipGraph and replayed per call (same trick the generator uses for
its decode step).  Dynamic CONTENT (token ids, valid lengths) flows through
static input buffers; shapes are the bucket's.

No reference counterpart (the reference called remote HTTP APIs —
reference src/core/embeddings/providers/jina.py:165)."""

from __future__ import annotations

import os
import threading
from typing import Callable

import torch


def graphs_enabled(device: str) -> bool:
    return (device != "cpu"
            and os.environ.get("SENTIO_DISABLE_HIPGRAPH", "0") != "1"
            and torch.cuda.is_available())


def bucket_batch(b: int, buckets=(1, 2, 4, 8, 16, 32, 64)) -> int:
    for cap in buckets:
        if b <= cap:
            return cap
    return b  # beyond the largest bucket: run unbucketed


class GraphedForward:
    """One captured (B, S) forward.  `fn(tokens, kv_lens) -> out` must be
    shape-static for fixed inputs and must not allocate new parameters."""

    def __init__(self, fn: Callable, batch: int, seq: int, device: str):
        self.fn = fn
        self.device = device
        self._lock = threading.Lock()   # static buffers: one replay at a time
        self.tokens = torch.zeros(batch, seq, dtype=torch.int64, device=device)
        self.kv_lens = torch.ones(batch, dtype=torch.int32, device=device)
        self.graph: torch.cuda.CUDAGraph | None = None
        self.out: torch.Tensor | None = None

    def _capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):                   # warmup (required pre-capture)
                self.fn(self.tokens, self.kv_lens)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        # relaxed: see generator._capture — cross-thread engine work
        # (decode on the main thread) must not invalidate this capture
        with torch.cuda.graph(self.graph, capture_error_mode="relaxed"):
            self.out = self.fn(self.tokens, self.kv_lens)

    def run(self, tokens: torch.Tensor, kv_lens: torch.Tensor) -> torch.Tensor:
      with self._lock:
        if self.graph is None:
            self._capture()
        b = tokens.shape[0]
        self.tokens[:b, : tokens.shape[1]].copy_(tokens)
        if tokens.shape[1] < self.tokens.shape[1]:
            self.tokens[:b, tokens.shape[1]:].zero_()
        if b < self.tokens.shape[0]:
            self.tokens[b:].zero_()
            self.kv_lens[b:].fill_(1)            # keep padded rows valid
        self.kv_lens[:b].copy_(kv_lens)
        self.graph.replay()
        # clone: the caller may hold the result across the next replay
        return self.out[:b].clone()


class GraphedEnginePool:
    """Per-(bucketed B, S) GraphedForward cache for an engine."""

    def __init__(self, fn: Callable, seq: int, device: str):
        self.fn = fn
        self.seq = seq
        self.device = device
        self._pool: dict[int, GraphedForward] = {}

    @property
    def active(self) -> bool:
        return graphs_enabled(self.device)

    def run(self, tokens: torch.Tensor, kv_lens: torch.Tensor) -> torch.Tensor:
        bb = bucket_batch(tokens.shape[0])
        if bb not in self._pool:
            self._pool[bb] = GraphedForward(self.fn, bb, self.seq, self.device)
        return self._pool[bb].run(tokens, kv_lens)
