"""Tensor parallelism for the generator.

Megatron-style decomposition over the node's GPUs:
* attention: QKV projection column-parallel (head-sharded), output
  projection row-parallel → one all-reduce per layer;
* FFN: gate+up column-parallel, down row-parallel → one all-reduce.
xGMI is point-to-point (7 links x ~153 GB/s), so the per-layer all-reduce
of [B·S, dim] bf16 activations is the bandwidth bound RCCL must schedule;
bucket sizes stay whole-tensor (decode activations are tiny).

The Transformer engine consumes this through `TPContext`, which tells the
weight container to materialize only this rank's shard (sliced from the
same seeded generator so TP=N matches TP=1 bit-for-bit in fp32).

Comm/compute overlap (SURVEY §7 hard part 2): every row-parallel output
projection goes through `linear_row_parallel`, which splits the GEMM along
its OUTPUT dim into chunks and launches each chunk's partial-sum
all-reduce with async_op=True as soon as that chunk's GEMM is enqueued —
RCCL runs the collective on its own stream, so chunk i's reduction rides
under chunk i+1's GEMM (and under the next kernel for the last chunk until
the wait).  xGMI is point-to-point (7 links x ~153 GB/s), so the prefill
reduces are per-link bandwidth-bound and chunking shortens the exposed
tail; decode reduces are latency-bound and the overlap window is the
launch gap.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class TPContext:
    rank: int = 0
    world: int = 1
    group: object = None

    @property
    def enabled(self) -> bool:
        return self.world > 1

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.enabled:
            dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.group)
        return t

    def all_reduce_async(self, t: torch.Tensor):
        """Launch the partial-sum reduce without blocking the compute
        stream; returns the Work handle (None when TP is off)."""
        if not self.enabled:
            return None
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.group,
                               async_op=True)

    @classmethod
    def from_env(cls) -> "TPContext":
        if dist.is_initialized() and dist.get_world_size() > 1:
            return cls(rank=dist.get_rank(), world=dist.get_world_size())
        return cls()


def linear_row_parallel(x: torch.Tensor, w: torch.Tensor, tp: TPContext,
                        chunks: int = 2, linear=None) -> torch.Tensor:
    """y = x @ W_shard^T summed over TP ranks, with the all-reduce
    OVERLAPPED: W (stored [out, in], TN layout) splits along `out` into
    `chunks`; chunk i's async all-reduce is in flight while chunk i+1's
    GEMM runs.  `linear` overrides the GEMM (e.g. ops.lt_linear for
    shape-stable decode rows).  TP off → one plain GEMM."""
    lin = linear or torch.nn.functional.linear
    if not tp.enabled:
        return lin(x, w)
    n = w.shape[0]
    if chunks <= 1 or n < 2 * chunks:
        out = lin(x, w)
        tp.all_reduce(out)
        return out
    step = (n + chunks - 1) // chunks
    outs, works = [], []
    for i in range(chunks):
        o = lin(x, w[i * step: (i + 1) * step])
        works.append(tp.all_reduce_async(o))   # rides under the next GEMM
        outs.append(o)
    for wk in works:
        if wk is not None:
            wk.wait()   # stream-ordered on NCCL/RCCL (no host sync)
    return torch.cat(outs, dim=-1)


def shard_columns(w: torch.Tensor, tp: TPContext) -> torch.Tensor:
    """Column-parallel shard of [in, out]: split the out dim."""
    if not tp.enabled:
        return w
    out = w.shape[1]
    assert out % tp.world == 0, "out dim must divide TP degree"
    step = out // tp.world
    return w[:, tp.rank * step : (tp.rank + 1) * step].contiguous()


def shard_rows(w: torch.Tensor, tp: TPContext) -> torch.Tensor:
    """Row-parallel shard of [in, out]: split the in dim."""
    if not tp.enabled:
        return w
    inp = w.shape[0]
    assert inp % tp.world == 0, "in dim must divide TP degree"
    step = inp // tp.world
    return w[tp.rank * step : (tp.rank + 1) * step].contiguous()


def shard_qkv(w: torch.Tensor, n_heads: int, n_kv: int, head_dim: int,
              tp: TPContext) -> torch.Tensor:
    """Shard the fused QKV [dim, (H + 2*Hkv)*hd] by heads: this rank keeps
    H/world query heads and Hkv/world kv heads (Hkv % world == 0)."""
    if not tp.enabled:
        return w
    assert n_heads % tp.world == 0 and n_kv % tp.world == 0
    hl = n_heads // tp.world
    kl = n_kv // tp.world
    q_end = n_heads * head_dim
    k_end = q_end + n_kv * head_dim
    q = w[:, tp.rank * hl * head_dim : (tp.rank + 1) * hl * head_dim]
    k = w[:, q_end + tp.rank * kl * head_dim : q_end + (tp.rank + 1) * kl * head_dim]
    v = w[:, k_end + tp.rank * kl * head_dim : k_end + (tp.rank + 1) * kl * head_dim]
    return torch.cat([q, k, v], dim=1).contiguous()
