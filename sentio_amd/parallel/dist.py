"""Distributed bootstrap: one process per GPU, RCCL over xGMI.

The reference had no distributed layer at all (SURVEY §2.4 — its
"communication backend" was HTTPS).  Here: torch.distributed with backend
"nccl" (which IS RCCL on ROCm) for GPU runs, "gloo" for CPU tests.  Rank and
world size come from the torchrun env (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None,
                     timeout_s: float = 600.0) -> tuple[int, int]:
    """Initialize from env; returns (rank, world_size).  No-op when launched
    as a single process."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))
    return rank, world


def is_distributed() -> bool:
    return dist.is_initialized() and dist.get_world_size() > 1


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def _account(op: str, nbytes: int) -> None:
    """RCCL traffic counters (surfaced as rccl_bytes_total in /metrics —
    the GPU-era analogue of the reference's HTTP client metrics)."""
    try:
        from sentio_amd.observability.metrics import metrics_collector

        metrics_collector.inc("rccl_bytes_total", float(nbytes), op=op)
    except Exception:
        pass


def all_gather_tensor(t: torch.Tensor) -> torch.Tensor:
    """Concatenate equal-shaped tensors from all ranks along dim 0."""
    if not is_distributed():
        return t
    world = get_world_size()
    out = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(out, t.contiguous())
    _account("all_gather", t.nelement() * t.element_size() * world)
    return torch.cat(out, dim=0)


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    if is_distributed():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        _account("all_reduce", t.nelement() * t.element_size())
    return t


def broadcast_object(obj, src: int = 0):
    if not is_distributed():
        return obj
    holder = [obj]
    dist.broadcast_object_list(holder, src=src)
    return holder[0]


def all_gather_objects(obj) -> list:
    if not is_distributed():
        return [obj]
    out = [None] * get_world_size()
    dist.all_gather_object(out, obj)
    return out
