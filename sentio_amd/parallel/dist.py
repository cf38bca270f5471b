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
    """Pickled object all-gather — ONLY for cold paths (init, tests).  The
    query hot path uses tensor collectives (all_gather_tensor) and
    exchange_bytes; see VERDICT r1 item 3."""
    if not is_distributed():
        return [obj]
    out = [None] * get_world_size()
    dist.all_gather_object(out, obj)
    return out


def collective_device() -> str:
    """Device the active backend can reduce on: NCCL(=RCCL) moves GPU
    tensors over xGMI; gloo wants CPU tensors."""
    if dist.is_initialized() and dist.get_backend() == "nccl":
        return f"cuda:{torch.cuda.current_device()}"
    return "cpu"


def exchange_bytes(to_send: dict[int, bytes]) -> dict[int, bytes]:
    """Targeted pairwise byte exchange: rank r receives exactly the payloads
    other ranks addressed to it (point-to-point send/recv — xGMI is p2p, so
    this is the native shape; replaces the O(W^2) object broadcast).
    Collective: EVERY rank must call it (with {} if it has nothing to send).
    Returns {src_rank: payload}."""
    if not is_distributed():
        return {}
    world, rank = get_world_size(), get_rank()
    dev = collective_device()
    # size matrix via one tiny all-gather: sizes[r][p] = bytes r sends to p
    sizes = torch.zeros(world, dtype=torch.int64, device=dev)
    for p, b in to_send.items():
        if p != rank:
            sizes[p] = len(b)
    all_sizes = [torch.zeros_like(sizes) for _ in range(world)]
    dist.all_gather(all_sizes, sizes)
    reqs = []
    recv_bufs: dict[int, torch.Tensor] = {}
    for src in range(world):
        if src == rank:
            continue
        n = int(all_sizes[src][rank])
        if n > 0:
            buf = torch.empty(n, dtype=torch.uint8, device=dev)
            recv_bufs[src] = buf
            reqs.append(dist.irecv(buf, src=src))
    total = 0
    for dst, b in to_send.items():
        if dst == rank or not b:
            continue
        t = torch.frombuffer(bytearray(b), dtype=torch.uint8).to(dev)
        reqs.append(dist.isend(t, dst=dst))
        total += len(b)
    for r in reqs:
        r.wait()
    _account("p2p_exchange", total)
    return {src: bytes(buf.cpu().numpy().tobytes())
            for src, buf in recv_bufs.items()}
