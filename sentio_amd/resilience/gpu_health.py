"""GPU device health probe + per-rank heartbeat.

SURVEY §5 (failure detection): the reference's health surface is HTTP pings
to its remote providers (reference src/api/handlers/health.py:214-249);
the MI355X equivalent is a local HIP probe — run a tiny op on the device,
synchronize, and treat any HIP runtime error (ECC, hung queue, lost
context) as unhealthy — plus a per-rank heartbeat so a stuck rank is
detectable from its last-progress timestamp.

The heartbeat deliberately does NOT issue collectives from the background
loop: a collective racing the serving path's RCCL traffic can deadlock the
communicator.  Cross-rank liveness exchange (`gather_heartbeats`) is an
explicit call for the caller's own cadence (e.g. bench teardown, an admin
endpoint), not something the loop does behind your back.
"""

from __future__ import annotations

import time
from threading import Lock

import torch

from sentio_amd.parallel import dist as D


def gpu_health_check(device: str = "cuda:0") -> bool:
    """One tiny round-trip op on `device`.  Returns False on any HIP error
    (surfaces as RuntimeError) instead of raising — a failed probe is a
    health signal, not a crash."""
    try:
        if device != "cpu" and not torch.cuda.is_available():
            return False
        x = torch.ones(8, device=device)
        s = float((x * 2).sum().item())
        return abs(s - 16.0) < 1e-6
    except RuntimeError:
        return False


class RankHeartbeat:
    """Monotonic per-rank progress marker.

    Hot-path code calls `beat()` after completed work units (a decode step,
    a served request); `age_s()` answers "how long since this rank last
    made progress".  Thread-safe, allocation-free on the beat path."""

    def __init__(self) -> None:
        self._lock = Lock()
        self._last = time.monotonic()
        self._count = 0
        self.rank = D.get_rank()

    def beat(self) -> None:
        with self._lock:
            self._last = time.monotonic()
            self._count += 1

    def age_s(self) -> float:
        with self._lock:
            return time.monotonic() - self._last

    @property
    def count(self) -> int:
        with self._lock:
            return self._count

    def healthy(self, max_age_s: float = 120.0) -> bool:
        return self.age_s() <= max_age_s

    def snapshot(self) -> dict:
        with self._lock:
            return {"rank": self.rank, "beats": self._count,
                    "age_s": round(time.monotonic() - self._last, 3)}

    def gather_heartbeats(self) -> list[dict]:
        """All ranks' snapshots (ALL ranks must call — it is a collective).
        Single-process: just this rank's."""
        snap = self.snapshot()
        if not D.is_distributed():
            return [snap]
        return D.all_gather_objects(snap)


def register_gpu_health(checker, device: str = "cuda:0",
                        heartbeat: RankHeartbeat | None = None,
                        max_age_s: float = 120.0) -> None:
    """Wire the device probe (and optionally a heartbeat-staleness check)
    into a resilience HealthChecker."""
    checker.register("gpu_device", lambda: gpu_health_check(device))
    if heartbeat is not None:
        checker.register("rank_heartbeat",
                         lambda: heartbeat.healthy(max_age_s))
