"""Performance / resource monitoring
(reference src/observability/monitoring.py:38-341: deque history, alert
thresholds + callbacks, psutil CPU/mem summaries, health checks)."""

from __future__ import annotations

import time
from collections import deque
from typing import Any, Callable

try:
    import psutil

    _HAS_PSUTIL = True
except ImportError:  # pragma: no cover
    _HAS_PSUTIL = False


class PerformanceMonitor:
    def __init__(self, history: int = 1000):
        self._values: dict[str, deque] = {}
        self._history = history
        self._thresholds: dict[str, float] = {}
        self._callbacks: list[Callable[[str, float], None]] = []

    def record_value(self, name: str, value: float) -> None:
        dq = self._values.setdefault(name, deque(maxlen=self._history))
        dq.append((time.time(), value))
        thr = self._thresholds.get(name)
        if thr is not None and value > thr:
            for cb in self._callbacks:
                try:
                    cb(name, value)
                except Exception:
                    pass

    def set_threshold(self, name: str, value: float) -> None:
        self._thresholds[name] = value

    def on_alert(self, cb: Callable[[str, float], None]) -> None:
        self._callbacks.append(cb)

    def summary(self, name: str) -> dict[str, Any] | None:
        dq = self._values.get(name)
        if not dq:
            return None
        vals = sorted(v for _, v in dq)
        return {
            "count": len(vals),
            "min": vals[0],
            "max": vals[-1],
            "mean": sum(vals) / len(vals),
            "p50": vals[len(vals) // 2],
            "p95": vals[min(len(vals) - 1, int(len(vals) * 0.95))],
        }

    def all_summaries(self) -> dict[str, Any]:
        return {name: self.summary(name) for name in self._values}


class ResourceMonitor:
    def snapshot(self) -> dict[str, Any]:
        out: dict[str, Any] = {"timestamp": time.time()}
        if _HAS_PSUTIL:
            proc = psutil.Process()
            out["cpu_percent"] = psutil.cpu_percent(interval=None)
            out["memory_percent"] = psutil.virtual_memory().percent
            out["process_rss_mb"] = proc.memory_info().rss / 1e6
        try:
            import torch

            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                out["gpu_hbm_used_gb"] = (total - free) / 1e9
                out["gpu_hbm_total_gb"] = total / 1e9
                out["gpu_count"] = torch.cuda.device_count()
        except Exception:
            pass
        return out

    def health(self) -> dict[str, Any]:
        snap = self.snapshot()
        alerts = []
        if snap.get("cpu_percent", 0) > 95:
            alerts.append("cpu_high")
        if snap.get("memory_percent", 0) > 90:
            alerts.append("memory_high")
        return {"status": "degraded" if alerts else "healthy",
                "alerts": alerts, "resources": snap}


performance_monitor = PerformanceMonitor()
resource_monitor = ResourceMonitor()
