"""Per-container usage records (reference: pkg/worker/usage.go — duration/
cost intervals pushed to the metrics plane; billing recorder deferred)."""

from __future__ import annotations

import time
from typing import Dict, Optional

from ..common.metrics import REGISTRY


class UsageTracker:
    def __init__(self, repo=None):
        self.repo = repo
        self._open: Dict[str, Dict] = {}

    def container_started(self, container_id: str, gpu: str = "", gpu_count: int = 0, cpu: int = 0, memory_mb: int = 0, gpu_fraction: float = 1.0) -> None:
        self._open[container_id] = {
            "started_at": time.time(),
            "gpu": gpu,
            "gpu_count": gpu_count,
            # shared-device containers bill their fraction of a GPU
            "gpu_units": (gpu_count or 0) * gpu_fraction if gpu_fraction < 1.0 else gpu_count,
            "cpu": cpu,
            "memory_mb": memory_mb,
        }

    def container_stopped(self, container_id: str, exit_code: Optional[int] = None) -> Optional[Dict]:
        rec = self._open.pop(container_id, None)
        if rec is None:
            return None
        dur = time.time() - rec["started_at"]
        units = rec.get("gpu_units", rec["gpu_count"])
        record = {
            "container_id": container_id,
            "duration_s": round(dur, 3),
            "gpu_seconds": round(dur * units, 3),
            "cpu_millicore_seconds": round(dur * rec["cpu"], 1),
            "exit_code": exit_code,
            **{k: rec[k] for k in ("gpu", "gpu_count", "cpu", "memory_mb")},
        }
        REGISTRY.inc("container_seconds_total", dur, gpu=rec["gpu"] or "cpu")
        if units:
            REGISTRY.inc("gpu_seconds_total", dur * units, gpu=rec["gpu"])
        REGISTRY.observe("container_duration_s", dur)
        return record
