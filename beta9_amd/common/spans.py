"""Container-lifecycle span instrumentation.

The de-facto profiler of the cold-start path (reference: pkg/types/
event.go:548-700 span schema + worker_startup_phase_duration_ms metrics,
pkg/metrics/metrics.go:296-309): every phase of RunContainer emits a span;
benches and the REST API read them back to decompose cold-start latency,
including the MI355X-specific phases (gpu_assignment, hip_preinit,
weight_stream, hbm_restore).
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Any, Callable, Dict, List, Optional

SpanSink = Callable[[Dict[str, Any]], None]


class SpanRecorder:
    def __init__(self, container_id: str = "", sink: Optional[SpanSink] = None):
        self.container_id = container_id
        self.spans: List[Dict[str, Any]] = []
        self._sink = sink

    @contextmanager
    def span(self, name: str, parent: str = "", **meta: Any):
        start = time.time()
        try:
            yield
        finally:
            end = time.time()
            rec = {
                "name": name,
                "container_id": self.container_id,
                "start": start,
                "end": end,
                "duration_ms": (end - start) * 1000.0,
                "parent": parent,
                "meta": meta,
            }
            self.spans.append(rec)
            if self._sink is not None:
                try:
                    self._sink(rec)
                except Exception:
                    pass

    def record(self, name: str, start: float, end: float, **meta: Any) -> None:
        rec = {
            "name": name,
            "container_id": self.container_id,
            "start": start,
            "end": end,
            "duration_ms": (end - start) * 1000.0,
            "parent": "",
            "meta": meta,
        }
        self.spans.append(rec)
        if self._sink is not None:
            try:
                self._sink(rec)
            except Exception:
                pass

    def phase_durations_ms(self) -> Dict[str, float]:
        return {s["name"]: s["duration_ms"] for s in self.spans}
