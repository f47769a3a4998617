"""Metrics registry: counters, gauges, histograms.

(reference: pkg/metrics/metrics.go — VictoriaMetrics push catalog:
scheduler durations/retries/backlog depth, container_start_latency_ms,
worker_startup_phase_duration_ms{phase}, proxy dial/wait). This build keeps
an in-process registry exposed over REST (/api/v1/metrics) and in Prometheus
text format (/metrics) for scrapes; no external TSDB required on a single
node.
"""

from __future__ import annotations

import math
import threading
import time
from typing import Dict, List, Optional, Tuple

_BUCKETS = [1, 2.5, 5, 10, 25, 50, 100, 250, 500, 1000, 2500, 5000, 10000, 30000]


class Histogram:
    def __init__(self, buckets: Optional[List[float]] = None):
        self.buckets = buckets or _BUCKETS
        self.counts = [0] * (len(self.buckets) + 1)
        self.total = 0.0
        self.n = 0

    def observe(self, value: float) -> None:
        self.n += 1
        self.total += value
        for i, b in enumerate(self.buckets):
            if value <= b:
                self.counts[i] += 1
                return
        self.counts[-1] += 1

    def percentile(self, q: float) -> float:
        if self.n == 0:
            return 0.0
        target = q * self.n
        acc = 0
        for i, c in enumerate(self.counts):
            acc += c
            if acc >= target:
                return self.buckets[i] if i < len(self.buckets) else self.buckets[-1]
        return self.buckets[-1]

    def snapshot(self) -> Dict:
        return {
            "count": self.n,
            "sum": round(self.total, 3),
            "avg": round(self.total / self.n, 3) if self.n else 0.0,
            "p50": self.percentile(0.5),
            "p95": self.percentile(0.95),
            "p99": self.percentile(0.99),
        }


class MetricsRegistry:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._counters: Dict[str, float] = {}
        self._gauges: Dict[str, float] = {}
        self._hists: Dict[str, Histogram] = {}

    def inc(self, name: str, by: float = 1.0, **labels) -> None:
        key = _key(name, labels)
        with self._lock:
            self._counters[key] = self._counters.get(key, 0.0) + by

    def set_gauge(self, name: str, value: float, **labels) -> None:
        with self._lock:
            self._gauges[_key(name, labels)] = value

    def observe(self, name: str, value: float, **labels) -> None:
        key = _key(name, labels)
        with self._lock:
            h = self._hists.get(key)
            if h is None:
                h = self._hists[key] = Histogram()
            h.observe(value)

    def timer(self, name: str, **labels):
        return _Timer(self, name, labels)

    def snapshot(self) -> Dict:
        with self._lock:
            return {
                "counters": dict(self._counters),
                "gauges": dict(self._gauges),
                "histograms": {k: h.snapshot() for k, h in self._hists.items()},
            }

    def prometheus_text(self) -> str:
        out = []
        with self._lock:
            for k, v in sorted(self._counters.items()):
                out.append(f"{k} {v}")
            for k, v in sorted(self._gauges.items()):
                out.append(f"{k} {v}")
            for k, h in sorted(self._hists.items()):
                s = h.snapshot()
                base = k.split("{")[0]
                labels = k[len(base):]
                out.append(f"{base}_count{labels} {s['count']}")
                out.append(f"{base}_sum{labels} {s['sum']}")
        return "\n".join(out) + "\n"


class _Timer:
    def __init__(self, reg: MetricsRegistry, name: str, labels: Dict):
        self.reg = reg
        self.name = name
        self.labels = labels

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.reg.observe(self.name, (time.perf_counter() - self.t0) * 1000.0, **self.labels)


def _key(name: str, labels: Dict) -> str:
    if not labels:
        return name
    inner = ",".join(f'{k}="{v}"' for k, v in sorted(labels.items()))
    return f"{name}{{{inner}}}"


# process-global default registry
REGISTRY = MetricsRegistry()
