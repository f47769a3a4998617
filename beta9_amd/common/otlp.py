"""OTLP-JSON span export for the lifecycle span system.

The reference exports OTel traces via OTLP-HTTP to Tempo (reference:
pkg/common/trace.go:58-190). This environment has no OTel SDK
and no collector, so spans are serialized in the OTLP/JSON wire shape
(`ExportTraceServiceRequest`) to a JSONL file — any OTLP collector ingests
the lines verbatim (`curl -d @line collector:4318/v1/traces`), and the
format keeps us wire-compatible when a collector exists.
"""

from __future__ import annotations

import json
import os
import secrets
from typing import Any, Dict, Optional


def _hex_id(nbytes: int) -> str:
    return secrets.token_hex(nbytes)


class OtlpSpanExporter:
    """Converts lifecycle span dicts ({name, start, end, container_id,
    ...meta}) to OTLP resourceSpans JSON lines. One trace per container."""

    def __init__(self, path: str, service_name: str = "beta9-amd"):
        self.path = path
        self.service_name = service_name
        self._traces: Dict[str, str] = {}  # container_id -> trace_id
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        self._f = open(path, "a", buffering=1)

    def export(self, span: Dict[str, Any]) -> None:
        cid = span.get("container_id", "")
        trace_id = self._traces.setdefault(cid, _hex_id(16))
        start_ns = int(float(span.get("start", 0.0)) * 1e9)
        end_ns = int(float(span.get("end", 0.0)) * 1e9)
        attrs = [
            {"key": k, "value": {"stringValue": str(v)}}
            for k, v in span.items()
            if k not in ("name", "start", "end", "duration_ms")
        ]
        req = {
            "resourceSpans": [
                {
                    "resource": {
                        "attributes": [
                            {"key": "service.name",
                             "value": {"stringValue": self.service_name}}
                        ]
                    },
                    "scopeSpans": [
                        {
                            "scope": {"name": "beta9_amd.lifecycle"},
                            "spans": [
                                {
                                    "traceId": trace_id,
                                    "spanId": _hex_id(8),
                                    "name": span.get("name", "span"),
                                    "kind": 1,  # SPAN_KIND_INTERNAL
                                    "startTimeUnixNano": str(start_ns),
                                    "endTimeUnixNano": str(end_ns),
                                    "attributes": attrs,
                                }
                            ],
                        }
                    ],
                }
            ]
        }
        self._f.write(json.dumps(req) + "\n")

    def close(self) -> None:
        try:
            self._f.close()
        except OSError:
            pass


_exporter: Optional[OtlpSpanExporter] = None


def get_exporter(path: str = "") -> Optional[OtlpSpanExporter]:
    """Process-wide exporter, enabled by config/env (B9_OTLP_PATH)."""
    global _exporter
    if _exporter is None:
        p = path or os.environ.get("B9_OTLP_PATH", "")
        if p:
            _exporter = OtlpSpanExporter(p)
    return _exporter
