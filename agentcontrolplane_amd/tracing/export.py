"""Span export.

Parity with the reference's otel init (acp/internal/otel/otel.go:23-80):
OTLP/HTTP export to ``OTEL_EXPORTER_OTLP_ENDPOINT`` with a graceful no-op
fallback, service name ``acp-controller``.  The otel SDK is not in this
image, so the exporter emits the OTLP JSON encoding directly (one
``/v1/traces`` POST per batch); a JSONL file sink covers air-gapped
deployments (and is what the tests read back).
"""
from __future__ import annotations

import json
import os
import threading
from typing import List, Optional

from .tracer import Span, Tracer


def span_to_otlp(span: Span) -> dict:
    return {
        "traceId": span.trace_id,
        "spanId": span.span_id,
        "parentSpanId": span.parent_span_id,
        "name": span.name,
        "kind": 1,
        "startTimeUnixNano": str(span.start_ns),
        "endTimeUnixNano": str(span.end_ns),
        "attributes": [
            {"key": k, "value": {"stringValue": str(v)}}
            for k, v in span.attributes.items()
        ],
        "events": [
            {
                "name": e["name"],
                "timeUnixNano": str(e["time_ns"]),
                "attributes": [
                    {"key": k, "value": {"stringValue": str(v)}}
                    for k, v in (e.get("attributes") or {}).items()
                ],
            }
            for e in span.events
        ],
        "status": {"code": {"OK": 1, "ERROR": 2}.get(span.status, 0),
                   "message": span.status_message},
    }


def otlp_payload(spans: List[Span], service: str = "acp-controller") -> dict:
    return {
        "resourceSpans": [
            {
                "resource": {
                    "attributes": [
                        {"key": "service.name", "value": {"stringValue": service}}
                    ]
                },
                "scopeSpans": [
                    {
                        "scope": {"name": "agentcontrolplane_amd"},
                        "spans": [span_to_otlp(s) for s in spans],
                    }
                ],
            }
        ]
    }


class SpanExporter:
    """Background exporter draining a Tracer's finished spans every
    ``interval_s`` to an OTLP endpoint and/or a JSONL file."""

    def __init__(
        self,
        tracer: Tracer,
        endpoint: Optional[str] = None,
        jsonl_path: Optional[str] = None,
        interval_s: float = 2.0,
    ):
        self.tracer = tracer
        self.endpoint = endpoint or os.environ.get("OTEL_EXPORTER_OTLP_ENDPOINT")
        self.jsonl_path = jsonl_path
        self.interval_s = interval_s
        self._cursor = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._client = None
        self.exported = 0
        self.export_errors = 0

    def start(self) -> "SpanExporter":
        self._thread = threading.Thread(target=self._loop, daemon=True, name="acp-otlp")
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
        self.flush()

    def _loop(self) -> None:
        while not self._stop.wait(self.interval_s):
            self.flush()

    def flush(self) -> None:
        spans = self.tracer.finished_spans()
        batch = spans[self._cursor :]
        if not batch:
            return
        self._cursor = len(spans)
        if self.jsonl_path:
            try:
                with open(self.jsonl_path, "a", encoding="utf-8") as f:
                    for s in batch:
                        f.write(json.dumps(span_to_otlp(s), separators=(",", ":")) + "\n")
                self.exported += len(batch)
            except OSError:
                self.export_errors += 1
        if self.endpoint:
            try:
                import httpx

                if self._client is None:
                    self._client = httpx.Client(timeout=5)
                url = self.endpoint.rstrip("/") + "/v1/traces"
                r = self._client.post(url, json=otlp_payload(batch))
                if r.status_code >= 400:
                    self.export_errors += 1
                else:
                    self.exported += len(batch)
            except Exception:
                # graceful no-op fallback, as the reference's init does
                self.export_errors += 1
