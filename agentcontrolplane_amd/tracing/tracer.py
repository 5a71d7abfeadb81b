"""Lightweight OpenTelemetry-shaped tracer.

The reference initializes OTLP-HTTP exporters with a graceful no-op fallback
(acp/internal/otel/otel.go:23-80) and keeps long-lived task traces alive
across reconcile loops by persisting {TraceID, SpanID} in CR status and
re-attaching each reconcile (task_types.go:100-106,
task/state_machine.go:119-145, task_helpers.go:58-81).

The otel SDK is not installed in this image, so this module implements the
same surface natively: 128-bit trace ids / 64-bit span ids (W3C format),
parent propagation from persisted SpanContext (``Remote=true`` semantics),
span events/attributes, and an in-process ring buffer that the REST server
and tests can read.  Engine metrics (tokens/s, KV occupancy, per-kernel
times) are reported through the same registry (see engine.metrics).
"""
from __future__ import annotations

import dataclasses
import secrets
import threading
import time
from typing import Any, Dict, List, Optional


def _gen_trace_id() -> str:
    return secrets.token_hex(16)


def _gen_span_id() -> str:
    return secrets.token_hex(8)


@dataclasses.dataclass
class SpanContextData:
    trace_id: str
    span_id: str
    remote: bool = False


def reconstruct_span_context(trace_id: str, span_id: str) -> SpanContextData:
    """task_helpers.go:58-81 — rebuild a remote, sampled parent context from
    the ids persisted in CR status."""
    if len(trace_id) != 32 or len(span_id) != 16:
        raise ValueError(f"invalid span context ids: {trace_id!r}/{span_id!r}")
    return SpanContextData(trace_id=trace_id, span_id=span_id, remote=True)


@dataclasses.dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_span_id: str = ""
    start_ns: int = 0
    end_ns: int = 0
    status: str = "UNSET"  # OK | ERROR | UNSET
    status_message: str = ""
    attributes: Dict[str, Any] = dataclasses.field(default_factory=dict)
    events: List[Dict[str, Any]] = dataclasses.field(default_factory=list)
    _tracer: Optional["Tracer"] = None

    def set_attribute(self, key: str, value: Any) -> None:
        self.attributes[key] = value

    def add_event(self, name: str, attributes: Optional[Dict[str, Any]] = None) -> None:
        self.events.append({"name": name, "time_ns": time.time_ns(), "attributes": attributes or {}})

    def record_error(self, err: BaseException) -> None:
        self.add_event("exception", {"exception.message": str(err), "exception.type": type(err).__name__})

    def set_status(self, status: str, message: str = "") -> None:
        self.status = status
        self.status_message = message

    def end(self) -> None:
        self.end_ns = time.time_ns()
        if self._tracer is not None:
            self._tracer._export(self)

    def context(self) -> SpanContextData:
        return SpanContextData(self.trace_id, self.span_id)

    def __enter__(self) -> "Span":
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        if exc is not None:
            self.record_error(exc)
            self.set_status("ERROR", str(exc))
        self.end()


class Tracer:
    """Per-service tracer with an exported-span ring buffer."""

    def __init__(self, service: str = "acp-controller", capacity: int = 65536):
        self.service = service
        self._lock = threading.Lock()
        self._spans: List[Span] = []
        self._capacity = capacity

    def start(
        self,
        name: str,
        parent: Optional[SpanContextData] = None,
        attributes: Optional[Dict[str, Any]] = None,
    ) -> Span:
        if parent is not None and parent.trace_id:
            trace_id, parent_id = parent.trace_id, parent.span_id
        else:
            trace_id, parent_id = _gen_trace_id(), ""
        span = Span(
            name=name,
            trace_id=trace_id,
            span_id=_gen_span_id(),
            parent_span_id=parent_id,
            start_ns=time.time_ns(),
            attributes=dict(attributes or {}),
            _tracer=self,
        )
        return span

    def _export(self, span: Span) -> None:
        with self._lock:
            self._spans.append(span)
            if len(self._spans) > self._capacity:
                self._spans = self._spans[-self._capacity // 2 :]

    def finished_spans(self, trace_id: Optional[str] = None) -> List[Span]:
        with self._lock:
            if trace_id is None:
                return list(self._spans)
            return [s for s in self._spans if s.trace_id == trace_id]

    def reset(self) -> None:
        with self._lock:
            self._spans.clear()


_global_tracer: Optional[Tracer] = None
_global_lock = threading.Lock()


def get_tracer(service: str = "acp-controller") -> Tracer:
    global _global_tracer
    with _global_lock:
        if _global_tracer is None:
            _global_tracer = Tracer(service)
        return _global_tracer
