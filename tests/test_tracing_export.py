"""Span export: OTLP JSON shape + JSONL sink round-trip; span parenting
across reconcile boundaries via persisted span context."""
import json

from agentcontrolplane_amd.tracing import Tracer, reconstruct_span_context
from agentcontrolplane_amd.tracing.export import SpanExporter, otlp_payload, span_to_otlp


def test_span_parenting_via_persisted_context():
    tr = Tracer()
    root = tr.start("Task")
    persisted = {"traceID": root.trace_id, "spanID": root.span_id}
    root.end()
    # a later reconcile rebuilds the parent from status (remote context)
    parent = reconstruct_span_context(persisted["traceID"], persisted["spanID"])
    child = tr.start("LLMRequest", parent=parent, attributes={"messages": 2})
    child.set_status("OK")
    child.end()
    spans = tr.finished_spans(root.trace_id)
    assert [s.name for s in spans] == ["Task", "LLMRequest"]
    assert spans[1].parent_span_id == root.span_id


def test_otlp_payload_shape(tmp_path):
    tr = Tracer()
    s = tr.start("LLMRequest", attributes={"messages": 3})
    s.add_event("retry", {"attempt": 1})
    s.set_status("ERROR", "boom")
    s.end()
    d = span_to_otlp(s)
    assert len(d["traceId"]) == 32 and len(d["spanId"]) == 16
    assert d["status"]["code"] == 2
    assert d["events"][0]["name"] == "retry"
    payload = otlp_payload([s])
    rs = payload["resourceSpans"][0]
    assert rs["resource"]["attributes"][0]["value"]["stringValue"] == "acp-controller"
    assert rs["scopeSpans"][0]["spans"][0]["name"] == "LLMRequest"

    # JSONL sink
    path = str(tmp_path / "spans.jsonl")
    exp = SpanExporter(tr, endpoint=None, jsonl_path=path)
    exp.flush()
    lines = [json.loads(l) for l in open(path)]
    assert lines and lines[0]["name"] == "LLMRequest"
    assert exp.exported == 1
