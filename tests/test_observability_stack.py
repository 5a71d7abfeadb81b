"""Structural validation of the observability stack (observability/).

No docker daemon exists here, so the stack is checked the way a config
reviewer would: every YAML parses, the compose graph is closed (services
reference each other consistently), and the Grafana dashboard's PromQL
queries only use metric names the REST server actually emits."""
from __future__ import annotations

import json
import os
import re

import yaml

ROOT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    "observability")


def test_compose_and_configs_parse():
    with open(os.path.join(ROOT, "docker-compose.yaml")) as f:
        compose = yaml.safe_load(f)
    services = compose["services"]
    assert {"otel-collector", "tempo", "prometheus", "grafana"} <= set(services)
    # every mounted config exists
    for svc in services.values():
        for vol in svc.get("volumes", []):
            src = vol.split(":")[0]
            assert os.path.exists(os.path.join(ROOT, src)), src
    with open(os.path.join(ROOT, "otel-collector.yaml")) as f:
        otel = yaml.safe_load(f)
    assert otel["exporters"]["otlp/tempo"]["endpoint"].startswith("tempo:")
    with open(os.path.join(ROOT, "prometheus.yaml")) as f:
        prom = yaml.safe_load(f)
    assert prom["scrape_configs"][0]["metrics_path"] == "/metrics"


def test_dashboard_queries_match_served_metrics():
    from starlette.testclient import TestClient

    from agentcontrolplane_amd.runtime import ControlPlane

    cp = ControlPlane(llm_probe=False)
    cp.start()
    try:
        # per-kind gauges only appear for kinds with live objects
        from agentcontrolplane_amd.api.types import LLM, make_resource

        cp.store.create(make_resource(LLM, "m", spec={"provider": "mock"}))
        client = TestClient(cp.rest_app)
        body = client.get("/metrics").text
    finally:
        cp.stop()
    served = set(re.findall(r"^([a-zA-Z_:][a-zA-Z0-9_:]*)", body, re.M))
    with open(os.path.join(ROOT, "grafana", "acp-dashboard.json")) as f:
        dash = json.load(f)
    used = set()
    for p in dash["panels"]:
        for t in p.get("targets", []):
            used.update(re.findall(r"acp_[a-z_]+", t["expr"]))
    missing = used - served
    # engine metrics only appear when an engine is attached; the
    # control-plane ones must all be live
    cp_metrics = {m for m in missing if not m.startswith("acp_engine_")}
    assert not cp_metrics, f"dashboard queries unknown metrics: {cp_metrics}"
    # engine metric names match the engine.metrics() keys
    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine

    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=64), start=False
    )
    keys = {f"acp_engine_{k}" for k in eng.metrics()}
    eng_missing = {m for m in missing if m.startswith("acp_engine_")} - keys
    assert not eng_missing, f"dashboard queries unknown engine metrics: {eng_missing}"
