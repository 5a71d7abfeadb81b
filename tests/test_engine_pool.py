"""EnginePool request router (parallel/router.py): least-loaded routing,
engine-facade parity, and the full control-plane loop over a pool."""
from __future__ import annotations

import threading

from agentcontrolplane_amd.api.types import (
    AGENT, LLM, MCP_SERVER, TASK, TaskPhase, make_resource,
)
from agentcontrolplane_amd.engine.config import EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.engine.request import SamplingParams
from agentcontrolplane_amd.parallel.router import EnginePool
from conftest import wait_for


def _pool(n=2):
    engines = [
        InferenceEngine(EngineConfig(model="tiny", device="cpu",
                                     num_kv_blocks=256, seed=i))
        for i in range(n)
    ]
    return EnginePool(engines)


def test_least_loaded_routing():
    pool = _pool(2)
    try:
        done = threading.Event()
        results = []

        def cb(r, e):
            results.append((r, e))
            if len(results) == 4:
                done.set()

        for _ in range(4):
            pool.chat_async([{"role": "user", "content": "hi"}], [],
                            SamplingParams(max_tokens=4), cb)
        assert done.wait(30)
        assert all(e is None for _, e in results)
        m = pool.metrics()
        # both engines took work (least-loaded spreads concurrent turns)
        assert m["pool_routed_0"] > 0 and m["pool_routed_1"] > 0
        assert m["pool_size"] == 2 and m["pool_inflight"] == 0
        assert m["generated_tokens"] > 0  # aggregated across engines
    finally:
        pool.stop()


def test_control_plane_over_pool():
    from agentcontrolplane_amd.runtime import ControlPlane

    pool = _pool(2)
    cp = ControlPlane(engine=pool, auto_approve="approve", llm_probe=False)
    cp.start()
    try:
        cp.store.create(make_resource(LLM, "pool-llm", spec={
            "provider": "local",
            "parameters": {"model": "tiny", "maxTokens": 8},
        }))
        llm = cp.store.get(LLM, "pool-llm")
        llm["status"].update({"ready": True, "status": "Ready"})
        cp.store.update_status(llm)
        cp.mcp.register_inproc("tools", {"noop": lambda **_: "ok"})
        cp.store.create(make_resource(MCP_SERVER, "tools",
                                      spec={"transport": "inproc"}))
        cp.store.create(make_resource(AGENT, "pool-agent", spec={
            "llmRef": {"name": "pool-llm"},
            "system": "sys",
            "mcpServers": [{"name": "tools"}],
        }))
        wait_for(lambda: (cp.store.get(AGENT, "pool-agent") or {})
                 .get("status", {}).get("ready"), timeout=15)
        for i in range(6):
            cp.store.create(make_resource(TASK, f"pool-task-{i}", spec={
                "agentRef": {"name": "pool-agent"},
                "userMessage": f"task {i}",
            }))
        for i in range(6):
            wait_for(
                lambda i=i: (cp.store.get(TASK, f"pool-task-{i}") or {})
                .get("status", {}).get("phase") == TaskPhase.FINAL_ANSWER,
                timeout=60,
            )
        m = pool.metrics()
        assert m["pool_routed_0"] > 0 and m["pool_routed_1"] > 0
    finally:
        cp.stop()
        pool.stop()
