"""End-to-end on CPU: control plane + tiny local engine — the agent loop
runs against real engine inference (constrained tool calls + free answers),
mirroring BASELINE.json config 2 at test scale."""
import pytest

from agentcontrolplane_amd.api.types import (
    AGENT,
    LLM,
    MCP_SERVER,
    TASK,
    TaskPhase,
    make_resource,
)
from agentcontrolplane_amd.engine.config import EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for


@pytest.fixture(scope="module")
def engine():
    eng = InferenceEngine(
        EngineConfig(
            model="tiny", device="cpu", num_kv_blocks=4096, kv_block_size=4,
            max_prefill_tokens=512, request_timeout_s=300,
        )
    )
    yield eng
    eng.stop()


@pytest.fixture
def cp(engine):
    plane = ControlPlane(engine=engine, auto_approve="approve")
    plane.start()
    yield plane
    plane.stop()


def test_agent_loop_on_local_engine(cp):
    store = cp.store
    store.create(
        make_resource(
            LLM,
            "local-llm",
            spec={
                "provider": "local",
                "parameters": {"model": "tiny", "maxTokens": 32, "temperature": "0.8"},
            },
        )
    )
    cp.mcp.register_inproc("calc", {"add": lambda a=0, b=0, **_: str(float(a) + float(b))})
    store.create(make_resource(MCP_SERVER, "calc", spec={"transport": "inproc"}))
    store.create(
        make_resource(
            AGENT,
            "a1",
            spec={
                "llmRef": {"name": "local-llm"},
                "system": "you are a calculator",
                "mcpServers": [{"name": "calc"}],
            },
        )
    )
    store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "add"})
    )
    task = wait_for(
        lambda: (store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        in (TaskPhase.FINAL_ANSWER, TaskPhase.FAILED)
        and store.get(TASK, "t1"),
        timeout=120,
    )
    assert task["status"]["phase"] == TaskPhase.FINAL_ANSWER, task["status"]
    cw = task["status"]["contextWindow"]
    roles = [m["role"] for m in cw]
    assert roles == ["system", "user", "assistant", "tool", "assistant"]
    # the constrained tool call named a real tool with valid JSON arguments
    tc = cw[2]["toolCalls"][0]
    assert tc["function"]["name"] == "calc__add"
    import json

    json.loads(tc["function"]["arguments"])
    # the tool actually executed (result is a float string or an error-free str)
    assert cw[3]["content"] != ""


def test_sub_agent_delegation_via_engine(cp):
    """Config 4 at test scale: the ENGINE's constrained decoding picks among
    delegate_to_agent__ tools; delegation spawns a child Task that runs its
    own loop on the same engine."""
    store = cp.store
    store.create(
        make_resource(
            LLM,
            "local-llm",
            spec={
                "provider": "local",
                "parameters": {"model": "tiny", "maxTokens": 48, "temperature": "0.9"},
            },
        )
    )
    store.create(
        make_resource(AGENT, "worker", spec={"llmRef": {"name": "local-llm"}, "system": "worker"})
    )
    store.create(
        make_resource(
            AGENT,
            "lead",
            spec={
                "llmRef": {"name": "local-llm"},
                "system": "lead",
                "subAgents": [{"name": "worker"}],
            },
        )
    )
    store.create(
        make_resource(TASK, "d1", spec={"agentRef": {"name": "lead"}, "userMessage": "do it"})
    )
    task = wait_for(
        lambda: (store.get(TASK, "d1") or {}).get("status", {}).get("phase")
        in (TaskPhase.FINAL_ANSWER, TaskPhase.FAILED)
        and store.get(TASK, "d1"),
        timeout=240,
    )
    assert task["status"]["phase"] == TaskPhase.FINAL_ANSWER, task["status"]
    cw = task["status"]["contextWindow"]
    # the only offered tool is the delegate tool → the constrained turn
    # must have called it, and the child's answer is the tool result
    tc = cw[2]["toolCalls"][0]
    assert tc["function"]["name"] == "delegate_to_agent__worker"
    children = [
        t for t in store.list(TASK)
        if (t["metadata"].get("labels") or {}).get("acp.humanlayer.dev/parent-toolcall")
    ]
    assert children and children[0]["status"]["phase"] == TaskPhase.FINAL_ANSWER
