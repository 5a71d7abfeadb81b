"""REST API tier — mirrors the reference's server tests
(server_test.go: gin handlers against a fake client; here FastAPI TestClient
against a live control plane)."""
import json

import pytest
from fastapi.testclient import TestClient

from agentcontrolplane_amd.api.types import AGENT, LLM, TASK, TaskPhase, make_resource
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for


@pytest.fixture
def cp():
    plane = ControlPlane(auto_approve="approve")
    plane.start()
    yield plane
    plane.stop()


@pytest.fixture
def client(cp):
    return TestClient(cp.rest_app)


def _ready_agent(cp, name="a1"):
    cp.store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    cp.store.create(
        make_resource(AGENT, name, spec={"llmRef": {"name": "llm1"}, "system": "sys"})
    )
    wait_for(lambda: (cp.store.get(AGENT, name) or {}).get("status", {}).get("ready"))


def test_status(client):
    r = client.get("/status")
    assert r.status_code == 200 and r.json()["status"] == "ok"


def test_create_task_agent_missing_404(client):
    r = client.post("/v1/tasks", json={"agentName": "ghost", "userMessage": "hi"})
    assert r.status_code == 404


def test_create_task_unknown_field_400(client, cp):
    _ready_agent(cp)
    r = client.post("/v1/tasks", json={"agentName": "a1", "userMessage": "hi", "zzz": 1})
    assert r.status_code == 400
    assert "Unknown field" in r.json()["error"]


def test_create_task_both_inputs_400(client, cp):
    _ready_agent(cp)
    r = client.post(
        "/v1/tasks",
        json={
            "agentName": "a1",
            "userMessage": "hi",
            "contextWindow": [{"role": "user", "content": "x"}],
        },
    )
    assert r.status_code == 400


def test_create_task_runs_loop(client, cp):
    _ready_agent(cp)
    r = client.post("/v1/tasks", json={"agentName": "a1", "userMessage": "hello"})
    assert r.status_code == 201
    body = r.json()
    assert body["name"].startswith("a1-task-")
    task = wait_for(
        lambda: (cp.store.get(TASK, body["name"]) or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, body["name"]),
        timeout=20,
    )
    r2 = client.get(f'/v1/tasks/{body["name"]}')
    assert r2.status_code == 200
    assert r2.json()["output"] == "mock final answer"


def test_agent_crud(client, cp):
    r = client.post(
        "/v1/agents",
        json={
            "name": "api-agent",
            "systemPrompt": "from the API",
            "llm": {"provider": "mock", "model": "m"},
        },
    )
    assert r.status_code == 201
    wait_for(lambda: (cp.store.get(AGENT, "api-agent") or {}).get("status", {}).get("ready"))
    r = client.get("/v1/agents/api-agent")
    assert r.json()["ready"] is True
    r = client.put("/v1/agents/api-agent", json={"systemPrompt": "updated"})
    assert r.status_code == 200
    assert cp.store.get(AGENT, "api-agent")["spec"]["system"] == "updated"
    r = client.delete("/v1/agents/api-agent")
    assert r.status_code == 204
    assert cp.store.get(AGENT, "api-agent") is None


def test_agent_duplicate_409(client, cp):
    _ready_agent(cp)
    r = client.post(
        "/v1/agents",
        json={"name": "a1", "systemPrompt": "x", "llm": {"provider": "mock", "model": "m"}},
    )
    assert r.status_code == 409


def test_v1beta3_event_flow(client, cp):
    """server.go:1384-1545: the event auto-creates Secret + ContactChannel +
    labeled Task, and the loop ends with a respond_to_human ToolCall."""
    _ready_agent(cp)
    r = client.post(
        "/v1/beta3/events",
        json={
            "type": "agent_email.received",
            "data": {
                "agentName": "a1",
                "fromAddress": "user@example.com",
                "body": "please help",
                "eventId": "ev12345",
                "apiKey": "hl-beta3key",
            },
        },
    )
    assert r.status_code == 201
    task_name = r.json()["taskName"]
    # reference parity (state_machine.go:609-611, 968-1067): the content
    # turn becomes a respond_to_human ToolCall, checkToolCalls loops the
    # task back to ReadyForLLM, and the thread parks awaiting the next
    # inbound event (which would arrive as a NEW Task on the same threadID)
    task = wait_for(
        lambda: (
            (cp.store.get(TASK, task_name) or {}).get("status", {}).get("phase")
            == TaskPhase.READY_FOR_LLM
            and (cp.store.get(TASK, task_name) or {})["status"].get("statusDetail")
            == "Awaiting next inbound event"
            and cp.store.get(TASK, task_name)
        ),
        timeout=30,
    )
    # final answer flowed through a respond_to_human tool call
    cw = task["status"]["contextWindow"]
    rth = [
        m
        for m in cw
        if m["role"] == "assistant"
        and any(tc["function"]["name"] == "respond_to_human" for tc in m.get("toolCalls", []))
    ]
    assert rth, f"no respond_to_human turn in {json.dumps(cw, indent=1)}"
    from agentcontrolplane_amd.api.types import CONTACT_CHANNEL

    assert cp.store.get(CONTACT_CHANNEL, "v1beta3-channel-ev12345") is not None
    assert task["metadata"]["labels"]["acp.humanlayer.dev/v1beta3"] == "true"


def test_approvals_endpoint(client, cp):
    # pending approval resolves through the REST surface
    from agentcontrolplane_amd.humanlayer import HumanLayerClientFactory

    cp.humanlayer.auto = None  # stop auto-approving
    hl = HumanLayerClientFactory(cp.store).new_client()
    call_id = hl.request_approval("tool", "{}")
    r = client.get("/v1/approvals")
    assert any(a["id"] == call_id and not a["resolved"] for a in r.json())
    r = client.post(f"/v1/approvals/{call_id}", json={"approved": True, "comment": "ok"})
    assert r.status_code == 200
    st = hl.get_function_call_status(call_id)
    assert st.approved is True


def test_metrics_endpoint(client, cp):
    r = client.get("/metrics")
    assert r.status_code == 200
    assert "acp_up 1" in r.text


def test_openai_chat_completions_endpoint():
    """OpenAI-wire completions against the in-process engine."""
    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine

    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=1024, kv_block_size=16,
                     max_prefill_tokens=256, request_timeout_s=120)
    )
    plane = ControlPlane(engine=eng, auto_approve="approve", llm_probe=False)
    plane.start()
    try:
        client = TestClient(plane.rest_app)
        r = client.get("/v1/models")
        assert r.json()["data"][0]["id"] == "tiny"
        r = client.post(
            "/v1/chat/completions",
            json={
                "model": "tiny",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 8,
                "temperature": 0.9,
            },
        )
        assert r.status_code == 200, r.text
        body = r.json()
        # random-init weights rarely emit EOT within 8 tokens: 'length'
        # must pass through unmapped (truncation visible to OpenAI clients)
        assert body["choices"][0]["finish_reason"] in ("stop", "length")
        assert body["usage"]["completion_tokens"] <= 8
        # tool-calling turn
        r = client.post(
            "/v1/chat/completions",
            json={
                "model": "tiny",
                "messages": [{"role": "user", "content": "use the tool"}],
                "tools": [{"type": "function",
                           "function": {"name": "calc__add", "parameters": {}}}],
                "tool_choice": "required",
                "max_tokens": 48,
            },
        )
        body = r.json()
        choice = body["choices"][0]
        assert choice["finish_reason"] == "tool_calls"
        assert choice["message"]["tool_calls"][0]["function"]["name"] == "calc__add"
        import json as _json

        _json.loads(choice["message"]["tool_calls"][0]["function"]["arguments"])
    finally:
        plane.stop()
        eng.stop()


def test_openai_streaming_completions():
    """SSE streaming: per-token content deltas concatenate to the full
    answer; terminal chunk carries finish_reason; stream ends with [DONE]."""
    import json as _json

    from agentcontrolplane_amd.engine.config import EngineConfig
    from agentcontrolplane_amd.engine.engine import InferenceEngine

    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=1024, kv_block_size=16,
                     max_prefill_tokens=256, request_timeout_s=120)
    )
    plane = ControlPlane(engine=eng, auto_approve="approve", llm_probe=False)
    plane.start()
    try:
        client = TestClient(plane.rest_app)
        with client.stream(
            "POST", "/v1/chat/completions",
            json={"model": "tiny", "stream": True, "max_tokens": 8,
                  "temperature": 0.9,
                  "messages": [{"role": "user", "content": "hi"}]},
        ) as r:
            assert r.status_code == 200
            assert r.headers["content-type"].startswith("text/event-stream")
            events = []
            for line in r.iter_lines():
                if line.startswith("data: "):
                    events.append(line[len("data: "):])
        assert events[-1] == "[DONE]"
        chunks = [_json.loads(e) for e in events[:-1]]
        assert chunks[0]["choices"][0]["delta"] == {"role": "assistant"}
        content = "".join(
            c["choices"][0]["delta"].get("content", "") for c in chunks
        )
        finishes = [c["choices"][0]["finish_reason"] for c in chunks]
        assert finishes[-1] in ("stop", "length")
        # the streamed text matches a non-streamed run of the same request
        # only for greedy sampling; here just require SOME deltas arrived
        assert len(content) >= 0 and len(chunks) >= 2
    finally:
        plane.stop()
