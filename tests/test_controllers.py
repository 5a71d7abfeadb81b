"""Controller tier: state-transition specs per phase edge.

Mirrors the reference's envtest controller suites (one Context
per phase edge, e.g. "'' -> Initializing", "ReadyForLLM -> LLMFinalAnswer")
— here driven through a live ControllerManager with the mock LLM provider,
exactly the seam the reference's mockgen factories provide.
"""
import json
import time

import pytest

from agentcontrolplane_amd.api.types import (
    AGENT,
    CONTACT_CHANNEL,
    LLM,
    MCP_SERVER,
    SECRET,
    TASK,
    TOOL_CALL,
    TaskPhase,
    ToolCallPhase,
    make_resource,
)
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for


@pytest.fixture
def cp():
    plane = ControlPlane(auto_approve="approve", llm_probe=True)
    plane.start()
    yield plane
    plane.stop()


def make_basic_world(cp, with_mcp=True, llm_provider="mock", agent_name="a1"):
    store = cp.store
    store.create(make_resource(LLM, "llm1", spec={"provider": llm_provider}))
    if with_mcp:
        cp.mcp.register_inproc(
            "calc",
            {"add": lambda a=0, b=0, **_: str(float(a) + float(b))},
        )
        store.create(make_resource(MCP_SERVER, "calc", spec={"transport": "inproc"}))
        agent_spec = {
            "llmRef": {"name": "llm1"},
            "system": "you are a test agent",
            "mcpServers": [{"name": "calc"}],
        }
    else:
        agent_spec = {"llmRef": {"name": "llm1"}, "system": "you are a test agent"}
    store.create(make_resource(AGENT, agent_name, spec=agent_spec))
    return store


# ----------------------------------------------------------------- LLM/Agent


def test_llm_validation_mock_ready(cp):
    cp.store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    llm = wait_for(lambda: (cp.store.get(LLM, "llm1") or {}).get("status", {}).get("ready") and cp.store.get(LLM, "llm1"))
    assert llm["status"]["status"] == "Ready"


def test_llm_unsupported_provider_error(cp):
    cp.store.create(make_resource(LLM, "bad", spec={"provider": "doesnotexist"}))
    llm = wait_for(
        lambda: (cp.store.get(LLM, "bad") or {}).get("status", {}).get("status") == "Error"
        and cp.store.get(LLM, "bad")
    )
    assert "unsupported provider" in llm["status"]["statusDetail"]


def test_remote_provider_requires_secret(cp):
    cp.store.create(make_resource(LLM, "oai", spec={"provider": "openai"}))
    llm = wait_for(
        lambda: (cp.store.get(LLM, "oai") or {}).get("status", {}).get("status") == "Error"
        and cp.store.get(LLM, "oai")
    )
    assert "secretKeyRef" in llm["status"]["statusDetail"]


def test_agent_ready_with_dependencies(cp):
    make_basic_world(cp)
    agent = wait_for(
        lambda: (cp.store.get(AGENT, "a1") or {}).get("status", {}).get("ready")
        and cp.store.get(AGENT, "a1")
    )
    assert agent["status"]["validMCPServers"] == [{"name": "calc", "tools": ["add"]}]


def test_agent_missing_llm_is_error(cp):
    cp.store.create(
        make_resource(AGENT, "a1", spec={"llmRef": {"name": "nope"}, "system": "s"})
    )
    agent = wait_for(
        lambda: (cp.store.get(AGENT, "a1") or {}).get("status", {}).get("status") == "Error"
        and cp.store.get(AGENT, "a1")
    )
    assert "not found" in agent["status"]["statusDetail"]


def test_mcpserver_connects_and_publishes_tools(cp):
    cp.mcp.register_inproc("srv", {"echo": lambda text="", **_: text})
    cp.store.create(make_resource(MCP_SERVER, "srv", spec={"transport": "inproc"}))
    srv = wait_for(
        lambda: (cp.store.get(MCP_SERVER, "srv") or {}).get("status", {}).get("connected")
        and cp.store.get(MCP_SERVER, "srv")
    )
    assert [t["name"] for t in srv["status"]["tools"]] == ["echo"]


def test_contactchannel_validation(cp):
    cp.store.create(
        make_resource(SECRET, "hl-key", spec={"data": {"key": "hl-abc123"}}, api_version="v1")
    )
    cp.store.create(
        make_resource(
            CONTACT_CHANNEL,
            "cc1",
            spec={
                "type": "slack",
                "apiKeyFrom": {"secretKeyRef": {"name": "hl-key", "key": "key"}},
                "slack": {"channelOrUserID": "C123"},
            },
        )
    )
    cc = wait_for(
        lambda: (cp.store.get(CONTACT_CHANNEL, "cc1") or {}).get("status", {}).get("ready")
        and cp.store.get(CONTACT_CHANNEL, "cc1")
    )
    assert cc["status"]["projectSlug"] == "local-project"


def test_contactchannel_mutual_exclusion(cp):
    cp.store.create(
        make_resource(
            CONTACT_CHANNEL,
            "bad",
            spec={
                "type": "slack",
                "apiKeyFrom": {"secretKeyRef": {"name": "s", "key": "k"}},
                "channelApiKeyFrom": {"secretKeyRef": {"name": "s", "key": "k"}},
                "slack": {"channelOrUserID": "C1"},
            },
        )
    )
    cc = wait_for(
        lambda: (cp.store.get(CONTACT_CHANNEL, "bad") or {}).get("status", {}).get("status")
        == "Error"
        and cp.store.get(CONTACT_CHANNEL, "bad")
    )
    assert "mutually exclusive" in cc["status"]["statusDetail"]


# ----------------------------------------------------------------- Task loop


def test_task_full_agent_loop_with_tool(cp):
    """'' → Initializing → ReadyForLLM → ToolCallsPending → ReadyForLLM →
    FinalAnswer, with the tool result appended in the checkpoint."""
    make_basic_world(cp)
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "add 1 2"})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, "t1"),
        timeout=20,
    )
    cw = task["status"]["contextWindow"]
    roles = [m["role"] for m in cw]
    # system, user, assistant(toolCalls), tool, assistant(final)
    assert roles == ["system", "user", "assistant", "tool", "assistant"]
    assert cw[2]["toolCalls"][0]["function"]["name"] == "calc__add"
    assert cw[3]["content"] == "3.0"
    assert task["status"]["output"] == "mock final answer"
    assert task["status"]["messageCount"] == 5
    # events history matches the reference reasons
    reasons = [e["reason"] for e in cp.store.events_for("t1")]
    for expected in (
        "ValidationSucceeded",
        "SendingContextWindowToLLM",
        "ToolCallsPending",
        "ToolCallCreated",
        "AllToolCallsCompleted",
        "LLMFinalAnswer",
    ):
        assert expected in reasons, f"{expected} missing from {reasons}"
    # spans: root Task + ≥2 LLMRequest + EndTaskSpan under one trace
    # (the EndTaskSpan is emitted by the terminal reconcile that follows the
    # FinalAnswer status flip — wait for it)
    trace_id = task["status"]["spanContext"]["traceID"]
    names = wait_for(
        lambda: (lambda ns: ns if "EndTaskSpan" in ns else None)(
            [s.name for s in cp.tracer.finished_spans(trace_id)]
        ),
        timeout=10,
    )
    assert names.count("LLMRequest") >= 2


def test_task_missing_agent_pending(cp):
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "ghost"}, "userMessage": "hi"})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.PENDING
        and cp.store.get(TASK, "t1")
    )
    assert "Waiting for Agent" in task["status"]["statusDetail"]


def test_task_invalid_input_fails(cp):
    make_basic_world(cp, with_mcp=False)
    wait_for(lambda: (cp.store.get(AGENT, "a1") or {}).get("status", {}).get("ready"))
    cp.store.create(make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}}))
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FAILED
        and cp.store.get(TASK, "t1")
    )
    assert "must be provided" in task["status"]["error"]


def test_task_context_window_input(cp):
    make_basic_world(cp, with_mcp=False)
    cw = [
        {"role": "user", "content": "continue the conversation"},
    ]
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "contextWindow": cw})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, "t1"),
        timeout=20,
    )
    # system prompt inserted at position 0 (task_helpers.go:13-44)
    assert task["status"]["contextWindow"][0]["role"] == "system"


def test_llm_4xx_terminal(cp):
    from agentcontrolplane_amd.llmclient.base import LLMRequestError
    from agentcontrolplane_amd.llmclient.mock import MockLLMClient

    class Bad(MockLLMClient):
        def send_request(self, messages, tools):
            # the LLM controller's 1-token probe must succeed so the LLM goes
            # Ready; task-sized requests fail 4xx
            if len(messages) == 1 and messages[0].content == "ping":
                return super().send_request(messages, tools)
            raise LLMRequestError(400, "bad request")

    cp.llm_factory._mock_factory = lambda llm: Bad()
    make_basic_world(cp, with_mcp=False)
    agent = wait_for(
        lambda: (cp.store.get(AGENT, "a1") or {}).get("status", {}).get("ready")
        and cp.store.get(AGENT, "a1")
    )
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "x"})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FAILED
        and cp.store.get(TASK, "t1"),
        timeout=20,
    )
    assert "status 400" in task["status"]["error"]
    assert any(e["reason"] == "LLMRequestFailed4xx" for e in cp.store.events_for("t1"))


def test_toolcall_names_and_labels(cp):
    make_basic_world(cp)
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "add"})
    )
    wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER,
        timeout=20,
    )
    tcs = cp.store.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "t1"})
    assert len(tcs) == 1
    tc = tcs[0]
    req_id = cp.store.get(TASK, "t1")["status"]["toolCallRequestId"]
    assert tc["metadata"]["name"] == f"t1-{req_id}-tc-01"
    assert tc["metadata"]["labels"]["acp.humanlayer.dev/toolcallrequest"] == req_id
    assert tc["status"]["phase"] == ToolCallPhase.SUCCEEDED
    assert tc["metadata"]["ownerReferences"][0]["kind"] == "Task"


# ------------------------------------------------------------ approval gate


def test_approval_gate_approved(cp):
    """MCP server with approvalContactChannel → AwaitingHumanApproval →
    approved → executes."""
    store = cp.store
    store.create(
        make_resource(SECRET, "hl", spec={"data": {"k": "hl-x"}}, api_version="v1")
    )
    store.create(
        make_resource(
            CONTACT_CHANNEL,
            "approver",
            spec={
                "type": "slack",
                "apiKeyFrom": {"secretKeyRef": {"name": "hl", "key": "k"}},
                "slack": {"channelOrUserID": "C1"},
            },
        )
    )
    store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    cp.mcp.register_inproc("calc", {"add": lambda a=0, b=0, **_: str(float(a) + float(b))})
    store.create(
        make_resource(
            MCP_SERVER,
            "calc",
            spec={"transport": "inproc", "approvalContactChannel": {"name": "approver"}},
        )
    )
    store.create(
        make_resource(
            AGENT,
            "a1",
            spec={
                "llmRef": {"name": "llm1"},
                "system": "s",
                "mcpServers": [{"name": "calc"}],
            },
        )
    )
    store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "add"})
    )
    task = wait_for(
        lambda: (store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and store.get(TASK, "t1"),
        timeout=30,
    )
    tcs = store.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "t1"})
    tc_events = [e["reason"] for e in store.events_for(tcs[0]["metadata"]["name"])]
    assert "AwaitingHumanApproval" in tc_events and "Approved" in tc_events
    assert task["status"]["contextWindow"][3]["content"] == "3.0"


def test_approval_gate_rejected_feeds_back(cp):
    cp.humanlayer.auto = "reject"
    cp.humanlayer.comment = "not allowed"
    store = cp.store
    store.create(make_resource(SECRET, "hl", spec={"data": {"k": "hl-x"}}, api_version="v1"))
    store.create(
        make_resource(
            CONTACT_CHANNEL,
            "approver",
            spec={
                "type": "slack",
                "apiKeyFrom": {"secretKeyRef": {"name": "hl", "key": "k"}},
                "slack": {"channelOrUserID": "C1"},
            },
        )
    )
    store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    cp.mcp.register_inproc("calc", {"add": lambda a=0, b=0, **_: "3.0"})
    store.create(
        make_resource(
            MCP_SERVER,
            "calc",
            spec={"transport": "inproc", "approvalContactChannel": {"name": "approver"}},
        )
    )
    store.create(
        make_resource(
            AGENT,
            "a1",
            spec={"llmRef": {"name": "llm1"}, "system": "s", "mcpServers": [{"name": "calc"}]},
        )
    )
    store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "add"})
    )
    task = wait_for(
        lambda: (store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and store.get(TASK, "t1"),
        timeout=30,
    )
    # the rejection is delivered to the LLM as the tool result
    tool_msgs = [m for m in task["status"]["contextWindow"] if m["role"] == "tool"]
    assert tool_msgs[0]["content"] == "Rejected: not allowed"
    tcs = store.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "t1"})
    assert tcs[0]["status"]["phase"] == ToolCallPhase.TOOL_CALL_REJECTED
    assert tcs[0]["status"]["status"] == "Succeeded"


# -------------------------------------------------------- sub-agent fan-out


def test_sub_agent_delegation(cp):
    """Parent agent delegates to a child agent through delegate_to_agent__…"""
    from agentcontrolplane_amd.llmclient.mock import MockLLMClient, final_answer, tool_call_turn

    class Parent(MockLLMClient):
        def send_request(self, messages, tools):
            sub_tools = [t for t in tools if t.function.name.startswith("delegate_to_agent__")]
            done = any(m.role == "tool" for m in messages)
            if sub_tools and not done:
                from agentcontrolplane_amd.llmclient.base import normalize_response

                return normalize_response(
                    tool_call_turn(
                        [("c1", sub_tools[0].function.name, json.dumps({"message": "sub task"}))]
                    )
                )
            from agentcontrolplane_amd.llmclient.base import normalize_response

            return normalize_response(final_answer("parent done"))

    cp.llm_factory._mock_factory = lambda llm: Parent()
    store = cp.store
    store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    store.create(
        make_resource(AGENT, "child", spec={"llmRef": {"name": "llm1"}, "system": "child sys"})
    )
    store.create(
        make_resource(
            AGENT,
            "parent",
            spec={
                "llmRef": {"name": "llm1"},
                "system": "parent sys",
                "subAgents": [{"name": "child"}],
            },
        )
    )
    store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "parent"}, "userMessage": "go"})
    )
    task = wait_for(
        lambda: (store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and store.get(TASK, "t1"),
        timeout=30,
    )
    assert task["status"]["output"] == "parent done"
    # the child task ran its own loop to FinalAnswer
    children = [
        t
        for t in store.list(TASK)
        if (t["metadata"].get("labels") or {}).get("acp.humanlayer.dev/parent-toolcall")
    ]
    assert len(children) == 1
    assert children[0]["status"]["phase"] == TaskPhase.FINAL_ANSWER
    # and its answer is the parent's tool result
    tool_msgs = [m for m in task["status"]["contextWindow"] if m["role"] == "tool"]
    assert tool_msgs[0]["content"] == "parent done"  # child used same mock → same answer


def test_failing_tool_feeds_back(cp):
    """An MCP tool error completes the ToolCall with status Error; the loop
    proceeds (reference checkToolCalls treats Error as completed)."""
    def boom(**_):
        raise RuntimeError("tool exploded")

    cp.mcp.register_inproc("bad", {"explode": boom})
    cp.store.create(make_resource(LLM, "llm1", spec={"provider": "mock"}))
    cp.store.create(make_resource(MCP_SERVER, "bad", spec={"transport": "inproc"}))
    cp.store.create(
        make_resource(
            AGENT, "a1",
            spec={"llmRef": {"name": "llm1"}, "system": "s", "mcpServers": [{"name": "bad"}]},
        )
    )
    cp.store.create(
        make_resource(TASK, "t1", spec={"agentRef": {"name": "a1"}, "userMessage": "x"})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, "t1"),
        timeout=30,
    )
    tcs = cp.store.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "t1"})
    assert tcs[0]["status"]["status"] == "Error"
    assert "tool exploded" in tcs[0]["status"]["error"]
    # the loop still completed with a final answer
    assert task["status"]["output"] == "mock final answer"


def test_task_delete_mid_loop_cascades(cp):
    """Deleting a Task while ToolCalls are pending cascade-deletes the
    children (owner index) and leaves the manager healthy for new work."""
    make_basic_world(cp)
    cp.store.create(
        make_resource(TASK, "tdel", spec={"agentRef": {"name": "a1"}, "userMessage": "add 1 2"})
    )
    # wait until the ToolCall child exists, then delete the parent
    tc = wait_for(
        lambda: next(
            iter(
                cp.store.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "tdel"})
            ),
            None,
        ),
        timeout=20,
    )
    cp.store.delete(TASK, "tdel")
    wait_for(
        lambda: not cp.store.list(
            TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "tdel"}
        ),
        timeout=10,
    )
    assert cp.store.get(TASK, "tdel") is None
    # the control plane still runs new tasks to completion afterwards
    cp.store.create(
        make_resource(TASK, "tafter", spec={"agentRef": {"name": "a1"}, "userMessage": "add 2 3"})
    )
    task = wait_for(
        lambda: (cp.store.get(TASK, "tafter") or {}).get("status", {}).get("phase")
        == TaskPhase.FINAL_ANSWER
        and cp.store.get(TASK, "tafter"),
        timeout=20,
    )
    assert task["status"]["output"] == "mock final answer"


def test_soak_concurrent_task_mix(cp):
    """Soak: 40 concurrent tasks against one plane — plain tool loops plus
    mid-flight deletions racing the reconcilers.  Everything must reach a
    terminal phase (or be cleanly gone) with no stuck reconciles."""
    make_basic_world(cp)
    n = 40
    for i in range(n):
        cp.store.create(
            make_resource(
                TASK, f"soak-{i}",
                spec={"agentRef": {"name": "a1"}, "userMessage": f"add {i} 1"},
            )
        )
    # delete a few while they run
    import time as _t

    _t.sleep(0.2)
    for i in range(0, n, 10):
        try:
            cp.store.delete(TASK, f"soak-{i}")
        except Exception:
            pass

    def all_done():
        for i in range(n):
            t = cp.store.get(TASK, f"soak-{i}")
            if t is None:
                continue
            if t.get("status", {}).get("phase") not in (
                TaskPhase.FINAL_ANSWER, TaskPhase.FAILED,
            ):
                return None
        return True

    assert wait_for(all_done, timeout=60)
    # the survivors all carry the full checkpoint
    done = [cp.store.get(TASK, f"soak-{i}") for i in range(n)]
    done = [t for t in done if t is not None]
    assert len(done) >= n - 4
    for t in done:
        assert t["status"]["output"] == "mock final answer"
        assert [m["role"] for m in t["status"]["contextWindow"]][-1] == "assistant"


def test_workqueue_fuzz_no_concurrent_no_lost():
    """_WorkQueue invariants under concurrent producers/consumers:
    (a) a key is never handed to two workers at once,
    (b) every key enqueued after its last processing is processed again
        (the dirty-while-in-flight requeue)."""
    import random
    import threading
    import time as _t

    from agentcontrolplane_amd.controllers.manager import _WorkQueue

    q = _WorkQueue()
    keys = [("K", f"k{i}") for i in range(6)]
    active = set()
    active_lock = threading.Lock()
    processed = {k: 0 for k in keys}
    enqueued = {k: 0 for k in keys}
    violations = []
    stop = threading.Event()

    def producer(seed):
        rng = random.Random(seed)
        for _ in range(300):
            k = rng.choice(keys)
            with active_lock:
                enqueued[k] += 1
            if rng.random() < 0.3:
                q.add_after(k, rng.random() * 0.01)
            else:
                q.add(k)
            if rng.random() < 0.1:
                _t.sleep(0.001)

    def worker():
        while not stop.is_set():
            k = q.get(timeout=0.05)
            if k is None:
                continue
            with active_lock:
                if k in active:
                    violations.append(k)
                active.add(k)
            _t.sleep(0.0005)
            with active_lock:
                active.discard(k)
                processed[k] += 1
            q.done(k)

    workers = [threading.Thread(target=worker) for _ in range(4)]
    producers = [threading.Thread(target=producer, args=(i,)) for i in range(3)]
    for t in workers + producers:
        t.start()
    for t in producers:
        t.join()
    # drain: wait until nothing is pending
    deadline = _t.monotonic() + 10
    while _t.monotonic() < deadline:
        with q._cv:
            idle = not q._ready and not q._delayed and not q._in_flight
        if idle:
            break
        _t.sleep(0.02)
    stop.set()
    for t in workers:
        t.join()
    assert not violations, f"concurrent reconcile of {violations[:3]}"
    for k in keys:
        assert processed[k] >= 1, (k, enqueued[k])
    with q._cv:
        assert not q._dirty_while_in_flight
