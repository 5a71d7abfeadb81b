"""Human-contact tool flow (AwaitingHumanInput → response feeds the LLM)
and MCP stdio reconnection after a server crash."""
import sys
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
from agentcontrolplane_amd.humanlayer import MockHumanLayerClientFactory
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for


def test_human_contact_tool_round_trip(store):
    """Agent offers a contact-channel tool; the LLM calls it; the ToolCall
    waits on human input; the human's response becomes the tool result."""
    import json

    from agentcontrolplane_amd.llmclient.base import normalize_response
    from agentcontrolplane_amd.llmclient.mock import MockLLMClient, final_answer, tool_call_turn

    class ContactCaller(MockLLMClient):
        def send_request(self, messages, tools):
            contact_tools = [
                t for t in tools if t.function.name.endswith("__human_contact_slack")
            ]
            if contact_tools and not any(m.role == "tool" for m in messages):
                return normalize_response(
                    tool_call_turn(
                        [("c1", contact_tools[0].function.name,
                          json.dumps({"message": "may I proceed?"}))]
                    )
                )
            return normalize_response(final_answer("done after human input"))

    cp = ControlPlane(
        humanlayer_factory=None, auto_approve=None, llm_probe=True,
    )
    cp.humanlayer = MockHumanLayerClientFactory(
        cp.store, auto_response="yes, proceed", delay_s=0.1
    )
    # rebuild reconcilers' humanlayer reference: the factory was injected at
    # construction; patch the two reconcilers directly
    for rec in cp.manager._reconcilers:
        if hasattr(rec, "humanlayer"):
            rec.humanlayer = cp.humanlayer
    cp.llm_factory._mock_factory = lambda llm: ContactCaller()
    cp.start()
    try:
        s = cp.store
        s.create(make_resource(SECRET, "hl", spec={"data": {"k": "hl-x"}}, api_version="v1"))
        s.create(
            make_resource(
                CONTACT_CHANNEL,
                "oncall",
                spec={
                    "type": "slack",
                    "apiKeyFrom": {"secretKeyRef": {"name": "hl", "key": "k"}},
                    "slack": {"channelOrUserID": "C1", "contextAboutChannelOrUser": "the on-call human"},
                },
            )
        )
        s.create(make_resource(LLM, "l", spec={"provider": "mock"}))
        s.create(
            make_resource(
                AGENT,
                "a",
                spec={
                    "llmRef": {"name": "l"},
                    "system": "s",
                    "humanContactChannels": [{"name": "oncall"}],
                },
            )
        )
        s.create(make_resource(TASK, "t1", spec={"agentRef": {"name": "a"}, "userMessage": "go"}))
        task = wait_for(
            lambda: (s.get(TASK, "t1") or {}).get("status", {}).get("phase")
            == TaskPhase.FINAL_ANSWER
            and s.get(TASK, "t1"),
            timeout=40,
        )
        tool_msgs = [m for m in task["status"]["contextWindow"] if m["role"] == "tool"]
        assert tool_msgs[0]["content"] == "yes, proceed"
        tcs = s.list(TOOL_CALL, label_selector={"acp.humanlayer.dev/task": "t1"})
        events = [e["reason"] for e in s.events_for(tcs[0]["metadata"]["name"])]
        assert "AwaitingHumanInput" in events
        assert task["status"]["output"] == "done after human input"
    finally:
        cp.stop()


def test_mcp_stdio_reconnect_after_crash():
    """The connection-maintenance loop reconnects a dead stdio server and
    republishes tools (reference maintainConnection, state_machine.go:173-213)."""
    cp = ControlPlane(auto_approve="approve")
    # fast re-list for the test
    for rec in cp.manager._reconcilers:
        if rec.kind == MCP_SERVER:
            rec.relist_period = 0.5
    cp.start()
    try:
        s = cp.store
        s.create(
            make_resource(
                MCP_SERVER,
                "calc",
                spec={
                    "transport": "stdio",
                    "command": sys.executable,
                    "args": ["-m", "agentcontrolplane_amd.mcp.echo_server"],
                },
            )
        )
        wait_for(
            lambda: (s.get(MCP_SERVER, "calc") or {}).get("status", {}).get("connected")
        )
        conn = cp.mcp.get_connection("calc")
        pid1 = conn.client.proc.pid
        conn.client.proc.kill()  # the MCP server dies
        # the maintenance pass must reconnect with a fresh subprocess
        def reconnected():
            c = cp.mcp.get_connection("calc")
            return c is not None and c.client.proc.poll() is None and c.client.proc.pid != pid1

        wait_for(reconnected, timeout=30)
        assert cp.mcp.call_tool("calc", "add", {"a": 1, "b": 2}) == "3.0"
        srv = s.get(MCP_SERVER, "calc")
        assert srv["status"]["connected"] is True
    finally:
        cp.stop()
