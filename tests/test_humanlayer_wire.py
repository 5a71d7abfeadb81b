"""HumanLayer HTTP wire client + ContactChannel HTTP verification.

An in-process mock HumanLayer server (the reference's httptest pattern,
contactchannel_controller_test.go:63) backs: request/poll round-trips for
approvals and human contacts, project-key validation with project/org
slugs, channel-specific verification, and the full approval-gate e2e
through the HTTP path (VERDICT item 8's done-criterion)."""
from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from agentcontrolplane_amd.api.types import (
    AGENT,
    CONTACT_CHANNEL,
    LLM,
    MCP_SERVER,
    SECRET,
    TASK,
    TaskPhase,
    make_resource,
)
from agentcontrolplane_amd.humanlayer.wire import (
    HTTPHumanLayerClient,
    HTTPHumanLayerClientFactory,
    HumanLayerAPIError,
    verify_api_key_http,
)
from conftest import wait_for


class _MockHumanLayer(BaseHTTPRequestHandler):
    """State lives on the server instance: .calls, .contacts, .auto."""

    def _auth_ok(self):
        return self.headers.get("Authorization", "").startswith("Bearer hl-")

    def _json(self, code, obj):
        data = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_GET(self):  # noqa: N802
        if not self._auth_ok():
            self._json(401, {"error": "unauthorized"})
            return
        if self.path == "/humanlayer/v1/project":
            self._json(200, {"project_slug": "proj-x", "org_slug": "org-y"})
        elif self.path.startswith("/humanlayer/v1/contact_channel/"):
            cid = self.path.rsplit("/", 1)[1]
            if cid == "chan-ok":
                self._json(200, {"id": cid, "type": "slack"})
            else:
                self._json(404, {"error": "not found"})
        elif self.path.startswith("/humanlayer/v1/function_calls/"):
            cid = self.path.rsplit("/", 1)[1]
            fc = self.server.calls.get(cid)
            if fc is None:
                self._json(404, {"error": "not found"})
            else:
                self._json(200, fc)
        elif self.path.startswith("/humanlayer/v1/contact_requests/"):
            cid = self.path.rsplit("/", 1)[1]
            hc = self.server.contacts.get(cid)
            if hc is None:
                self._json(404, {"error": "not found"})
            else:
                self._json(200, hc)
        else:
            self._json(404, {"error": self.path})

    def do_POST(self):  # noqa: N802
        if not self._auth_ok():
            self._json(401, {"error": "unauthorized"})
            return
        n = int(self.headers.get("Content-Length", 0))
        body = json.loads(self.rfile.read(n) or b"{}")
        if self.path == "/humanlayer/v1/function_calls":
            cid = body["call_id"]
            fc = {"run_id": body.get("run_id", ""), "call_id": cid,
                  "spec": body.get("spec", {}),
                  "status": {"requested_at": 1.0}}
            if self.server.auto == "approve":
                fc["status"].update({"responded_at": 2.0, "approved": True,
                                     "comment": "lgtm"})
            elif self.server.auto == "reject":
                fc["status"].update({"responded_at": 2.0, "approved": False,
                                     "comment": "nope"})
            self.server.calls[cid] = fc
            self._json(201, fc)
        elif self.path == "/humanlayer/v1/contact_requests":
            cid = body["call_id"]
            hc = {"run_id": body.get("run_id", ""), "call_id": cid,
                  "spec": body.get("spec", {}),
                  "status": {"requested_at": 1.0}}
            if self.server.auto_response is not None:
                hc["status"].update({"responded_at": 2.0,
                                     "response": self.server.auto_response})
            self.server.contacts[cid] = hc
            self._json(201, hc)
        else:
            self._json(404, {"error": self.path})

    def log_message(self, *a):
        pass


@pytest.fixture()
def hl_server():
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _MockHumanLayer)
    srv.calls, srv.contacts = {}, {}
    srv.auto, srv.auto_response = None, None
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    srv.base = f"http://127.0.0.1:{srv.server_address[1]}"
    yield srv
    srv.shutdown()


def test_approval_roundtrip(hl_server):
    c = HTTPHumanLayerClient(api_base=hl_server.base, run_id="task-1",
                             api_key="hl-key",
                             channel={"slack": {"channelOrUserID": "C123"}})
    cid = c.request_approval("tools__rm", '{"path": "/tmp/x"}')
    # channel travels inside the spec (hlclient.go:149-166)
    sent = hl_server.calls[cid]["spec"]
    assert sent["fn"] == "tools__rm" and sent["kwargs"] == {"path": "/tmp/x"}
    assert sent["channel"]["slack"]["channel_or_user_id"] == "C123"
    st = c.get_function_call_status(cid)
    assert st.approved is None  # pending
    hl_server.calls[cid]["status"].update(
        {"responded_at": 2.0, "approved": True, "comment": "ok"})
    st = c.get_function_call_status(cid)
    assert st.approved is True and st.comment == "ok"


def test_channel_specific_auth_omits_channel(hl_server):
    c = HTTPHumanLayerClient(api_base=hl_server.base, api_key="hl-key",
                             channel={"slack": {"channelOrUserID": "C1"}},
                             channel_id="chan-ok")
    cid = c.request_approval("f", "{}")
    assert "channel" not in hl_server.calls[cid]["spec"]


def test_human_contact_roundtrip(hl_server):
    hl_server.auto_response = "go ahead"
    c = HTTPHumanLayerClient(api_base=hl_server.base, api_key="hl-key")
    cid = c.request_human_contact("should I?")
    st = c.get_human_contact_status(cid)
    assert st.response == "go ahead"
    assert hl_server.contacts[cid]["spec"]["msg"] == "should I?"


def test_auth_error_typed(hl_server):
    c = HTTPHumanLayerClient(api_base=hl_server.base, api_key="bad")
    with pytest.raises(HumanLayerAPIError) as ei:
        c.request_approval("f", "{}")
    assert ei.value.status_code == 401


def test_verify_api_key_http(hl_server):
    slugs = verify_api_key_http("hl-good", api_base=hl_server.base)
    assert slugs == {"projectSlug": "proj-x", "orgSlug": "org-y"}
    with pytest.raises(PermissionError):
        verify_api_key_http("wrong-prefix", api_base=hl_server.base)
    # channel-specific auth verifies the channel exists
    slugs = verify_api_key_http("hl-good", channel_id="chan-ok",
                                api_base=hl_server.base)
    assert slugs["projectSlug"] == "proj-x"
    with pytest.raises(LookupError):
        verify_api_key_http("hl-good", channel_id="chan-missing",
                            api_base=hl_server.base)


def test_contactchannel_controller_verifies_over_http(hl_server, monkeypatch):
    """ContactChannel reconcile → HTTP verification → slugs in status."""
    monkeypatch.setenv("HUMANLAYER_API_BASE", hl_server.base)
    from agentcontrolplane_amd.runtime import ControlPlane

    cp = ControlPlane(auto_approve="approve", llm_probe=False)
    cp.start()
    try:
        cp.store.create(make_resource(SECRET, "hl-secret",
                                      spec={"data": {"k": "hl-live"}},
                                      api_version="v1"))
        cp.store.create(make_resource(CONTACT_CHANNEL, "ch-http", spec={
            "type": "slack",
            "apiKeyFrom": {"secretKeyRef": {"name": "hl-secret", "key": "k"}},
            "slack": {"channelOrUserID": "C42"},
        }))
        ch = wait_for(lambda: (cp.store.get(CONTACT_CHANNEL, "ch-http") or {})
                      .get("status", {}).get("ready")
                      and cp.store.get(CONTACT_CHANNEL, "ch-http"), timeout=15)
        assert ch["status"]["projectSlug"] == "proj-x"
        assert ch["status"]["orgSlug"] == "org-y"
        # invalid key → Error status
        cp.store.create(make_resource(SECRET, "bad-secret",
                                      spec={"data": {"k": "nothl"}},
                                      api_version="v1"))
        cp.store.create(make_resource(CONTACT_CHANNEL, "ch-bad", spec={
            "type": "slack",
            "apiKeyFrom": {"secretKeyRef": {"name": "bad-secret", "key": "k"}},
            "slack": {"channelOrUserID": "C43"},
        }))
        bad = wait_for(lambda: (cp.store.get(CONTACT_CHANNEL, "ch-bad") or {})
                       .get("status", {}).get("status") == "Error"
                       and cp.store.get(CONTACT_CHANNEL, "ch-bad"), timeout=15)
        assert "invalid" in bad["status"]["statusDetail"].lower()
    finally:
        cp.stop()


def test_approval_gate_e2e_through_http(hl_server):
    """The full approval-gated tool call resolves through the HTTP client:
    MCP tool with approvalContactChannel → RequestApproval on the mock
    server (auto-approving) → ToolCall executes → FinalAnswer."""
    hl_server.auto = "approve"
    from agentcontrolplane_amd.llmclient.mock import MockLLMClient
    from agentcontrolplane_amd.llmclient.factory import LLMClientFactory
    from agentcontrolplane_amd.runtime import ControlPlane

    factory = HTTPHumanLayerClientFactory(api_base=hl_server.base,
                                          api_key="hl-key")
    cp = ControlPlane(
        humanlayer_factory=factory,
        llm_client_factory=LLMClientFactory(mock_factory=lambda llm: MockLLMClient()),
        llm_probe=False,
    )
    cp.start()
    try:
        cp.store.create(make_resource(SECRET, "hl-secret",
                                      spec={"data": {"k": "hl-key"}},
                                      api_version="v1"))
        cp.store.create(make_resource(CONTACT_CHANNEL, "approvals", spec={
            "type": "slack",
            "apiKeyFrom": {"secretKeyRef": {"name": "hl-secret", "key": "k"}},
            "slack": {"channelOrUserID": "CAPPROVE"},
        }))
        cp.store.create(make_resource(LLM, "mock-llm", spec={"provider": "mock"}))
        cp.mcp.register_inproc("tools", {"add": lambda a=0, b=0, **_: str(a + b)})
        cp.store.create(make_resource(MCP_SERVER, "tools", spec={
            "transport": "inproc",
            "approvalContactChannel": {"name": "approvals"},
        }))
        cp.store.create(make_resource(AGENT, "gated-agent", spec={
            "llmRef": {"name": "mock-llm"},
            "system": "sys",
            "mcpServers": [{"name": "tools"}],
        }))
        wait_for(lambda: (cp.store.get(AGENT, "gated-agent") or {})
                 .get("status", {}).get("ready"), timeout=15)
        cp.store.create(make_resource(TASK, "gated-task", spec={
            "agentRef": {"name": "gated-agent"},
            "userMessage": "use the add tool",
        }))
        task = wait_for(
            lambda: (cp.store.get(TASK, "gated-task") or {}).get("status", {})
            .get("phase") == TaskPhase.FINAL_ANSWER
            and cp.store.get(TASK, "gated-task"),
            timeout=30,
        )
        # the approval went over the wire to the mock HumanLayer server
        assert hl_server.calls, "no RequestApproval reached the mock server"
        fc = next(iter(hl_server.calls.values()))
        assert fc["status"]["approved"] is True
        assert any(m.get("role") == "tool" for m in task["status"]["contextWindow"])
    finally:
        cp.stop()


def test_rejection_feeds_back_through_http(hl_server):
    """A rejected approval must NOT fail the ToolCall: the reference marks
    it Succeeded with result "Rejected: <comment>" so the refusal feeds
    back into the next LLM turn (toolcall/state_machine.go:153-160) —
    here through the real HTTP client against the mock server."""
    hl_server.auto = "reject"
    from agentcontrolplane_amd.llmclient.mock import MockLLMClient
    from agentcontrolplane_amd.llmclient.factory import LLMClientFactory
    from agentcontrolplane_amd.runtime import ControlPlane

    factory = HTTPHumanLayerClientFactory(api_base=hl_server.base,
                                          api_key="hl-key")
    cp = ControlPlane(
        humanlayer_factory=factory,
        llm_client_factory=LLMClientFactory(mock_factory=lambda llm: MockLLMClient()),
        llm_probe=False,
    )
    cp.start()
    try:
        cp.store.create(make_resource(SECRET, "hl-secret",
                                      spec={"data": {"k": "hl-key"}},
                                      api_version="v1"))
        cp.store.create(make_resource(CONTACT_CHANNEL, "approvals", spec={
            "type": "slack",
            "apiKeyFrom": {"secretKeyRef": {"name": "hl-secret", "key": "k"}},
            "slack": {"channelOrUserID": "CREJECT"},
        }))
        cp.store.create(make_resource(LLM, "mock-llm", spec={"provider": "mock"}))
        cp.mcp.register_inproc("tools", {"add": lambda a=0, b=0, **_: str(a + b)})
        cp.store.create(make_resource(MCP_SERVER, "tools", spec={
            "transport": "inproc",
            "approvalContactChannel": {"name": "approvals"},
        }))
        cp.store.create(make_resource(AGENT, "rej-agent", spec={
            "llmRef": {"name": "mock-llm"}, "system": "sys",
            "mcpServers": [{"name": "tools"}],
        }))
        wait_for(lambda: (cp.store.get(AGENT, "rej-agent") or {})
                 .get("status", {}).get("ready"), timeout=15)
        cp.store.create(make_resource(TASK, "rej-task", spec={
            "agentRef": {"name": "rej-agent"},
            "userMessage": "use the add tool",
        }))
        task = wait_for(
            lambda: (cp.store.get(TASK, "rej-task") or {}).get("status", {})
            .get("phase") == TaskPhase.FINAL_ANSWER
            and cp.store.get(TASK, "rej-task"),
            timeout=30,
        )
        # the rejection reached the context window as a tool result
        tool_msgs = [m for m in task["status"]["contextWindow"]
                     if m.get("role") == "tool"]
        assert tool_msgs and tool_msgs[0]["content"].startswith("Rejected")
        fc = next(iter(hl_server.calls.values()))
        assert fc["status"]["approved"] is False
    finally:
        cp.stop()
