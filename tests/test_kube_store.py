"""KubeStore against a mock Kubernetes apiserver.

Proves the north-star deployment path: CRs created through the Kubernetes
REST wire (what `kubectl apply` does) drive the controllers unchanged,
with the REFERENCE's own sample manifests applied byte-for-byte and
validated against this repo's generated CRDs (config/crd/bases/)."""
from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest
import yaml

from agentcontrolplane_amd.api.types import TASK, TaskPhase
from agentcontrolplane_amd.store import ConflictError
from agentcontrolplane_amd.store.kube import KubeStore
from conftest import wait_for
from mock_apiserver import MockAPIServer

REF_SAMPLES = "/root/reference/acp/config/samples"


@pytest.fixture()
def apiserver():
    srv = MockAPIServer()
    yield srv
    srv.shutdown()


@pytest.fixture()
def kube(apiserver):
    ks = KubeStore(base_url=apiserver.base)
    yield ks
    ks.close()


def test_crud_roundtrip_and_conflicts(kube):
    obj = kube.create({
        "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "Agent",
        "metadata": {"name": "a1"},
        "spec": {"llmRef": {"name": "l1"}, "system": "sys"},
    })
    assert obj["metadata"]["resourceVersion"]
    got = kube.get("Agent", "a1")
    assert got["spec"]["system"] == "sys"
    # status subresource: spec PUT does not touch status and vice versa
    got["status"] = {"ready": True, "status": "Ready"}
    kube.update_status(got)
    fresh = kube.get("Agent", "a1")
    assert fresh["status"]["ready"] is True
    # stale resourceVersion -> ConflictError (apiserver 409 semantics)
    stale = dict(got)
    stale["metadata"] = dict(got["metadata"])
    kube.update_status(fresh)  # no-op write does not bump rv
    fresh["status"] = {"ready": False, "status": "Error", "statusDetail": "x"}
    kube.update_status(fresh)
    with pytest.raises(ConflictError):
        stale["status"] = {"ready": True, "status": "Pending"}
        kube.update_status(stale)
    assert kube.delete("Agent", "a1") is True
    assert kube.get("Agent", "a1") is None


def test_label_selector_list(kube):
    for i, lbl in enumerate(["x", "x", "y"]):
        kube.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "ToolCall",
            "metadata": {"name": f"tc{i}", "labels": {"acp.humanlayer.dev/task": lbl}},
            "spec": {"taskRef": {"name": "t"}, "toolCallId": "c", "toolRef":
                     {"name": "n"}, "arguments": "{}"},
        })
    out = kube.list("ToolCall", label_selector={"acp.humanlayer.dev/task": "x"})
    assert {o["metadata"]["name"] for o in out} == {"tc0", "tc1"}


def test_crd_validation_rejects_bad_spec(kube):
    with pytest.raises(RuntimeError) as ei:
        kube.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "Agent",
            "metadata": {"name": "bad"},
            "spec": {"system": "no llmRef"},
        })
    assert "llmRef" in str(ei.value)
    with pytest.raises(RuntimeError):
        kube.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "ContactChannel",
            "metadata": {"name": "bad2"},
            "spec": {"type": "carrier-pigeon"},  # not in the enum
        })


def test_watch_stream(kube):
    q = kube.watch(kinds={"Task"})
    kube.create({
        "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "Task",
        "metadata": {"name": "t1"},
        "spec": {"agentRef": {"name": "a"}, "userMessage": "hi"},
    })
    ev = q.get(timeout=5)
    assert ev.type == "ADDED" and ev.obj["metadata"]["name"] == "t1"


def test_lease_semantics(kube):
    assert kube.acquire_lease("task-llm-x", "pod-a", 30) is True
    assert kube.acquire_lease("task-llm-x", "pod-b", 30) is False  # held
    assert kube.acquire_lease("task-llm-x", "pod-a", 30) is True   # renew
    kube.release_lease("task-llm-x", "pod-a")
    assert kube.acquire_lease("task-llm-x", "pod-b", 30) is True


def test_reference_samples_validate(apiserver, kube):
    """Reference sample manifests apply byte-for-byte against our CRDs."""
    import glob
    import os

    applied = 0
    for path in sorted(glob.glob(os.path.join(REF_SAMPLES, "*.yaml"))):
        if path.endswith("kustomization.yaml"):
            continue
        with open(path) as f:
            for doc in yaml.safe_load_all(f):
                if not doc or doc.get("kind") not in (
                    "LLM", "Agent", "Task", "ToolCall", "MCPServer", "ContactChannel"
                ):
                    continue
                try:
                    kube.create(doc)
                except Exception as e:
                    # duplicate names across sample files are fine
                    assert "already exists" in str(e), e
                    continue
                applied += 1
    assert applied >= 8  # agents, tasks, channels, llm, mcpservers...
    # and they landed with server-assigned metadata
    assert kube.get("Agent", "web-fetch-agent")["metadata"]["uid"]


# ------------------------------------------------------------ full e2e


class _MockOpenAI(BaseHTTPRequestHandler):
    def do_POST(self):  # noqa: N802
        n = int(self.headers.get("Content-Length", 0))
        body = json.loads(self.rfile.read(n) or b"{}")
        msgs = body.get("messages", [])
        tools = body.get("tools", []) or []
        has_tool_result = any(m.get("role") == "tool" for m in msgs)
        add = next((t for t in tools
                    if t["function"]["name"].endswith("__add")), None)
        if add and not has_tool_result and body.get("max_tokens") != 1:
            reply = {"choices": [{"message": {
                "content": None,
                "tool_calls": [{"id": "call_k8s1", "type": "function",
                                "function": {"name": add["function"]["name"],
                                             "arguments": '{"a": 2, "b": 3}'}}]}}]}
        else:
            reply = {"choices": [{"message": {"content": "k8s final answer"}}]}
        data = json.dumps(reply).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def log_message(self, *a):
        pass


def test_kubectl_apply_e2e(apiserver):
    """Reference Agent + Task sample YAML applied unchanged through the
    Kubernetes wire reach FinalAnswer (VERDICT item 2's done-criterion,
    with the mock apiserver standing in for kind)."""
    import os

    oai = ThreadingHTTPServer(("127.0.0.1", 0), _MockOpenAI)
    threading.Thread(target=oai.serve_forever, daemon=True).start()
    oai_base = f"http://127.0.0.1:{oai.server_address[1]}/v1"

    from agentcontrolplane_amd.runtime import ControlPlane

    store = KubeStore(base_url=apiserver.base)
    cp = ControlPlane(store=store)
    try:
        # the LLM the samples reference, pointed at the mock server the way
        # the reference's own e2e does (test_getting_started.go:250-261)
        store.create({
            "apiVersion": "v1", "kind": "Secret", "metadata": {"name": "openai"},
            "spec": {"data": {"OPENAI_API_KEY": "sk-mock"}},
        })
        store.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "LLM",
            "metadata": {"name": "gpt-4o"},
            "spec": {"provider": "openai",
                     "apiKeyFrom": {"secretKeyRef": {"name": "openai",
                                                     "key": "OPENAI_API_KEY"}},
                     "parameters": {"model": "gpt-4o", "baseUrl": oai_base}},
        })
        # the MCP server the agent sample references — stdio transport with
        # the in-repo echo server (the sample's uvx fetch server needs
        # network; same substitution the reference CI makes)
        import sys

        store.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "MCPServer",
            "metadata": {"name": "fetch-server"},
            "spec": {"transport": "stdio", "command": sys.executable,
                     "args": ["-m", "agentcontrolplane_amd.mcp.echo_server"]},
        })
        cp.start()
        # reference sample manifests, byte-for-byte
        for fname in ("acp_v1alpha1_agent.yaml", "acp_v1alpha1_task.yaml"):
            with open(os.path.join(REF_SAMPLES, fname)) as f:
                for doc in yaml.safe_load_all(f):
                    if doc:
                        store.create(doc)
        task_name = "fetch-example"  # acp_v1alpha1_task.yaml metadata.name
        task = wait_for(
            lambda: (store.get(TASK, task_name) or {}).get("status", {})
            .get("phase") == TaskPhase.FINAL_ANSWER
            and store.get(TASK, task_name),
            timeout=60,
        )
        assert task["status"]["output"] == "k8s final answer"
        tool_msgs = [m for m in task["status"]["contextWindow"]
                     if m.get("role") == "tool"]
        assert tool_msgs and tool_msgs[0]["content"] == "5.0"
    finally:
        cp.stop()
        store.close()
        oai.shutdown()


def test_leader_election_failover():
    """Two replicas over one shared store: only the leader reconciles;
    when it stops, the follower takes over (cmd/main.go:208-226)."""
    import time

    from agentcontrolplane_amd.runtime import ControlPlane
    from agentcontrolplane_amd.store import ResourceStore

    shared = ResourceStore()
    a = ControlPlane(store=shared, pod_name="pod-a", llm_probe=False)
    b = ControlPlane(store=shared, pod_name="pod-b", llm_probe=False)
    # fast failover for the test
    for cp in (a, b):
        cp.LEADER_LEASE_DURATION = 1.0
        cp.LEADER_RETRY = 0.1
    a.start(leader_elect=True)
    wait_for(lambda: a.is_leader, timeout=15)
    b.start(leader_elect=True)
    time.sleep(0.5)
    assert a.is_leader and not b.is_leader
    # the leader reconciles; the follower does not
    shared.create({"apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "LLM",
                   "metadata": {"name": "l1"}, "spec": {"provider": "mock"},
                   "status": {}})
    wait_for(lambda: (shared.get("LLM", "l1") or {}).get("status", {}).get("ready"),
             timeout=10)
    a.stop()  # releases the lease
    wait_for(lambda: b.is_leader, timeout=10)
    shared.create({"apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "LLM",
                   "metadata": {"name": "l2"}, "spec": {"provider": "mock"},
                   "status": {}})
    wait_for(lambda: (shared.get("LLM", "l2") or {}).get("status", {}).get("ready"),
             timeout=10)
    b.stop()


def test_watch_survives_apiserver_restart():
    """The per-kind watch threads re-list and reconnect after the
    apiserver drops (KubeStore._watch_loop's 410/error path) — events for
    objects created after the restart still arrive."""
    import queue as _q
    import time

    srv = MockAPIServer()
    port = srv._srv.server_address[1]
    ks = KubeStore(base_url=srv.base)
    try:
        q = ks.watch(kinds={"Task"})
        ks.create({
            "apiVersion": "acp.humanlayer.dev/v1alpha1", "kind": "Task",
            "metadata": {"name": "before"},
            "spec": {"agentRef": {"name": "a"}, "userMessage": "x"},
        })
        ev = q.get(timeout=5)
        assert ev.obj["metadata"]["name"] == "before"
        # kill the apiserver; bring a fresh one up on the SAME port
        srv.shutdown()
        time.sleep(0.3)
        srv2 = MockAPIServer(port=port)
        try:
            # wait for the store's reconnect loop to re-establish, then
            # a new object's ADDED must flow through the same queue
            deadline = time.monotonic() + 15
            created = False
            seen = False
            while time.monotonic() < deadline and not seen:
                if not created:
                    try:
                        ks.create({
                            "apiVersion": "acp.humanlayer.dev/v1alpha1",
                            "kind": "Task", "metadata": {"name": "after"},
                            "spec": {"agentRef": {"name": "a"},
                                     "userMessage": "y"},
                        })
                        created = True
                    except Exception:
                        time.sleep(0.2)  # server not up yet
                        continue
                try:
                    ev = q.get(timeout=1.0)
                except _q.Empty:
                    continue
                if ev.obj.get("metadata", {}).get("name") == "after":
                    seen = True
            assert seen, "watch did not recover after apiserver restart"
        finally:
            srv2.shutdown()
    finally:
        ks.close()
