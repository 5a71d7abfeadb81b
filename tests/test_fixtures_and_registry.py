"""Fixture-builder + external registry tiers, and multi-replica lease
semantics (two managers sharing one store — the reference's multi-pod
locking, exercised rather than just unit-tested)."""
import time

from agentcontrolplane_amd.api.types import LLM, TASK, TaskPhase
from agentcontrolplane_amd.external_api import ExternalClient, Registry
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for
from utils import READY, TestAgent, TestLLM, TestSecret, TestTask


def test_fixture_builders(store):
    TestSecret("test-secret").setup(store)
    llm = TestLLM("test-llm").setup_with_status(store, READY)
    assert llm["status"]["ready"] is True
    agent = TestAgent("test-agent").setup_with_status(store, READY)
    task = TestTask("test-task").setup_with_status(
        store, {"phase": TaskPhase.READY_FOR_LLM, "contextWindow": [
            {"role": "system", "content": "s"}, {"role": "user", "content": "u"},
        ]}
    )
    assert task["status"]["phase"] == TaskPhase.READY_FOR_LLM
    TestTask("test-task").teardown(store)
    assert store.get(TASK, "test-task") is None


def test_external_registry():
    reg = Registry()

    class Echo(ExternalClient):
        def __init__(self, key):
            self.key = key

        def call(self, arguments):
            return f"{self.key}:{arguments.get('x')}"

    reg.register("echo_tool", lambda key: Echo(key))
    assert reg.has("echo_tool")
    c = reg.get_client("echo_tool", api_key="k1")
    assert c.call({"x": 1}) == "k1:1"
    assert reg.get_client("missing") is None
    try:
        reg.register("echo_tool", lambda key: Echo(key))
        raise AssertionError("duplicate registration must fail")
    except ValueError:
        pass


def test_two_replicas_share_one_store():
    """Two ControlPlane replicas (distinct pod names) over one store — the
    reference's multi-pod deployment against one etcd.  Both managers'
    reconcilers race on the same Tasks; the dual-layer locking (in-memory
    mutex is per-process, so the store lease does the cross-replica work)
    must keep the loop correct: tasks complete with exactly one final
    assistant message each."""
    cp1 = ControlPlane(auto_approve="approve", pod_name="pod-a")
    cp2 = ControlPlane(auto_approve="approve", pod_name="pod-b", store=cp1.store)
    cp1.start()
    cp2.start()
    try:
        s = cp1.store
        TestLLM("test-llm", {"provider": "mock"}).setup(s)
        TestAgent("test-agent", {"llmRef": {"name": "test-llm"}, "system": "x"}).setup(s)
        names = [f"t-shared-{i}" for i in range(6)]
        for n in names:
            TestTask(n).setup(s)
        for n in names:
            wait_for(
                lambda n=n: (s.get(TASK, n) or {}).get("status", {}).get("phase")
                == TaskPhase.FINAL_ANSWER,
                timeout=40,
            )
        for n in names:
            cw = s.get(TASK, n)["status"]["contextWindow"]
            finals = [m for m in cw if m["role"] == "assistant" and m.get("content")]
            assert len(finals) == 1, f"{n}: duplicated turn {cw}"
        # cross-holder lease exclusion on the same store
        assert s.acquire_lease("task-llm-x", "pod-a", 30)
        assert not s.acquire_lease("task-llm-x", "pod-b", 30)
        s.release_lease("task-llm-x", "pod-a")
        assert s.acquire_lease("task-llm-x", "pod-b", 30)
    finally:
        cp2.stop()
        cp1.stop()
