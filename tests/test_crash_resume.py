"""Crash-resume: the core durability design (SURVEY.md §5 checkpoint/resume).

The control plane is killed mid-agent-loop; a new one starts from the WAL
and every task finishes from its checkpointed phase + context window —
"async/await at the infrastructure layer"."""
import time

import pytest

from agentcontrolplane_amd.api.types import (
    AGENT,
    LLM,
    MCP_SERVER,
    TASK,
    TaskPhase,
    make_resource,
)
from agentcontrolplane_amd.llmclient.mock import MockLLMClient
from agentcontrolplane_amd.runtime import ControlPlane

from conftest import wait_for


class SlowMock(MockLLMClient):
    def __init__(self):
        super().__init__(latency_s=0.4)


def _world(cp):
    s = cp.store
    if s.get(LLM, "l") is None:
        s.create(make_resource(LLM, "l", spec={"provider": "mock"}))
        cp.mcp.register_inproc("tools", {"noop": lambda **_: "ok"})
        s.create(make_resource(MCP_SERVER, "tools", spec={"transport": "inproc"}))
        s.create(
            make_resource(
                AGENT, "a",
                spec={"llmRef": {"name": "l"}, "system": "sys",
                      "mcpServers": [{"name": "tools"}]},
            )
        )
    else:
        cp.mcp.register_inproc("tools", {"noop": lambda **_: "ok"})


def test_crash_mid_loop_resume_from_wal(tmp_path):
    wal = str(tmp_path / "acp.wal")
    cp1 = ControlPlane(
        wal_path=wal, fsync="always", auto_approve="approve",
        llm_client_factory=None, llm_probe=True,
    )
    cp1.llm_factory._mock_factory = lambda llm: SlowMock()
    cp1.start()
    _world(cp1)
    names = [f"t{i}" for i in range(6)]
    for n in names:
        cp1.store.create(
            make_resource(TASK, n, spec={"agentRef": {"name": "a"}, "userMessage": "go"})
        )
    # let the loop get partway (some tasks mid-phase), then crash hard:
    # no graceful drain — threads are daemonic, just drop the manager
    time.sleep(1.0)
    phases_at_crash = {
        n: (cp1.store.get(TASK, n) or {}).get("status", {}).get("phase") for n in names
    }
    cp1.manager.stop()
    cp1.store.close()
    assert any(p not in (TaskPhase.FINAL_ANSWER, None) for p in phases_at_crash.values()), (
        f"crash happened too late to be interesting: {phases_at_crash}"
    )

    # resurrect from the WAL
    cp2 = ControlPlane(wal_path=wal, auto_approve="approve")
    cp2.llm_factory._mock_factory = lambda llm: SlowMock()
    cp2.start()
    _world(cp2)
    try:
        for n in names:
            task = wait_for(
                lambda n=n: (cp2.store.get(TASK, n) or {}).get("status", {}).get("phase")
                == TaskPhase.FINAL_ANSWER
                and cp2.store.get(TASK, n),
                timeout=60,
            )
            cw = task["status"]["contextWindow"]
            # exactly one complete loop: no duplicated turns despite the crash
            finals = [m for m in cw if m["role"] == "assistant" and m.get("content")]
            assert len(finals) == 1, f"{n}: {cw}"
            tools = [m for m in cw if m["role"] == "tool"]
            assert len(tools) <= 1, f"{n}: duplicated tool turns {cw}"
    finally:
        cp2.stop()


def test_engine_error_mid_task_retries():
    """A transient engine failure (5xx-class) retries in place and the task
    still completes (reference handleLLMError semantics)."""
    from agentcontrolplane_amd.llmclient.base import LLMRequestError

    failed = {"done": False}

    class Flaky(MockLLMClient):
        def send_request(self, messages, tools):
            is_probe = len(messages) == 1 and messages[0].content == "ping"
            if not is_probe and not failed["done"]:
                failed["done"] = True  # fail exactly the first real task turn
                raise LLMRequestError(503, "transient engine failure")
            return super().send_request(messages, tools)

    cp = ControlPlane(auto_approve="approve")
    cp.llm_factory._mock_factory = lambda llm: Flaky()
    cp.start()
    try:
        _world(cp)
        cp.store.create(
            make_resource(TASK, "t1", spec={"agentRef": {"name": "a"}, "userMessage": "x"})
        )
        task = wait_for(
            lambda: (cp.store.get(TASK, "t1") or {}).get("status", {}).get("phase")
            == TaskPhase.FINAL_ANSWER
            and cp.store.get(TASK, "t1"),
            timeout=40,
        )
        reasons = [e["reason"] for e in cp.store.events_for("t1")]
        assert "LLMRequestFailed" in reasons  # the transient error was recorded
        assert task["status"]["output"] == "mock final answer"
    finally:
        cp.stop()
