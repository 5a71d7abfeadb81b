#!/usr/bin/env python3
"""Programmatic demo (reference: acp/examples/simple_agent.go): build an
agent + task in-process and watch the loop run.

    python examples/simple_agent.py            # mock LLM (no GPU needed)
    python examples/simple_agent.py --local    # in-process engine
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from agentcontrolplane_amd.api.types import (  # noqa: E402
    AGENT, LLM, MCP_SERVER, TASK, make_resource,
)
from agentcontrolplane_amd.runtime import ControlPlane  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--local", action="store_true", help="use the in-process engine")
    args = p.parse_args()

    engine = None
    if args.local:
        import torch

        from agentcontrolplane_amd.engine.config import EngineConfig
        from agentcontrolplane_amd.engine.engine import InferenceEngine

        device = "cuda" if torch.cuda.is_available() else "cpu"
        engine = InferenceEngine(
            EngineConfig(model="llama3-8b" if device == "cuda" else "tiny", device=device)
        )

    cp = ControlPlane(engine=engine, auto_approve="approve").start()
    try:
        provider = "local" if args.local else "mock"
        cp.store.create(
            make_resource(LLM, "demo-llm", spec={"provider": provider,
                                                 "parameters": {"model": "llama3-8b", "maxTokens": 64}})
        )
        cp.mcp.register_inproc("calc", {"add": lambda a=0, b=0, **_: str(float(a) + float(b))})
        cp.store.create(make_resource(MCP_SERVER, "calc", spec={"transport": "inproc"}))
        cp.store.create(
            make_resource(
                AGENT, "demo-agent",
                spec={"llmRef": {"name": "demo-llm"}, "system": "You are a calculator agent.",
                      "mcpServers": [{"name": "calc"}]},
            )
        )
        cp.store.create(
            make_resource(TASK, "demo-task",
                          spec={"agentRef": {"name": "demo-agent"}, "userMessage": "add 2 and 40"})
        )
        print("watching demo-task ...")
        for _ in range(240):
            t = cp.store.get(TASK, "demo-task")
            phase = t["status"].get("phase", "")
            print(f"  phase={phase:<18} detail={t['status'].get('statusDetail', '')}")
            if phase in ("FinalAnswer", "Failed"):
                break
            time.sleep(0.5)
        print("\ncontext window:")
        for m in t["status"].get("contextWindow", []):
            calls = [tc["function"]["name"] for tc in m.get("toolCalls", [])]
            print(f"  [{m['role']}] {m.get('content', '')[:70]!r} {('tools: ' + str(calls)) if calls else ''}")
        print("\nevents:")
        for e in cp.store.events_for("demo-task"):
            print(f"  {e['reason']}: {e['message'][:70]}")
    finally:
        cp.stop()
        if engine is not None:
            engine.stop()


if __name__ == "__main__":
    main()
