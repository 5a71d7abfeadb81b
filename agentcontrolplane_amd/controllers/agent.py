"""Agent reconciler.

Parity with acp/internal/controller/agent/state_machine.go (307 LoC):
validate the LLM ref, MCP servers, contact channels and sub-agents, and
publish ``validMCPServers`` / ``validHumanContactChannels`` /
``validSubAgents`` in status.  A missing dependency is a terminal Error;
a present-but-not-ready dependency is Pending with a 5 s requeue
(state_machine.go:219-232).
"""
from __future__ import annotations

from ..api.types import AGENT, CONTACT_CHANNEL, LLM, MCP_SERVER
from .manager import Reconciler, Result

PENDING_REQUEUE = 5.0


class AgentReconciler(Reconciler):
    kind = AGENT
    workers = 2

    def reconcile(self, name: str, namespace: str) -> Result:
        agent = self.store.get(AGENT, name, namespace)
        if agent is None:
            return Result()
        spec = agent.get("spec", {})
        status = agent.setdefault("status", {})

        def set_state(state: str, detail: str, reason: str = "", event: str = "Normal",
                      requeue: float = 0.0) -> Result:
            status.update(
                {"ready": state == "Ready", "status": state, "statusDetail": detail}
            )
            if reason:
                self.store.record_event(agent, event, reason, detail)
            self.store.update_status(agent)
            return Result(requeue_after=requeue)

        # LLM (state_machine.go:207-216)
        llm_name = (spec.get("llmRef") or {}).get("name", "")
        llm = self.store.get(LLM, llm_name, namespace)
        if llm is None:
            return set_state("Error", f'LLM "{llm_name}" not found', "ValidationFailed", "Warning")
        if not llm.get("status", {}).get("ready", False):
            return set_state(
                "Pending", f'LLM "{llm_name}" is not ready', "Waiting", requeue=PENDING_REQUEUE
            )

        # MCP servers (235-257)
        valid_mcp = []
        for ref in spec.get("mcpServers", []) or []:
            srv = self.store.get(MCP_SERVER, ref["name"], namespace)
            if srv is None:
                return set_state(
                    "Error", f'MCPServer "{ref["name"]}" not found', "ValidationFailed", "Warning"
                )
            if not srv.get("status", {}).get("connected", False):
                return set_state(
                    "Pending",
                    f'MCPServer "{ref["name"]}" is not connected',
                    "Waiting",
                    requeue=PENDING_REQUEUE,
                )
            tools = [t.get("name", "") for t in srv.get("status", {}).get("tools", []) or []]
            valid_mcp.append({"name": ref["name"], "tools": tools})

        # contact channels
        valid_channels = []
        for ref in spec.get("humanContactChannels", []) or []:
            ch = self.store.get(CONTACT_CHANNEL, ref["name"], namespace)
            if ch is None:
                return set_state(
                    "Error",
                    f'ContactChannel "{ref["name"]}" not found',
                    "ValidationFailed",
                    "Warning",
                )
            if not ch.get("status", {}).get("ready", False):
                return set_state(
                    "Pending",
                    f'ContactChannel "{ref["name"]}" is not ready',
                    "Waiting",
                    requeue=PENDING_REQUEUE,
                )
            valid_channels.append({"name": ref["name"], "type": ch.get("spec", {}).get("type", "")})

        # sub-agents (219-232): pending + requeue when not ready
        valid_sub = []
        for ref in spec.get("subAgents", []) or []:
            sub = self.store.get(AGENT, ref["name"], namespace)
            if sub is None:
                return set_state(
                    "Error", f'sub-agent "{ref["name"]}" not found', "ValidationFailed", "Warning"
                )
            if not sub.get("status", {}).get("ready", False):
                return set_state(
                    "Pending",
                    f'sub-agent "{ref["name"]}" is not ready',
                    "Waiting",
                    requeue=PENDING_REQUEUE,
                )
            valid_sub.append({"name": ref["name"]})

        status["validMCPServers"] = valid_mcp
        status["validHumanContactChannels"] = valid_channels
        status["validSubAgents"] = valid_sub
        return set_state("Ready", "All dependencies validated", "ValidationSucceeded")
