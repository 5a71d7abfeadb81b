"""MCPServer reconciler.

Parity with acp/internal/controller/mcpserver/state_machine.go (315 LoC):
spec validation, env-var/secret resolution, connect through the shared
MCPServerManager, publish discovered tools in status, and a periodic
re-list/reconnect loop (10 min in the reference, state_machine.go:170;
configurable here).
"""
from __future__ import annotations

from ..api.types import CONTACT_CHANNEL, MCP_SERVER
from ..mcp.manager import MCPServerManager
from .manager import Reconciler, Result

RELIST_PERIOD = 600.0  # 10 min (mcpserver/state_machine.go:170)


class MCPServerReconciler(Reconciler):
    kind = MCP_SERVER
    workers = 2

    def __init__(self, store, mcp_manager: MCPServerManager, relist_period: float = RELIST_PERIOD):
        super().__init__(store)
        self.mcp = mcp_manager
        self.relist_period = relist_period

    def reconcile(self, name: str, namespace: str) -> Result:
        srv = self.store.get(MCP_SERVER, name, namespace)
        if srv is None:
            self.mcp.disconnect_server(name)
            return Result()
        spec = srv.get("spec", {})
        status = srv.setdefault("status", {})

        def fail(detail: str, requeue: float = 15.0) -> Result:
            status.update({"connected": False, "status": "Error", "statusDetail": detail})
            self.store.record_event(srv, "Warning", "ConnectionFailed", detail)
            self.store.update_status(srv)
            return Result(requeue_after=requeue)

        # spec validation (state_machine.go:85-120)
        transport = spec.get("transport", "")
        if transport not in ("stdio", "http", "inproc"):
            return fail(f"invalid transport {transport!r}", requeue=0)
        if transport == "stdio" and not spec.get("command"):
            return fail("stdio transport requires command", requeue=0)
        if transport == "http" and not spec.get("url"):
            return fail("http transport requires url", requeue=0)

        # approval contact channel reference must exist when set
        acc = spec.get("approvalContactChannel")
        if acc:
            ch = self.store.get(CONTACT_CHANNEL, acc.get("name", ""), namespace)
            if ch is None:
                return fail(f'approvalContactChannel "{acc.get("name")}" not found', requeue=10.0)
            if not ch.get("status", {}).get("ready", False):
                return fail(
                    f'approvalContactChannel "{acc.get("name")}" not ready', requeue=5.0
                )

        # connect (or maintain an existing connection; reconnect on tool drift —
        # maintainConnection/toolsChanged, state_machine.go:173-211)
        conn = self.mcp.get_connection(name)
        try:
            if conn is None:
                tools = self.mcp.connect_server(srv)
                self.store.record_event(srv, "Normal", "Connected", "MCP server connected")
            else:
                tools = conn.tools
                if conn.transport == "stdio":
                    result = conn.client.call("tools/list", {})
                    fresh = result.get("tools", []) if isinstance(result, dict) else []
                    if [t.get("name") for t in fresh] != [t.get("name") for t in tools]:
                        tools = fresh
                        conn.tools = fresh
                        self.store.record_event(
                            srv, "Normal", "ToolsChanged", "MCP server tool list changed"
                        )
        except Exception as e:
            self.mcp.disconnect_server(name)
            return fail(f"connect failed: {e}")

        status.update(
            {
                "connected": True,
                "status": "Ready",
                "statusDetail": f"Connected with {len(tools)} tools",
                "tools": tools,
            }
        )
        self.store.update_status(srv)
        return Result(requeue_after=self.relist_period)
