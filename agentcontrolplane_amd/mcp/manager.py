"""MCP server connection pool.

Parity with the reference's MCPServerManager (acp/internal/mcpmanager/
mcpmanager.go:24-341): a lock-guarded connection map, stdio (subprocess) and
http transports, tool discovery at connect, tool invocation with text-content
concatenation, env-var secret resolution, and a ``server__tool`` naming
convention for routing (mcpmanager.go:304-331).

The stdio transport speaks MCP JSON-RPC 2.0 over newline-delimited JSON on
the child's stdin/stdout (the mark3labs/mcp-go NewStdioMCPClient wire format).
An ``inproc`` transport binds python callables directly for tests and
benchmarks that do not want subprocess overhead in the measured loop.
"""
from __future__ import annotations

import json
import subprocess
import threading
import time
from typing import Any, Callable, Dict, List, Optional

from ..api.types import SECRET


class MCPError(RuntimeError):
    pass


class _StdioClient:
    """Newline-delimited JSON-RPC 2.0 client over a child process."""

    def __init__(self, command: str, args: List[str], env: Optional[Dict[str, str]] = None):
        import os

        full_env = dict(os.environ)
        if env:
            full_env.update(env)
        self.proc = subprocess.Popen(
            [command] + list(args),
            stdin=subprocess.PIPE,
            stdout=subprocess.PIPE,
            stderr=subprocess.DEVNULL,
            env=full_env,
            text=True,
            bufsize=1,
        )
        self._id = 0
        self._lock = threading.Lock()

    def call(self, method: str, params: Optional[Dict[str, Any]] = None, timeout: float = 30.0) -> Any:
        with self._lock:
            self._id += 1
            req = {"jsonrpc": "2.0", "id": self._id, "method": method}
            if params is not None:
                req["params"] = params
            try:
                self.proc.stdin.write(json.dumps(req) + "\n")
                self.proc.stdin.flush()
            except (BrokenPipeError, ValueError) as e:
                raise MCPError(f"stdio MCP server pipe closed: {e}")
            deadline = time.monotonic() + timeout
            while time.monotonic() < deadline:
                line = self.proc.stdout.readline()
                if not line:
                    raise MCPError("stdio MCP server closed stdout")
                line = line.strip()
                if not line:
                    continue
                try:
                    msg = json.loads(line)
                except json.JSONDecodeError:
                    continue
                if msg.get("id") == self._id:
                    if "error" in msg:
                        raise MCPError(str(msg["error"]))
                    return msg.get("result")
                # notifications are ignored
            raise MCPError(f"timeout waiting for {method}")

    def notify(self, method: str, params: Optional[Dict[str, Any]] = None) -> None:
        msg = {"jsonrpc": "2.0", "method": method}
        if params is not None:
            msg["params"] = params
        self.proc.stdin.write(json.dumps(msg) + "\n")
        self.proc.stdin.flush()

    def close(self) -> None:
        try:
            self.proc.terminate()
            self.proc.wait(timeout=2)
        except Exception:
            try:
                self.proc.kill()
            except Exception:
                pass


class MCPConnection:
    __slots__ = ("name", "transport", "client", "tools", "funcs", "connected_at")

    def __init__(self, name: str, transport: str, client=None, tools=None, funcs=None):
        self.name = name
        self.transport = transport
        self.client = client
        self.tools: List[Dict[str, Any]] = tools or []
        self.funcs: Dict[str, Callable] = funcs or {}
        self.connected_at = time.time()


class MCPServerManager:
    """Connection pool + tool router (mcpmanager.go:24-341)."""

    def __init__(self, store=None):
        self.store = store
        self._lock = threading.RLock()
        self._conns: Dict[str, MCPConnection] = {}
        # registry of in-process tool servers: name -> {tool: callable}
        self._inproc_registry: Dict[str, Dict[str, Callable]] = {}

    # ------------------------------------------------------------- lifecycle

    def register_inproc(self, server_name: str, tools: Dict[str, Callable]) -> None:
        """Register python callables as an in-process MCP server (tests/bench)."""
        self._inproc_registry[server_name] = dict(tools)

    def convert_env_vars(self, env_spec: List[Dict[str, Any]], namespace: str) -> Dict[str, str]:
        """Resolve literal values and secretKeyRef values
        (mcpmanager.go:73-111)."""
        out: Dict[str, str] = {}
        for ev in env_spec or []:
            name = ev.get("name", "")
            if not name:
                continue
            if ev.get("value"):
                out[name] = ev["value"]
            elif ev.get("valueFrom", {}).get("secretKeyRef"):
                ref = ev["valueFrom"]["secretKeyRef"]
                if self.store is None:
                    raise MCPError("secretKeyRef used but manager has no store")
                secret = self.store.get(SECRET, ref["name"], namespace)
                if secret is None:
                    raise MCPError(f'secret "{ref["name"]}" not found')
                data = secret.get("spec", {}).get("data", {}) or secret.get("data", {})
                if ref["key"] not in data:
                    raise MCPError(f'key "{ref["key"]}" not in secret "{ref["name"]}"')
                out[name] = str(data[ref["key"]])
        return out

    def connect_server(self, server: Dict[str, Any]) -> List[Dict[str, Any]]:
        """Connect + initialize + list tools; store the connection
        (mcpmanager.go:114-218).  Returns the discovered tool list."""
        name = server["metadata"]["name"]
        ns = server["metadata"].get("namespace", "default")
        spec = server.get("spec", {})
        transport = spec.get("transport", "stdio")

        if transport == "inproc" or (
            transport == "stdio" and spec.get("command", "") == "inproc"
        ):
            funcs = self._inproc_registry.get(name)
            if funcs is None and spec.get("command", "") == "inproc":
                # allow inline tool specs: args name the built-in tools
                funcs = {t: _BUILTIN_TOOLS[t] for t in spec.get("args", []) if t in _BUILTIN_TOOLS}
            if not funcs:
                raise MCPError(f"no in-process tools registered for server {name!r}")
            tools = [
                {
                    "name": t,
                    "description": getattr(f, "__doc__", "") or t,
                    "inputSchema": getattr(f, "input_schema", {"type": "object"}),
                }
                for t, f in funcs.items()
            ]
            conn = MCPConnection(name, "inproc", tools=tools, funcs=funcs)
        elif transport == "stdio":
            env = self.convert_env_vars(spec.get("env", []), ns)
            client = _StdioClient(spec.get("command", ""), spec.get("args", []) or [], env)
            client.call(
                "initialize",
                {
                    "protocolVersion": "2024-11-05",
                    "clientInfo": {"name": "acp-amd", "version": "0.1.0"},
                    "capabilities": {},
                },
            )
            client.notify("notifications/initialized")
            result = client.call("tools/list", {})
            tools = result.get("tools", []) if isinstance(result, dict) else []
            conn = MCPConnection(name, "stdio", client=client, tools=tools)
        elif transport == "http":
            raise MCPError("http/SSE MCP transport requires a reachable URL (no egress here)")
        else:
            raise MCPError(f"unknown MCP transport {transport!r}")

        with self._lock:
            old = self._conns.get(name)
            self._conns[name] = conn
        if old is not None and old.client is not None:
            old.client.close()
        return conn.tools

    def disconnect_server(self, name: str) -> None:
        with self._lock:
            conn = self._conns.pop(name, None)
        if conn is not None and conn.client is not None:
            conn.client.close()

    def get_connection(self, name: str) -> Optional[MCPConnection]:
        with self._lock:
            return self._conns.get(name)

    def get_tools(self, name: str) -> Optional[List[Dict[str, Any]]]:
        conn = self.get_connection(name)
        return list(conn.tools) if conn else None

    # ------------------------------------------------------------- execution

    def call_tool(self, server_name: str, tool_name: str, arguments: Dict[str, Any]) -> str:
        """Invoke a tool; concatenate text content; raise on IsError
        (mcpmanager.go:259-300)."""
        conn = self.get_connection(server_name)
        if conn is None:
            raise MCPError(f"no connection to MCP server {server_name!r}")
        if conn.transport == "inproc":
            func = conn.funcs.get(tool_name)
            if func is None:
                raise MCPError(f"tool {tool_name!r} not found on {server_name!r}")
            result = func(**arguments) if isinstance(arguments, dict) else func(arguments)
            return result if isinstance(result, str) else json.dumps(result)
        result = conn.client.call(
            "tools/call", {"name": tool_name, "arguments": arguments or {}}
        )
        if not isinstance(result, dict):
            return str(result)
        if result.get("isError"):
            raise MCPError(_concat_text(result))
        return _concat_text(result)

    def find_server_for_tool(self, full_tool_name: str):
        """``server__tool`` convention (mcpmanager.go:304-331)."""
        if "__" not in full_tool_name:
            return None, None
        server_name, tool_name = full_tool_name.split("__", 1)
        conn = self.get_connection(server_name)
        if conn is None:
            return None, None
        for t in conn.tools:
            if t.get("name") == tool_name:
                return server_name, tool_name
        return None, None

    def close(self) -> None:
        with self._lock:
            conns = list(self._conns.values())
            self._conns.clear()
        for c in conns:
            if c.client is not None:
                c.client.close()


def _concat_text(result: Dict[str, Any]) -> str:
    parts = []
    for item in result.get("content", []) or []:
        if item.get("type") == "text":
            parts.append(item.get("text", ""))
    return "".join(parts)


# ---------------------------------------------------------------- built-ins


def _tool_add(a=0, b=0, **_):
    """Add two numbers."""
    return str(float(a) + float(b))


_tool_add.input_schema = {
    "type": "object",
    "properties": {"a": {"type": "number"}, "b": {"type": "number"}},
    "required": ["a", "b"],
}


def _tool_echo(text="", **kw):
    """Echo the input text."""
    return text or json.dumps(kw)


_tool_echo.input_schema = {
    "type": "object",
    "properties": {"text": {"type": "string"}},
}


def _tool_noop(**_):
    """No-op tool: returns 'ok' (bench config 1)."""
    return "ok"


_tool_noop.input_schema = {"type": "object", "properties": {}}

_BUILTIN_TOOLS = {"add": _tool_add, "echo": _tool_echo, "noop": _tool_noop}
