"""A minimal stdio MCP server (newline-delimited JSON-RPC 2.0).

Run as ``python -m agentcontrolplane_amd.mcp.echo_server``; used by tests and
by MCPServer resources with transport=stdio, mirroring the reference's use of
`uvx`/`npx`-spawned MCP servers inside the controller pod (acp/Dockerfile).

Tools: add(a, b), echo(text), noop().
"""
from __future__ import annotations

import json
import sys
import threading

_OUT_LOCK = threading.Lock()

TOOLS = [
    {
        "name": "add",
        "description": "Add two numbers",
        "inputSchema": {
            "type": "object",
            "properties": {"a": {"type": "number"}, "b": {"type": "number"}},
            "required": ["a", "b"],
        },
    },
    {
        "name": "echo",
        "description": "Echo the input text",
        "inputSchema": {"type": "object", "properties": {"text": {"type": "string"}}},
    },
    {"name": "noop", "description": "No-op", "inputSchema": {"type": "object", "properties": {}}},
    {
        "name": "sleep",
        "description": "Sleep for `seconds`, then return 'slept' (tests the client's concurrent in-flight requests)",
        "inputSchema": {"type": "object", "properties": {"seconds": {"type": "number"}}},
    },
]


def handle(method, params):
    if method == "initialize":
        return {
            "protocolVersion": "2024-11-05",
            "serverInfo": {"name": "acp-echo-server", "version": "0.1.0"},
            "capabilities": {"tools": {}},
        }
    if method == "tools/list":
        return {"tools": TOOLS}
    if method == "tools/call":
        name = params.get("name")
        args = params.get("arguments", {}) or {}
        if name == "add":
            text = str(float(args.get("a", 0)) + float(args.get("b", 0)))
        elif name == "echo":
            text = args.get("text", "")
        elif name == "noop":
            text = "ok"
        elif name == "sleep":
            import time

            time.sleep(float(args.get("seconds", 0)))
            text = "slept"
        else:
            return {"isError": True, "content": [{"type": "text", "text": f"unknown tool {name}"}]}
        return {"content": [{"type": "text", "text": text}]}
    raise KeyError(method)


def _respond(msg) -> None:
    try:
        result = handle(msg.get("method", ""), msg.get("params", {}) or {})
        resp = {"jsonrpc": "2.0", "id": msg["id"], "result": result}
    except KeyError:
        resp = {
            "jsonrpc": "2.0",
            "id": msg["id"],
            "error": {"code": -32601, "message": "method not found"},
        }
    with _OUT_LOCK:
        sys.stdout.write(json.dumps(resp) + "\n")
        sys.stdout.flush()


def main() -> None:
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        try:
            msg = json.loads(line)
        except json.JSONDecodeError:
            continue
        if "id" not in msg:
            continue  # notification
        # thread-per-request: a slow tool must not block other callers
        # (exercises the client's concurrent in-flight request ids)
        threading.Thread(target=_respond, args=(msg,), daemon=True).start()


if __name__ == "__main__":
    main()
