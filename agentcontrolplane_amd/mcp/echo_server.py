"""A minimal stdio MCP server (newline-delimited JSON-RPC 2.0).

Run as ``python -m agentcontrolplane_amd.mcp.echo_server``; used by tests and
by MCPServer resources with transport=stdio, mirroring the reference's use of
`uvx`/`npx`-spawned MCP servers inside the controller pod (acp/Dockerfile).

Tools: add(a, b), echo(text), noop().
"""
from __future__ import annotations

import json
import sys

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
        else:
            return {"isError": True, "content": [{"type": "text", "text": f"unknown tool {name}"}]}
        return {"content": [{"type": "text", "text": text}]}
    raise KeyError(method)


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
        try:
            result = handle(msg.get("method", ""), msg.get("params", {}) or {})
            resp = {"jsonrpc": "2.0", "id": msg["id"], "result": result}
        except KeyError:
            resp = {
                "jsonrpc": "2.0",
                "id": msg["id"],
                "error": {"code": -32601, "message": "method not found"},
            }
        sys.stdout.write(json.dumps(resp) + "\n")
        sys.stdout.flush()


if __name__ == "__main__":
    main()
