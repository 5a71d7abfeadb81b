"""MCP http/SSE transports + concurrent stdio requests.

The reference connects http MCP servers through NewSSEMCPClient
(mcpmanager.go:161-175); here both the modern streamable-HTTP wire and the
legacy HTTP+SSE wire are served by an in-process ThreadingHTTPServer and
exercised over real sockets."""
from __future__ import annotations

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from agentcontrolplane_amd.mcp.manager import (
    MCPError,
    MCPServerManager,
    _SSEClient,
    _StdioClient,
)

TOOLS = [
    {"name": "greet", "description": "greet", "inputSchema": {
        "type": "object", "properties": {"who": {"type": "string"}}}},
]


def _rpc_result(msg):
    method = msg.get("method", "")
    if method == "initialize":
        return {"protocolVersion": "2024-11-05",
                "serverInfo": {"name": "mock-http-mcp", "version": "0"},
                "capabilities": {"tools": {}}}
    if method == "tools/list":
        return {"tools": TOOLS}
    if method == "tools/call":
        args = msg["params"].get("arguments", {}) or {}
        if msg["params"].get("name") != "greet":
            return {"isError": True,
                    "content": [{"type": "text", "text": "unknown tool"}]}
        return {"content": [{"type": "text", "text": f"hello {args.get('who', '')}"}]}
    return None


class _StreamableHTTPHandler(BaseHTTPRequestHandler):
    """Streamable-HTTP MCP: POST with JSON response + Mcp-Session-Id."""

    def do_POST(self):  # noqa: N802
        n = int(self.headers.get("Content-Length", 0))
        msg = json.loads(self.rfile.read(n) or b"{}")
        if "id" not in msg:  # notification
            self.send_response(202)
            self.send_header("Content-Length", "0")
            self.end_headers()
            return
        result = _rpc_result(msg)
        reply = {"jsonrpc": "2.0", "id": msg["id"], "result": result}
        data = json.dumps(reply).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Mcp-Session-Id", "sess-123")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def log_message(self, *a):
        pass


class _SSEHandler(BaseHTTPRequestHandler):
    """Legacy HTTP+SSE MCP: GET /sse streams endpoint + message events;
    POST /messages accepts JSON-RPC, responses go down the stream."""

    server_version = "MockSSE/0"
    # per-server shared state
    streams = None  # set on the server instance

    def do_GET(self):  # noqa: N802
        if self.path != "/sse":
            self.send_error(404)
            return
        self.send_response(200)
        self.send_header("Content-Type", "text/event-stream")
        self.send_header("Cache-Control", "no-cache")
        self.end_headers()
        self.wfile.write(b"event: endpoint\ndata: /messages\n\n")
        self.wfile.flush()
        q = self.server.out_queue
        while True:
            item = q.get()
            if item is None:
                break
            payload = f"event: message\ndata: {json.dumps(item)}\n\n".encode()
            try:
                self.wfile.write(payload)
                self.wfile.flush()
            except BrokenPipeError:
                break

    def do_POST(self):  # noqa: N802
        if self.path != "/messages":
            self.send_error(404)
            return
        n = int(self.headers.get("Content-Length", 0))
        msg = json.loads(self.rfile.read(n) or b"{}")
        self.send_response(202)
        self.send_header("Content-Length", "0")
        self.end_headers()
        if "id" in msg:
            def reply():
                params = msg.get("params", {}) or {}
                if msg.get("method") == "tools/call" and \
                        params.get("name") == "greet" and \
                        params.get("arguments", {}).get("slow"):
                    time.sleep(0.5)
                self.server.out_queue.put(
                    {"jsonrpc": "2.0", "id": msg["id"], "result": _rpc_result(msg)}
                )
            threading.Thread(target=reply, daemon=True).start()

    def log_message(self, *a):
        pass


@pytest.fixture()
def http_mcp_server():
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _StreamableHTTPHandler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}/mcp"
    srv.shutdown()


@pytest.fixture()
def sse_mcp_server():
    import queue

    srv = ThreadingHTTPServer(("127.0.0.1", 0), _SSEHandler)
    srv.out_queue = queue.Queue()
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}/sse"
    srv.out_queue.put(None)
    srv.shutdown()


def test_streamable_http_connect_and_call(http_mcp_server):
    mgr = MCPServerManager()
    tools = mgr.connect_server({
        "metadata": {"name": "web"},
        "spec": {"transport": "http", "url": http_mcp_server},
    })
    assert [t["name"] for t in tools] == ["greet"]
    assert mgr.call_tool("web", "greet", {"who": "world"}) == "hello world"
    # session id captured from the response header
    assert mgr.get_connection("web").client._session_id == "sess-123"
    # server__tool routing works across transports (mcpmanager.go:304-331)
    assert mgr.find_server_for_tool("web__greet") == ("web", "greet")
    mgr.close()


def test_sse_connect_and_call(sse_mcp_server):
    mgr = MCPServerManager()
    tools = mgr.connect_server({
        "metadata": {"name": "ssesrv"},
        "spec": {"transport": "sse", "url": sse_mcp_server},
    })
    assert [t["name"] for t in tools] == ["greet"]
    assert mgr.call_tool("ssesrv", "greet", {"who": "sse"}) == "hello sse"
    mgr.close()


def test_sse_concurrent_requests(sse_mcp_server):
    client = _SSEClient(sse_mcp_server)
    client.call("initialize", {})
    results = {}

    def slow():
        results["slow"] = (client.call(
            "tools/call", {"name": "greet", "arguments": {"who": "s", "slow": 1}}),
            time.monotonic())

    t = threading.Thread(target=slow)
    t.start()
    time.sleep(0.05)
    fast = client.call("tools/call", {"name": "greet", "arguments": {"who": "f"}})
    t_fast = time.monotonic()
    t.join(timeout=5)
    # the fast call completed while the slow one was still in flight
    assert fast["content"][0]["text"] == "hello f"
    assert results["slow"][0]["content"][0]["text"] == "hello s"
    assert t_fast < results["slow"][1]
    client.close()


def test_http_fallback_to_sse(sse_mcp_server):
    """spec.transport=http against a legacy SSE-only server falls back."""
    mgr = MCPServerManager()
    tools = mgr.connect_server({
        "metadata": {"name": "legacy"},
        "spec": {"transport": "http", "url": sse_mcp_server},
    })
    assert [t["name"] for t in tools] == ["greet"]
    mgr.close()


def test_http_error_surfaces():
    mgr = MCPServerManager()
    with pytest.raises(MCPError):
        mgr.connect_server({
            "metadata": {"name": "down"},
            "spec": {"transport": "http", "url": "http://127.0.0.1:9/nothing"},
        })


def test_stdio_concurrent_in_flight():
    """One slow tool must not serialize other callers (VERDICT weak #7)."""
    import sys

    client = _StdioClient(sys.executable,
                          ["-m", "agentcontrolplane_amd.mcp.echo_server"])
    client.call("initialize", {"protocolVersion": "2024-11-05",
                               "clientInfo": {"name": "t", "version": "0"},
                               "capabilities": {}})
    done = {}

    def slow():
        client.call("tools/call", {"name": "sleep", "arguments": {"seconds": 1.0}})
        done["slow"] = time.monotonic()

    t = threading.Thread(target=slow)
    t.start()
    time.sleep(0.1)
    t0 = time.monotonic()
    r = client.call("tools/call", {"name": "echo", "arguments": {"text": "quick"}})
    t_fast = time.monotonic()
    assert r["content"][0]["text"] == "quick"
    assert t_fast - t0 < 0.5, "echo was serialized behind the sleeping call"
    t.join(timeout=5)
    assert done["slow"] > t_fast
    client.close()
