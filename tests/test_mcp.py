"""MCP manager tier: stdio subprocess JSON-RPC, tool discovery, routing,
secret-resolved env vars (mcpmanager_test.go's role)."""
import sys

import pytest

from agentcontrolplane_amd.api.types import SECRET, make_resource
from agentcontrolplane_amd.mcp.manager import MCPError, MCPServerManager


def _stdio_server_obj(name="calc"):
    return {
        "apiVersion": "acp.humanlayer.dev/v1alpha1",
        "kind": "MCPServer",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "transport": "stdio",
            "command": sys.executable,
            "args": ["-m", "agentcontrolplane_amd.mcp.echo_server"],
        },
        "status": {},
    }


def test_stdio_connect_list_call(store):
    mgr = MCPServerManager(store)
    try:
        tools = mgr.connect_server(_stdio_server_obj())
        assert {t["name"] for t in tools} == {"add", "echo", "noop", "sleep"}
        assert mgr.call_tool("calc", "add", {"a": 2, "b": 3}) == "5.0"
        assert mgr.call_tool("calc", "echo", {"text": "hi"}) == "hi"
        server, tool = mgr.find_server_for_tool("calc__add")
        assert (server, tool) == ("calc", "add")
        assert mgr.find_server_for_tool("calc__missing") == (None, None)
        assert mgr.find_server_for_tool("nounderscore") == (None, None)
    finally:
        mgr.close()


def test_stdio_unknown_tool_error(store):
    mgr = MCPServerManager(store)
    try:
        mgr.connect_server(_stdio_server_obj())
        with pytest.raises(MCPError):
            mgr.call_tool("calc", "nope", {})
    finally:
        mgr.close()


def test_env_var_secret_resolution(store):
    store.create(
        make_resource(SECRET, "creds", spec={"data": {"token": "s3cret"}}, api_version="v1")
    )
    mgr = MCPServerManager(store)
    env = mgr.convert_env_vars(
        [
            {"name": "PLAIN", "value": "x"},
            {"name": "FROM_SECRET", "valueFrom": {"secretKeyRef": {"name": "creds", "key": "token"}}},
        ],
        "default",
    )
    assert env == {"PLAIN": "x", "FROM_SECRET": "s3cret"}
    with pytest.raises(MCPError):
        mgr.convert_env_vars(
            [{"name": "V", "valueFrom": {"secretKeyRef": {"name": "creds", "key": "nope"}}}],
            "default",
        )


def test_inproc_registry(store):
    mgr = MCPServerManager(store)
    mgr.register_inproc("local", {"double": lambda x=0, **_: str(2 * float(x))})
    obj = {
        "kind": "MCPServer",
        "metadata": {"name": "local", "namespace": "default"},
        "spec": {"transport": "inproc"},
    }
    tools = mgr.connect_server(obj)
    assert tools[0]["name"] == "double"
    assert mgr.call_tool("local", "double", {"x": 4}) == "8.0"


def test_stdio_concurrent_calls(store):
    """Many threads calling one stdio server: the per-connection lock
    serializes frames — every call gets ITS OWN response."""
    import threading

    mgr = MCPServerManager(store)
    try:
        mgr.connect_server(_stdio_server_obj())
        results = [None] * 24

        def go(i):
            results[i] = mgr.call_tool("calc", "add", {"a": i, "b": 1000})

        ts = [threading.Thread(target=go, args=(i,)) for i in range(24)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        for i, r in enumerate(results):
            assert r == str(float(i + 1000)), (i, r)
    finally:
        mgr.close()
