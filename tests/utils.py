"""Fixture builders — parity with the reference's acp/test/utils
(TestLLM/TestAgent/TestTask/… with Setup / SetupWithStatus / Teardown):
specs can start mid-state-machine by force-writing status."""
from __future__ import annotations

from typing import Any, Dict, Optional

from agentcontrolplane_amd.api.types import (
    AGENT,
    CONTACT_CHANNEL,
    LLM,
    MCP_SERVER,
    SECRET,
    TASK,
    TOOL_CALL,
    make_resource,
)


class _Fixture:
    kind = ""
    __test__ = False  # not pytest collectibles

    def __init__(self, name: str, spec: Optional[Dict[str, Any]] = None, namespace="default",
                 labels: Optional[Dict[str, str]] = None):
        self.name = name
        self.namespace = namespace
        self.spec = spec or self.default_spec()
        self.labels = labels

    def default_spec(self) -> Dict[str, Any]:
        return {}

    def setup(self, store) -> Dict[str, Any]:
        return store.create(
            make_resource(self.kind, self.name, self.namespace, self.spec, labels=self.labels)
        )

    def setup_with_status(self, store, status: Dict[str, Any]) -> Dict[str, Any]:
        """Force-write status so a controller test starts mid-machine
        (reference test/utils/llm.go:48-59)."""
        obj = self.setup(store)
        obj["status"] = dict(status)
        return store.update_status(obj)

    def teardown(self, store) -> None:
        store.delete(self.kind, self.name, self.namespace)


class TestSecret(_Fixture):
    kind = SECRET

    def default_spec(self):
        return {"data": {"api-key": "sk-test"}}

    def setup(self, store):
        return store.create(
            make_resource(self.kind, self.name, self.namespace, self.spec, api_version="v1")
        )


class TestLLM(_Fixture):
    kind = LLM

    def default_spec(self):
        return {"provider": "mock"}


class TestAgent(_Fixture):
    kind = AGENT

    def default_spec(self):
        return {"llmRef": {"name": "test-llm"}, "system": "test system prompt"}


class TestTask(_Fixture):
    kind = TASK

    def default_spec(self):
        return {"agentRef": {"name": "test-agent"}, "userMessage": "test message"}


class TestToolCall(_Fixture):
    kind = TOOL_CALL

    def default_spec(self):
        return {
            "toolCallId": "call_000001",
            "taskRef": {"name": "test-task"},
            "toolRef": {"name": "srv__tool"},
            "toolType": "MCP",
            "arguments": "{}",
        }


class TestMCPServer(_Fixture):
    kind = MCP_SERVER

    def default_spec(self):
        return {"transport": "inproc"}


class TestContactChannel(_Fixture):
    kind = CONTACT_CHANNEL

    def default_spec(self):
        return {
            "type": "slack",
            "apiKeyFrom": {"secretKeyRef": {"name": "test-secret", "key": "api-key"}},
            "slack": {"channelOrUserID": "C123"},
        }


READY = {"ready": True, "status": "Ready"}
