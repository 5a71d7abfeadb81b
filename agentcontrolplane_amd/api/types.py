"""Resource schemas for the acp.humanlayer.dev/v1alpha1 API group.

The JSON wire/checkpoint format is kept identical to the reference CRDs so a
Task checkpoint (``status.contextWindow``) or an Agent manifest written for the
reference operator round-trips through this store unchanged:

- Task:           /root/reference/acp/api/v1alpha1/task_types.go:24-207
- ToolCall:       /root/reference/acp/api/v1alpha1/toolcall_types.go:26-130
- Agent:          /root/reference/acp/api/v1alpha1/agent_types.go:8-87
- LLM:            /root/reference/acp/api/v1alpha1/llm_types.go:141-199
- MCPServer:      /root/reference/acp/api/v1alpha1/mcpserver_types.go:10-62
- ContactChannel: /root/reference/acp/api/v1alpha1/contactchannel_types.go:59-120

Resources are stored as plain JSON-style dicts (the store is schemaless, like
etcd); this module provides the enums, the typed ``Message`` checkpoint format,
and constructors.  Spec/status field names are the Go struct json tags.
"""
from __future__ import annotations

import dataclasses
import datetime
import secrets
import uuid
from typing import Any, Dict, List, Optional

API_VERSION = "acp.humanlayer.dev/v1alpha1"

# Kinds
LLM = "LLM"
AGENT = "Agent"
TASK = "Task"
TOOL_CALL = "ToolCall"
MCP_SERVER = "MCPServer"
CONTACT_CHANNEL = "ContactChannel"
SECRET = "Secret"            # core/v1 in the reference cluster
EVENT = "Event"              # core/v1 Events — the user-facing execution history
LEASE = "Lease"              # coordination.k8s.io Lease — task locking

KINDS = [LLM, AGENT, TASK, TOOL_CALL, MCP_SERVER, CONTACT_CHANNEL, SECRET, EVENT, LEASE]


# ---------------------------------------------------------------------------
# Message — the Task-status context-window checkpoint format.
# Byte-identical JSON to task_types.go:57-97.
# ---------------------------------------------------------------------------

MESSAGE_ROLE_SYSTEM = "system"
MESSAGE_ROLE_USER = "user"
MESSAGE_ROLE_ASSISTANT = "assistant"
MESSAGE_ROLE_TOOL = "tool"
VALID_MESSAGE_ROLES = {
    MESSAGE_ROLE_SYSTEM,
    MESSAGE_ROLE_USER,
    MESSAGE_ROLE_ASSISTANT,
    MESSAGE_ROLE_TOOL,
}


@dataclasses.dataclass
class ToolCallFunction:
    name: str = ""
    arguments: str = ""

    def to_dict(self) -> Dict[str, Any]:
        return {"name": self.name, "arguments": self.arguments}

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "ToolCallFunction":
        return ToolCallFunction(name=d.get("name", ""), arguments=d.get("arguments", ""))


@dataclasses.dataclass
class MessageToolCall:
    id: str = ""
    function: ToolCallFunction = dataclasses.field(default_factory=ToolCallFunction)
    type: str = "function"

    def to_dict(self) -> Dict[str, Any]:
        return {"id": self.id, "function": self.function.to_dict(), "type": self.type}

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "MessageToolCall":
        return MessageToolCall(
            id=d.get("id", ""),
            function=ToolCallFunction.from_dict(d.get("function", {}) or {}),
            type=d.get("type", "function"),
        )


@dataclasses.dataclass
class Message:
    """One conversation message (task_types.go:57-78).

    Serialized keys: role, content, toolCalls, toolCallId, name — omitempty
    semantics preserved (absent when empty), so checkpoints match the
    reference byte format.
    """

    role: str = ""
    content: str = ""
    tool_calls: List[MessageToolCall] = dataclasses.field(default_factory=list)
    tool_call_id: str = ""
    name: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"role": self.role, "content": self.content}
        if self.tool_calls:
            d["toolCalls"] = [tc.to_dict() for tc in self.tool_calls]
        if self.tool_call_id:
            d["toolCallId"] = self.tool_call_id
        if self.name:
            d["name"] = self.name
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Message":
        return Message(
            role=d.get("role", ""),
            content=d.get("content", ""),
            tool_calls=[MessageToolCall.from_dict(t) for t in d.get("toolCalls", []) or []],
            tool_call_id=d.get("toolCallId", ""),
            name=d.get("name", ""),
        )


@dataclasses.dataclass
class SpanContext:
    """task_types.go:100-106 — persisted OTel span identity."""

    trace_id: str = ""
    span_id: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.trace_id:
            d["traceID"] = self.trace_id
        if self.span_id:
            d["spanID"] = self.span_id
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "SpanContext":
        return SpanContext(trace_id=d.get("traceID", ""), span_id=d.get("spanID", ""))


# ---------------------------------------------------------------------------
# Phase / status enums
# ---------------------------------------------------------------------------


class TaskStatusType:
    READY = "Ready"
    ERROR = "Error"
    PENDING = "Pending"


class TaskPhase:
    """task_types.go:167-193."""

    INITIALIZING = "Initializing"
    PENDING = "Pending"
    READY_FOR_LLM = "ReadyForLLM"
    SEND_CONTEXT_WINDOW_TO_LLM = "SendContextWindowToLLM"
    TOOL_CALLS_PENDING = "ToolCallsPending"
    CHECKING_TOOL_CALLS = "CheckingToolCalls"
    FINAL_ANSWER = "FinalAnswer"
    ERROR_BACKOFF = "ErrorBackoff"
    FAILED = "Failed"


class ToolCallStatusType:
    READY = "Ready"
    ERROR = "Error"
    PENDING = "Pending"
    SUCCEEDED = "Succeeded"


class ToolType:
    """toolcall_types.go:17-23."""

    MCP = "MCP"
    HUMAN_CONTACT = "HumanContact"
    DELEGATE_TO_AGENT = "DelegateToAgent"


class ToolCallPhase:
    """toolcall_types.go:92-116."""

    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"
    AWAITING_HUMAN_INPUT = "AwaitingHumanInput"
    AWAITING_SUB_AGENT = "AwaitingSubAgent"
    AWAITING_HUMAN_APPROVAL = "AwaitingHumanApproval"
    READY_TO_EXECUTE_APPROVED_TOOL = "ReadyToExecuteApprovedTool"
    ERROR_REQUESTING_HUMAN_APPROVAL = "ErrorRequestingHumanApproval"
    ERROR_REQUESTING_HUMAN_INPUT = "ErrorRequestingHumanInput"
    TOOL_CALL_REJECTED = "ToolCallRejected"


class AgentStatusType:
    READY = "Ready"
    ERROR = "Error"
    PENDING = "Pending"


class ContactChannelType:
    SLACK = "slack"
    EMAIL = "email"


# ---------------------------------------------------------------------------
# Object construction helpers
# ---------------------------------------------------------------------------


def now_iso() -> str:
    return datetime.datetime.now(datetime.timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%fZ")


def new_object_meta(
    name: str,
    namespace: str = "default",
    labels: Optional[Dict[str, str]] = None,
    owner_references: Optional[List[Dict[str, Any]]] = None,
) -> Dict[str, Any]:
    m: Dict[str, Any] = {
        "name": name,
        "namespace": namespace,
        "uid": str(uuid.uuid4()),
        "creationTimestamp": now_iso(),
    }
    if labels:
        m["labels"] = dict(labels)
    if owner_references:
        m["ownerReferences"] = list(owner_references)
    return m


def make_resource(
    kind: str,
    name: str,
    namespace: str = "default",
    spec: Optional[Dict[str, Any]] = None,
    labels: Optional[Dict[str, str]] = None,
    owner_references: Optional[List[Dict[str, Any]]] = None,
    api_version: str = API_VERSION,
) -> Dict[str, Any]:
    return {
        "apiVersion": api_version,
        "kind": kind,
        "metadata": new_object_meta(name, namespace, labels, owner_references),
        "spec": dict(spec or {}),
        "status": {},
    }


def owner_ref(obj: Dict[str, Any], controller: bool = True) -> Dict[str, Any]:
    """Build an ownerReference to ``obj`` (used for Task → ToolCall ownership)."""
    return {
        "apiVersion": obj.get("apiVersion", API_VERSION),
        "kind": obj["kind"],
        "name": obj["metadata"]["name"],
        "uid": obj["metadata"].get("uid", ""),
        "controller": controller,
    }


class meta:
    """Accessor helpers for unstructured resources."""

    @staticmethod
    def name(obj: Dict[str, Any]) -> str:
        return obj.get("metadata", {}).get("name", "")

    @staticmethod
    def namespace(obj: Dict[str, Any]) -> str:
        return obj.get("metadata", {}).get("namespace", "default")

    @staticmethod
    def labels(obj: Dict[str, Any]) -> Dict[str, str]:
        return obj.get("metadata", {}).get("labels", {}) or {}

    @staticmethod
    def uid(obj: Dict[str, Any]) -> str:
        return obj.get("metadata", {}).get("uid", "")

    @staticmethod
    def rv(obj: Dict[str, Any]) -> int:
        return int(obj.get("metadata", {}).get("resourceVersion", 0))


def rand_suffix(n: int = 8) -> str:
    """Lowercase alphanumeric random suffix (validation.go GenerateK8sRandomString)."""
    alphabet = "abcdefghijklmnopqrstuvwxyz0123456789"
    return "".join(secrets.choice(alphabet) for _ in range(n))
