"""The LLMClient seam.

Interface parity with the reference's ``LLMClient`` (llm_client.go:11-14):
``send_request(messages, tools) -> Message``.  The reference implements it by
HTTPS calls to 5 remote providers through langchaingo; here the interesting
implementation is the *local* provider (``agentcontrolplane_amd.llmclient.local``)
backed by the in-process MI355X inference engine.  The mock provider keeps
controller tests GPU-free, mirroring the reference's mockgen seam
(task_controller.go:131-138).
"""
from __future__ import annotations

import dataclasses
from typing import Any, Dict, List

from ..api.types import Message, ToolType


class LLMRequestError(Exception):
    """LLM request error carrying an HTTP-style status code (llm_client.go:18-30).

    The Task controller treats 4xx as terminal (task → Failed) and everything
    else as retryable (task/state_machine.go:733-789)."""

    def __init__(self, status_code: int, message: str):
        super().__init__(f"LLM request failed with status {status_code}: {message}")
        self.status_code = status_code
        self.message = message


@dataclasses.dataclass
class ToolFunction:
    name: str = ""
    description: str = ""
    parameters: Dict[str, Any] = dataclasses.field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "description": self.description,
            "parameters": self.parameters,
        }


@dataclasses.dataclass
class Tool:
    """llm_client.go:33-50 — OpenAI-style function tool; ``acp_tool_type`` is
    internal routing metadata (MCP / HumanContact / DelegateToAgent) and never
    reaches the model."""

    function: ToolFunction
    type: str = "function"
    acp_tool_type: str = ToolType.MCP

    def to_dict(self) -> Dict[str, Any]:
        return {"type": self.type, "function": self.function.to_dict()}


class LLMClient:
    """Uniform chat-completion interface (llm_client.go:11-14).

    ``send_request`` blocks; ``send_request_async`` delivers the assistant
    Message (or error) through a callback so reconciler workers never block
    on a turn — engine-backed clients override it with a true async submit,
    the default runs synchronously in the calling thread (mock/remote)."""

    def send_request(self, messages: List[Message], tools: List[Tool]) -> Message:
        raise NotImplementedError

    def send_request_async(self, messages, tools, callback) -> None:
        try:
            msg = self.send_request(messages, tools)
        except Exception as e:  # noqa: BLE001 — delivered to the callback
            callback(None, e)
            return
        callback(msg, None)


def tool_from_contact_channel(channel: Dict[str, Any]) -> Tool:
    """Build a human-contact tool from a ContactChannel resource
    (llm_client.go:53-99): ``<channel>__human_contact_<type>`` with a
    one-required-string-arg schema."""
    params = {
        "type": "object",
        "properties": {"message": {"type": "string"}},
        "required": ["message"],
    }
    spec = channel.get("spec", {})
    cname = channel["metadata"]["name"]
    ctype = spec.get("type", "")
    if ctype == "email":
        name = f"{cname}__human_contact_email"
        description = (spec.get("email") or {}).get("contextAboutUser", "") or (
            "Contact a human via email"
        )
    elif ctype == "slack":
        name = f"{cname}__human_contact_slack"
        description = (spec.get("slack") or {}).get("contextAboutChannelOrUser", "") or (
            "Contact a human via Slack"
        )
    else:
        name = f"{cname}__human_contact"
        description = f"Contact a human via {ctype} channel"
    return Tool(
        function=ToolFunction(name=name, description=description, parameters=params),
        acp_tool_type=ToolType.HUMAN_CONTACT,
    )


def normalize_response(message: Message) -> Message:
    """'Tool calls win over content' (langchaingo_client.go:208-282): when an
    assistant turn contains tool calls, content is cleared so the controller
    takes the tool-call execution path."""
    if message.tool_calls:
        message.content = ""
    message.role = "assistant"
    return message
