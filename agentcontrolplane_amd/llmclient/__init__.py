from .base import (  # noqa: F401
    LLMClient,
    LLMRequestError,
    Tool,
    ToolFunction,
    tool_from_contact_channel,
)
from .factory import LLMClientFactory  # noqa: F401
from .mock import MockLLMClient, ScriptedResponder  # noqa: F401
