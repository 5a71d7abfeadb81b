"""Mock LLM provider for GPU-free controller tests and the config-1 bench.

Plays the role of the reference's mockgen LLM client + the httptest mock
OpenAI server (test/e2e/getting_started/test_getting_started.go:250-338):
a deterministic scripted responder that can return final answers or
tool-call turns.
"""
from __future__ import annotations

import itertools
import json
import threading
from typing import Callable, List, Optional, Sequence

from ..api.types import Message, MessageToolCall, ToolCallFunction
from .base import LLMClient, Tool, normalize_response


def final_answer(content: str) -> Message:
    return Message(role="assistant", content=content)


def tool_call_turn(calls: Sequence[tuple]) -> Message:
    """calls: sequence of (id, tool_name, arguments_json)."""
    return Message(
        role="assistant",
        tool_calls=[
            MessageToolCall(
                id=cid, function=ToolCallFunction(name=name, arguments=args), type="function"
            )
            for cid, name, args in calls
        ],
    )


class ScriptedResponder:
    """Returns each scripted message once, then repeats the last one."""

    def __init__(self, script: List[Message]):
        self._script = list(script)
        self._i = 0
        self._lock = threading.Lock()

    def __call__(self, messages, tools) -> Message:
        with self._lock:
            msg = self._script[min(self._i, len(self._script) - 1)]
            self._i += 1
        return Message.from_dict(msg.to_dict())  # fresh copy


class MockLLMClient(LLMClient):
    """Deterministic mock.

    Default behavior: if the conversation's last message is a tool result,
    return a final answer; otherwise, if tools are offered and
    ``tool_turns`` > completed tool rounds, request one call of the first
    MCP tool; else return a final answer.  A custom ``responder`` callable
    overrides everything.
    """

    def __init__(
        self,
        responder: Optional[Callable] = None,
        tool_turns: int = 1,
        answer: str = "mock final answer",
        latency_s: float = 0.0,
    ):
        self._responder = responder
        self._tool_turns = tool_turns
        self._answer = answer
        self._latency = latency_s
        self._ids = itertools.count(1)

    def send_request(self, messages: List[Message], tools: List[Tool]) -> Message:
        if self._latency > 0:
            import time

            time.sleep(self._latency)
        if self._responder is not None:
            return normalize_response(self._responder(messages, tools))
        tool_rounds = sum(1 for m in messages if m.role == "assistant" and m.tool_calls)
        if tools and tool_rounds < self._tool_turns:
            tool = tools[0]
            args = json.dumps({"a": 1, "b": 2})
            return normalize_response(
                tool_call_turn([(f"call_{next(self._ids):06d}", tool.function.name, args)])
            )
        return normalize_response(final_answer(self._answer))
