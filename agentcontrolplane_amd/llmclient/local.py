"""The *local* LLM provider — chat completion against the in-process engine.

This is the component that replaces the reference's entire remote-provider
path (langchaingo_client.go): ``send_request`` renders the ACP context
window into the model's chat template token stream, submits it to the
continuous-batching engine, and converts the engine result (text or
constrained tool-call JSON) back into an ACP assistant ``Message``.

The call blocks the calling reconciler worker on a future; the engine
batches every in-flight task's request into shared prefill/decode steps on
the GPU (one paged-KV pool per GPU), so concurrency comes from the number
of tasks, not the number of HTTP connections.
"""
from __future__ import annotations

from typing import Any, Dict, List

from ..api.types import Message, MessageToolCall, ToolCallFunction
from .base import LLMClient, LLMRequestError, Tool, normalize_response


class LocalEngineClient(LLMClient):
    def __init__(self, engine, llm: Dict[str, Any]):
        self.engine = engine
        params = llm.get("spec", {}).get("parameters", {}) or {}
        self.model = params.get("model", "")
        self.max_tokens = params.get("maxTokens") or 256
        self.temperature = float(params.get("temperature") or 0.7)
        self.top_p = float(params.get("topP") or 1.0)
        self.top_k = int(params.get("topK") or 0)
        self.frequency_penalty = float(params.get("frequencyPenalty") or 0.0)
        self.presence_penalty = float(params.get("presencePenalty") or 0.0)

    def _sampling(self):
        from ..engine.request import SamplingParams

        return SamplingParams(
            max_tokens=int(self.max_tokens),
            temperature=self.temperature,
            top_p=self.top_p,
            top_k=self.top_k,
            frequency_penalty=self.frequency_penalty,
            presence_penalty=self.presence_penalty,
        )

    @staticmethod
    def _map_error(e: BaseException) -> LLMRequestError:
        if isinstance(e, TimeoutError):
            return LLMRequestError(503, f"engine timeout: {e}")
        if isinstance(e, ValueError):
            # invalid request (e.g. context-limit overflow) — terminal 4xx so
            # the Task fails instead of retrying forever
            return LLMRequestError(400, f"invalid request: {e}")
        return LLMRequestError(500, f"engine error: {e}")

    def send_request(self, messages: List[Message], tools: List[Tool]) -> Message:
        try:
            result = self.engine.chat(
                messages=[m.to_dict() for m in messages],
                tools=[t.to_dict() for t in tools],
                sampling=self._sampling(),
            )
        except Exception as e:
            raise self._map_error(e)
        return self._to_message(result)

    def send_request_async(self, messages, tools, callback) -> None:
        """Non-blocking submit into the continuous-batching engine; the
        callback fires from the engine thread at turn completion."""

        def _cb(result, error):
            if error is not None:
                callback(None, self._map_error(error))
            else:
                callback(self._to_message(result), None)

        self.engine.chat_async(
            [m.to_dict() for m in messages],
            [t.to_dict() for t in tools],
            self._sampling(),
            _cb,
        )

    def _to_message(self, result) -> Message:
        msg = Message(role="assistant")
        if result.tool_calls:
            msg.tool_calls = [
                MessageToolCall(
                    id=tc["id"],
                    function=ToolCallFunction(
                        name=tc["function"]["name"],
                        arguments=tc["function"]["arguments"],
                    ),
                    type="function",
                )
                for tc in result.tool_calls
            ]
        else:
            msg.content = result.text
        return normalize_response(msg)
