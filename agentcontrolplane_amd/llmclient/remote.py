"""Remote LLM provider clients (httpx).

The reference reaches openai/anthropic/mistral/google/vertex through
langchaingo (langchaingo_client.go:27-80 provider construction,
118-185 message conversion, 208-282 response conversion with the
"tool-calls win over content" rule).  This module implements the same
five providers natively over httpx:

- role mapping system/user/assistant/tool (convertToLangchainMessages),
- assistant tool-call messages and tool-result messages on each wire,
- tools as function declarations,
- response conversion that collects tool calls across ALL choices and
  clears content when any tool call is present (langchaingo_client.go:
  255-268),
- HTTP 4xx surfaces as LLMRequestError(status) so the Task controller's
  4xx-terminal logic applies (task/state_machine.go:733-789),
- ``parameters.baseUrl`` points the client anywhere — the reference's
  e2e tests drive the full tool-calling loop against an httptest mock
  OpenAI server this way (test_getting_started.go:250-261), and
  tests/test_remote_providers.py does the same here,
- Azure APIType/api-version (llm/state_machine.go:267-280), the
  Anthropic beta header (state_machine.go:293-295), Mistral randomSeed,
  and Vertex project/location config are honored.

There is no egress in this deployment, so every network-touching test
runs against an in-process mock server; the wire formats follow the
providers' public REST APIs.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Optional

from ..api.types import Message, MessageToolCall, ToolCallFunction
from .base import LLMClient, LLMRequestError, Tool


def _f(v) -> Optional[float]:
    """BaseConfig carries numeric knobs as strings (llm_types.go:50-70)."""
    if v in (None, ""):
        return None
    try:
        return float(v)
    except (TypeError, ValueError):
        return None


class RemoteLLMClient(LLMClient):
    """Shared httpx plumbing: timeout, retries, error typing."""

    provider = "remote"

    def __init__(self, parameters: Dict[str, Any], api_key: str,
                 provider_config: Optional[Dict[str, Any]] = None,
                 transport=None):
        import httpx

        self.params = parameters or {}
        self.api_key = api_key
        self.pconf = provider_config or {}
        timeout = self.params.get("timeout") or 30
        self._client = httpx.Client(
            timeout=float(timeout), transport=transport, trust_env=False
        )
        self.max_retries = int(self.params.get("maxRetries") or 2)

    def close(self) -> None:
        self._client.close()

    # -- subclass surface --------------------------------------------------

    def _build(self, messages: List[Message], tools: List[Tool]):
        """-> (url, headers, body_dict)"""
        raise NotImplementedError

    def _parse(self, body: Dict[str, Any]) -> Message:
        raise NotImplementedError

    # ----------------------------------------------------------------------

    def send_request(self, messages: List[Message], tools: List[Tool]) -> Message:
        import httpx

        url, headers, body = self._build(messages, tools)
        last_err: Optional[Exception] = None
        for attempt in range(self.max_retries + 1):
            try:
                r = self._client.post(url, headers=headers, json=body)
            except httpx.HTTPError as e:
                last_err = LLMRequestError(502, f"{self.provider}: {e}")
                continue
            if 200 <= r.status_code < 300:
                try:
                    return self._parse(r.json())
                except (KeyError, ValueError, TypeError) as e:
                    raise LLMRequestError(
                        502, f"{self.provider}: malformed response: {e}"
                    ) from e
            err = LLMRequestError(r.status_code, f"{self.provider}: {r.text[:500]}")
            if 400 <= r.status_code < 500 and r.status_code != 429:
                raise err  # non-retryable: the controller treats 4xx terminal
            last_err = err
        raise last_err if last_err is not None else LLMRequestError(
            502, f"{self.provider}: no response"
        )


def _winning_message(content: str, tool_calls: List[MessageToolCall]) -> Message:
    """langchaingo_client.go:255-275 — tool calls win over content."""
    if tool_calls:
        return Message(role="assistant", content="", tool_calls=tool_calls)
    return Message(role="assistant", content=content or "")


class OpenAIClient(RemoteLLMClient):
    """OpenAI chat-completions wire (also Azure OpenAI via apiType)."""

    provider = "openai"
    default_base = "https://api.openai.com/v1"

    def _headers(self) -> Dict[str, str]:
        api_type = (self.pconf.get("apiType") or "").upper()
        if api_type in ("AZURE", "AZURE_AD"):
            # Azure uses api-key (or a bearer AD token) + api-version query
            if api_type == "AZURE":
                return {"api-key": self.api_key}
            return {"Authorization": f"Bearer {self.api_key}"}
        h = {"Authorization": f"Bearer {self.api_key}"}
        if self.pconf.get("organization"):
            h["OpenAI-Organization"] = self.pconf["organization"]
        return h

    def _build(self, messages, tools):
        base = (self.params.get("baseUrl") or self.default_base).rstrip("/")
        url = f"{base}/chat/completions"
        api_type = (self.pconf.get("apiType") or "").upper()
        if api_type in ("AZURE", "AZURE_AD"):
            api_version = self.pconf.get("apiVersion") or "2024-02-01"
            model = self.params.get("model", "")
            url = f"{base}/openai/deployments/{model}/chat/completions?api-version={api_version}"
        wire_msgs = []
        for m in messages:
            d: Dict[str, Any] = {"role": m.role or "user", "content": m.content or ""}
            if m.tool_calls:
                d["tool_calls"] = [
                    {
                        "id": tc.id,
                        "type": tc.type or "function",
                        "function": {
                            "name": tc.function.name,
                            "arguments": tc.function.arguments,
                        },
                    }
                    for tc in m.tool_calls
                ]
                d["content"] = m.content or None
            if m.tool_call_id:
                d["tool_call_id"] = m.tool_call_id
            wire_msgs.append(d)
        body: Dict[str, Any] = {
            "model": self.params.get("model", ""),
            "messages": wire_msgs,
        }
        if tools:
            body["tools"] = [t.to_dict() for t in tools]
        if self.params.get("maxTokens") is not None:
            body["max_tokens"] = int(self.params["maxTokens"])
        for src, dst in (
            ("temperature", "temperature"),
            ("topP", "top_p"),
            ("frequencyPenalty", "frequency_penalty"),
            ("presencePenalty", "presence_penalty"),
        ):
            v = _f(self.params.get(src))
            if v is not None:
                body[dst] = v
        return url, self._headers(), body

    def _parse(self, body):
        content = ""
        tool_calls: List[MessageToolCall] = []
        for choice in body.get("choices", []) or []:
            msg = choice.get("message", {}) or {}
            if not content and msg.get("content"):
                content = msg["content"]
            for tc in msg.get("tool_calls", []) or []:
                fn = tc.get("function", {}) or {}
                tool_calls.append(
                    MessageToolCall(
                        id=tc.get("id", ""),
                        type=tc.get("type", "function"),
                        function=ToolCallFunction(
                            name=fn.get("name", ""),
                            arguments=fn.get("arguments", ""),
                        ),
                    )
                )
        return _winning_message(content, tool_calls)


class MistralClient(OpenAIClient):
    """Mistral's chat API is OpenAI-compatible; adds randomSeed."""

    provider = "mistral"
    default_base = "https://api.mistral.ai/v1"

    def _build(self, messages, tools):
        url, headers, body = super()._build(messages, tools)
        if self.pconf.get("randomSeed") is not None:
            body["random_seed"] = int(self.pconf["randomSeed"])
        if self.pconf.get("maxTokens") is not None:
            body.setdefault("max_tokens", int(self.pconf["maxTokens"]))
        return url, headers, body


class AnthropicClient(RemoteLLMClient):
    """Anthropic messages wire: system extracted to the top level, tool
    results as user-role tool_result blocks, tools as input_schema."""

    provider = "anthropic"
    default_base = "https://api.anthropic.com"

    def _build(self, messages, tools):
        base = (self.params.get("baseUrl") or self.default_base).rstrip("/")
        url = f"{base}/v1/messages"
        headers = {
            "x-api-key": self.api_key,
            "anthropic-version": "2023-06-01",
        }
        if self.pconf.get("anthropicBetaHeader"):
            headers["anthropic-beta"] = self.pconf["anthropicBetaHeader"]
        system = ""
        wire_msgs: List[Dict[str, Any]] = []
        for m in messages:
            if m.role == "system":
                system = (system + "\n\n" + m.content).strip() if system else m.content
                continue
            if m.role == "tool":
                wire_msgs.append(
                    {
                        "role": "user",
                        "content": [
                            {
                                "type": "tool_result",
                                "tool_use_id": m.tool_call_id,
                                "content": m.content or "",
                            }
                        ],
                    }
                )
                continue
            if m.role == "assistant" and m.tool_calls:
                blocks: List[Dict[str, Any]] = []
                if m.content:
                    blocks.append({"type": "text", "text": m.content})
                for tc in m.tool_calls:
                    try:
                        args = json.loads(tc.function.arguments or "{}")
                    except json.JSONDecodeError:
                        args = {}
                    blocks.append(
                        {
                            "type": "tool_use",
                            "id": tc.id,
                            "name": tc.function.name,
                            "input": args,
                        }
                    )
                wire_msgs.append({"role": "assistant", "content": blocks})
                continue
            wire_msgs.append(
                {"role": "assistant" if m.role == "assistant" else "user",
                 "content": m.content or ""}
            )
        body: Dict[str, Any] = {
            "model": self.params.get("model", ""),
            "max_tokens": int(self.params.get("maxTokens") or 1024),
            "messages": wire_msgs,
        }
        if system:
            body["system"] = system
        if tools:
            body["tools"] = [
                {
                    "name": t.function.name,
                    "description": t.function.description,
                    "input_schema": t.function.parameters
                    or {"type": "object", "properties": {}},
                }
                for t in tools
            ]
        v = _f(self.params.get("temperature"))
        if v is not None:
            body["temperature"] = v
        v = _f(self.params.get("topP"))
        if v is not None:
            body["top_p"] = v
        if self.params.get("topK") is not None:
            body["top_k"] = int(self.params["topK"])
        return url, headers, body

    def _parse(self, body):
        content = ""
        tool_calls: List[MessageToolCall] = []
        for block in body.get("content", []) or []:
            if block.get("type") == "text" and not content:
                content = block.get("text", "")
            elif block.get("type") == "tool_use":
                tool_calls.append(
                    MessageToolCall(
                        id=block.get("id", ""),
                        function=ToolCallFunction(
                            name=block.get("name", ""),
                            arguments=json.dumps(block.get("input", {}) or {}),
                        ),
                    )
                )
        return _winning_message(content, tool_calls)


class GoogleClient(RemoteLLMClient):
    """Gemini generateContent wire (generativelanguage REST)."""

    provider = "google"
    default_base = "https://generativelanguage.googleapis.com/v1beta"

    def _auth(self, url: str) -> tuple:
        return f"{url}?key={self.api_key}", {}

    def _build(self, messages, tools):
        base = (self.params.get("baseUrl") or self.default_base).rstrip("/")
        model = self.params.get("model", "gemini-pro")
        url = f"{base}/models/{model}:generateContent"
        url, headers = self._auth(url)
        system_parts: List[Dict[str, Any]] = []
        contents: List[Dict[str, Any]] = []
        for m in messages:
            if m.role == "system":
                system_parts.append({"text": m.content or ""})
                continue
            if m.role == "tool":
                contents.append(
                    {
                        "role": "user",
                        "parts": [
                            {
                                "functionResponse": {
                                    "name": m.name or m.tool_call_id,
                                    "response": {"content": m.content or ""},
                                }
                            }
                        ],
                    }
                )
                continue
            role = "model" if m.role == "assistant" else "user"
            parts: List[Dict[str, Any]] = []
            if m.content:
                parts.append({"text": m.content})
            for tc in m.tool_calls:
                try:
                    args = json.loads(tc.function.arguments or "{}")
                except json.JSONDecodeError:
                    args = {}
                parts.append({"functionCall": {"name": tc.function.name, "args": args}})
            contents.append({"role": role, "parts": parts or [{"text": ""}]})
        body: Dict[str, Any] = {"contents": contents}
        if system_parts:
            body["systemInstruction"] = {"parts": system_parts}
        if tools:
            body["tools"] = [
                {
                    "functionDeclarations": [
                        {
                            "name": t.function.name,
                            "description": t.function.description,
                            "parameters": t.function.parameters
                            or {"type": "object", "properties": {}},
                        }
                        for t in tools
                    ]
                }
            ]
        gen: Dict[str, Any] = {}
        if self.params.get("maxTokens") is not None:
            gen["maxOutputTokens"] = int(self.params["maxTokens"])
        v = _f(self.params.get("temperature"))
        if v is not None:
            gen["temperature"] = v
        v = _f(self.params.get("topP"))
        if v is not None:
            gen["topP"] = v
        if self.params.get("topK") is not None:
            gen["topK"] = int(self.params["topK"])
        if gen:
            body["generationConfig"] = gen
        return url, headers, body

    def _parse(self, body):
        content = ""
        tool_calls: List[MessageToolCall] = []
        n = 0
        for cand in body.get("candidates", []) or []:
            for part in (cand.get("content", {}) or {}).get("parts", []) or []:
                if part.get("text") and not content:
                    content = part["text"]
                fc = part.get("functionCall")
                if fc:
                    n += 1
                    tool_calls.append(
                        MessageToolCall(
                            id=f"gc-{n}",
                            function=ToolCallFunction(
                                name=fc.get("name", ""),
                                arguments=json.dumps(fc.get("args", {}) or {}),
                            ),
                        )
                    )
        return _winning_message(content, tool_calls)


class VertexClient(GoogleClient):
    """Vertex AI: same generateContent shape, bearer credentials and a
    project/location endpoint (langchaingo_client.go:65-71 feeds the
    credentials JSON as the api key; here the key doubles as the bearer
    token since there is no egress to exchange credentials)."""

    provider = "vertex"

    def _auth(self, url: str) -> tuple:
        return url, {"Authorization": f"Bearer {self.api_key}"}

    def _build(self, messages, tools):
        if not self.params.get("baseUrl"):
            project = self.pconf.get("cloudProject", "")
            location = self.pconf.get("cloudLocation", "us-central1")
            self.params = dict(self.params)
            self.params["baseUrl"] = (
                f"https://{location}-aiplatform.googleapis.com/v1/projects/"
                f"{project}/locations/{location}/publishers/google"
            )
        return super()._build(messages, tools)


PROVIDER_CLIENTS = {
    "openai": OpenAIClient,
    "anthropic": AnthropicClient,
    "mistral": MistralClient,
    "google": GoogleClient,
    "vertex": VertexClient,
}


def create_remote_client(provider: str, llm_spec: Dict[str, Any], api_key: str,
                         transport=None) -> RemoteLLMClient:
    cls = PROVIDER_CLIENTS.get(provider)
    if cls is None:
        raise LLMRequestError(400, f"unsupported provider: {provider!r}")
    params = llm_spec.get("parameters", {}) or {}
    pconf = llm_spec.get(provider, {}) or {}
    return cls(params, api_key, pconf, transport=transport)
