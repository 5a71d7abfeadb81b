"""LLM client factory — the reconciler seam (factory.go:9-11).

The reference's factory returns a langchaingo client per provider; here the
providers are:

- ``mock``   deterministic scripted client (tests, config-1 bench)
- ``local``  the in-process MI355X inference engine (the whole point)
- openai/anthropic/mistral/google/vertex — recognized for spec parity; they
  validate config shape but cannot reach the network in this environment, so
  their ``send_request`` raises a 502-class LLMRequestError unless a
  ``transport`` is injected (tests inject an httpx mock).
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional

from .base import LLMClient, LLMRequestError

KNOWN_PROVIDERS = ("openai", "anthropic", "mistral", "google", "vertex", "mock", "local")


class _RemoteStubClient(LLMClient):
    def __init__(self, provider: str, parameters: Dict[str, Any], api_key: str,
                 transport: Optional[Callable] = None):
        self.provider = provider
        self.parameters = parameters
        self.api_key = api_key
        self.transport = transport

    def send_request(self, messages, tools):
        if self.transport is not None:
            return self.transport(self.provider, self.parameters, messages, tools)
        raise LLMRequestError(
            502, f"remote provider {self.provider!r} is unreachable in this deployment"
        )


class LLMClientFactory:
    """create_client(llm_resource, api_key) -> LLMClient.

    ``engine_provider`` is a callable returning the shared engine-backed
    client (injected by the process wiring so one engine serves every
    reconciler worker); ``mock_factory`` can be overridden by tests.
    """

    def __init__(
        self,
        engine_provider: Optional[Callable[[Dict[str, Any]], LLMClient]] = None,
        mock_factory: Optional[Callable[[Dict[str, Any]], LLMClient]] = None,
        remote_transport: Optional[Callable] = None,
    ):
        self._engine_provider = engine_provider
        self._mock_factory = mock_factory
        self._remote_transport = remote_transport

    def create_client(self, llm: Dict[str, Any], api_key: str = "") -> LLMClient:
        provider = llm.get("spec", {}).get("provider", "")
        if provider not in KNOWN_PROVIDERS:
            raise LLMRequestError(400, f"unsupported provider: {provider!r}")
        if provider == "mock":
            if self._mock_factory is not None:
                return self._mock_factory(llm)
            from .mock import MockLLMClient

            return MockLLMClient()
        if provider == "local":
            if self._engine_provider is None:
                raise LLMRequestError(
                    503, "local provider requested but no engine is attached to this manager"
                )
            return self._engine_provider(llm)
        return _RemoteStubClient(
            provider,
            llm.get("spec", {}).get("parameters", {}) or {},
            api_key,
            transport=self._remote_transport,
        )
