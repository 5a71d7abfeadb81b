"""LLM client factory — the reconciler seam (factory.go:9-11).

The reference's factory returns a langchaingo client per provider; here the
providers are:

- ``mock``   deterministic scripted client (tests, config-1 bench)
- ``local``  the in-process MI355X inference engine (the whole point)
- openai/anthropic/mistral/google/vertex — real httpx clients with the
  reference's message/tool conversion rules (llmclient/remote.py); there is
  no egress in this deployment, so production use targets
  ``parameters.baseUrl`` (gateway/self-hosted endpoints) and tests run
  against an in-process mock server exactly as the reference's e2e does
  (test_getting_started.go:250-261).
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional

from .base import LLMClient, LLMRequestError

KNOWN_PROVIDERS = ("openai", "anthropic", "mistral", "google", "vertex", "mock", "local")


class LLMClientFactory:
    """create_client(llm_resource, api_key) -> LLMClient.

    ``engine_provider`` is a callable returning the shared engine-backed
    client (injected by the process wiring so one engine serves every
    reconciler worker); ``mock_factory`` can be overridden by tests;
    ``http_transport`` (an httpx transport) lets tests intercept the remote
    providers without sockets.
    """

    def __init__(
        self,
        engine_provider: Optional[Callable[[Dict[str, Any]], LLMClient]] = None,
        mock_factory: Optional[Callable[[Dict[str, Any]], LLMClient]] = None,
        http_transport=None,
    ):
        self._engine_provider = engine_provider
        self._mock_factory = mock_factory
        self._http_transport = http_transport

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
        from .remote import create_remote_client

        return create_remote_client(
            provider, llm.get("spec", {}) or {}, api_key, transport=self._http_transport
        )
