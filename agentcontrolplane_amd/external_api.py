"""Pluggable external tool-client registry.

Parity with acp/internal/externalAPI/main.go:32-67: factories keyed by tool
name, resolved with a per-call API key from a secret ref.  The reference
ships this as unused scaffolding; here it is wired as an optional ToolCall
executor path: a ToolCall whose toolRef matches a registered client name is
executed through the registry instead of the MCP manager.
"""
from __future__ import annotations

import threading
from typing import Any, Callable, Dict, Optional


class ExternalClient:
    """One external tool client: ``call(arguments) -> str``."""

    def call(self, arguments: Dict[str, Any]) -> str:
        raise NotImplementedError


class Registry:
    def __init__(self):
        self._lock = threading.Lock()
        self._factories: Dict[str, Callable[[str], ExternalClient]] = {}

    def register(self, tool_name: str, factory: Callable[[str], ExternalClient]) -> None:
        with self._lock:
            if tool_name in self._factories:
                raise ValueError(f"tool {tool_name!r} already registered")
            self._factories[tool_name] = factory

    def get_client(self, tool_name: str, api_key: str = "") -> Optional[ExternalClient]:
        with self._lock:
            factory = self._factories.get(tool_name)
        return factory(api_key) if factory else None

    def has(self, tool_name: str) -> bool:
        with self._lock:
            return tool_name in self._factories


default_registry = Registry()
# reference naming (externalAPI/main.go:67 DefaultRegistry)
DEFAULT_REGISTRY = default_registry
