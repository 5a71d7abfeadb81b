"""MCP tool → LLM tool schema adapter.

Parity with acp/internal/adapters/mcp_adapter.go:12-51: discovered MCP tools
become OpenAI-style function tools named ``<server>__<tool>`` so one flat
namespace routes back to the owning server (mcpmanager.go:304-331).
"""
from __future__ import annotations

from typing import Any, Dict, List

from ..api.types import ToolType
from ..llmclient.base import Tool, ToolFunction


def convert_mcp_tools_to_llm_tools(
    mcp_tools: List[Dict[str, Any]], server_name: str
) -> List[Tool]:
    out: List[Tool] = []
    for t in mcp_tools or []:
        schema = t.get("inputSchema") or {"type": "object"}
        out.append(
            Tool(
                function=ToolFunction(
                    name=f'{server_name}__{t.get("name", "")}',
                    description=t.get("description", "") or t.get("name", ""),
                    parameters=schema,
                ),
                acp_tool_type=ToolType.MCP,
            )
        )
    return out
