from .manager import MCPConnection, MCPServerManager  # noqa: F401
from .adapter import convert_mcp_tools_to_llm_tools  # noqa: F401
