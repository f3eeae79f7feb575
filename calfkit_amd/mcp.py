"""SDK-free MCP vocabulary: the session port and result types.

Capability parity: reference ``mcp.py`` (/root/reference/src/calfkit/mcp.py:1-2)
names ``MCPSession``, ``MCPToolInfo``, ``MCPResult``.  The official MCP SDK is
quarantined in ``adapters/mcp.py``; the core speaks only these shapes.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Protocol, runtime_checkable

__all__ = ["MCPToolInfo", "MCPResult", "MCPSession"]


@dataclass(frozen=True)
class MCPToolInfo:
    name: str
    description: str = ""
    input_schema: dict[str, Any] = field(default_factory=dict)


@dataclass(frozen=True)
class MCPResult:
    content: Any = None
    is_error: bool = False


@runtime_checkable
class MCPSession(Protocol):
    """A live connection to one MCP server."""

    async def list_tools(self) -> list[MCPToolInfo]: ...

    async def call_tool(self, name: str, arguments: dict[str, Any]) -> MCPResult: ...

    async def close(self) -> None: ...
