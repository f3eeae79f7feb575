"""Sole importer of the official MCP SDK.

Capability parity: reference ``adapters/mcp.py``
(/root/reference/src/calfkit/adapters/mcp.py:1): ``OfficialMCPSession``.
The MCP SDK is not installed in this image (no network), so opening a real
session raises a clear error; scripted sessions (``MCPToolbox(session=...)``
with ``calfkit_amd.testing.FakeMCPSession``) cover the toolbox path.
"""

from __future__ import annotations

from typing import Any

from ..definitions.toolbox import MCPToolbox
from ..mcp import MCPResult, MCPSession, MCPToolInfo

__all__ = ["OfficialMCPSession", "open_session"]


class OfficialMCPSession:
    """Adapter from the official ``mcp`` SDK session to the MCPSession port."""

    def __init__(self, sdk_session: Any):
        self._s = sdk_session

    async def list_tools(self) -> list[MCPToolInfo]:
        resp = await self._s.list_tools()
        return [MCPToolInfo(name=t.name, description=t.description or "",
                            input_schema=dict(getattr(t, "inputSchema", {}) or {}))
                for t in resp.tools]

    async def call_tool(self, name: str, arguments: dict[str, Any]) -> MCPResult:
        resp = await self._s.call_tool(name, arguments)
        content = [getattr(c, "text", str(c)) for c in resp.content]
        return MCPResult(content="\n".join(content),
                         is_error=bool(getattr(resp, "isError", False)))

    async def close(self) -> None:
        close = getattr(self._s, "close", None) or getattr(self._s, "aclose", None)
        if close is not None:
            await close()


async def open_session(toolbox: MCPToolbox) -> MCPSession:
    """Dial an MCP server per the toolbox config."""
    try:
        import mcp  # noqa: F401
    except ImportError as e:
        raise RuntimeError(
            f"toolbox {toolbox.name!r}: the official MCP SDK is not installed; "
            "pass a pre-built session via MCPToolbox(session=...)") from e
    raise NotImplementedError(
        "official MCP transport dialing lands with the SDK available")
