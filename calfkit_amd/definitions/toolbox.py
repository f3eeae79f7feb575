"""MCPToolbox — the pure connection config, never the connection.

Capability parity: reference ``definitions/toolbox.py``
(/root/reference/src/calfkit/definitions/toolbox.py:1-2); MCP toolboxes as
mesh nodes per /root/reference/CHANGELOG.md:159.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

from .refs import validate_node_name

__all__ = ["MCPToolbox"]


@dataclass
class MCPToolbox:
    """Config for an MCP server proxied onto the mesh as a toolbox node.

    ``transport`` ∈ {"stdio", "sse", "http"}.  The live session is owned by
    the MCPToolboxProcessor via the lifecycle bracket; this object is data.
    """

    name: str
    transport: str = "stdio"
    command: str = ""
    args: tuple[str, ...] = ()
    url: str = ""
    env: dict[str, str] = field(default_factory=dict)
    headers: dict[str, str] = field(default_factory=dict)
    #: pre-scripted session for tests (anything satisfying MCPSession)
    session: Any = None
    kind: str = "toolbox"

    def __post_init__(self) -> None:
        validate_node_name(self.name)
        if self.transport not in ("stdio", "sse", "http"):
            raise ValueError(f"unknown MCP transport {self.transport!r}")
