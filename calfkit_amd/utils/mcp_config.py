"""mcp.json loading (capability parity: /root/reference/CHANGELOG.md:347 —
``mcp.json`` with a published JSON Schema).

Format (the de-facto standard):

    {"mcpServers": {
        "fs":  {"command": "mcp-fs", "args": ["--root", "."], "env": {...}},
        "web": {"url": "http://localhost:3000/sse", "headers": {...}}
    }}
"""

from __future__ import annotations

import json
from pathlib import Path

from ..definitions.toolbox import MCPToolbox

__all__ = ["load_mcp_config", "MCP_JSON_SCHEMA"]

MCP_JSON_SCHEMA = {
    "$schema": "https://json-schema.org/draft/2020-12/schema",
    "type": "object",
    "properties": {
        "mcpServers": {
            "type": "object",
            "additionalProperties": {
                "type": "object",
                "properties": {
                    "command": {"type": "string"},
                    "args": {"type": "array", "items": {"type": "string"}},
                    "env": {"type": "object",
                            "additionalProperties": {"type": "string"}},
                    "url": {"type": "string"},
                    "headers": {"type": "object",
                                "additionalProperties": {"type": "string"}},
                },
            },
        }
    },
    "required": ["mcpServers"],
}


def load_mcp_config(path: str | Path) -> list[MCPToolbox]:
    """Parse an mcp.json into MCPToolbox definitions (pure config)."""
    data = json.loads(Path(path).read_text())
    servers = data.get("mcpServers")
    if not isinstance(servers, dict):
        raise ValueError(f"{path}: expected an object under 'mcpServers'")
    boxes: list[MCPToolbox] = []
    for name, spec in servers.items():
        if not isinstance(spec, dict):
            raise ValueError(f"{path}: server {name!r} must be an object")
        if spec.get("url"):
            transport = "sse" if spec["url"].endswith("/sse") else "http"
            boxes.append(MCPToolbox(name=name, transport=transport,
                                    url=spec["url"],
                                    headers=dict(spec.get("headers", {}))))
        elif spec.get("command"):
            boxes.append(MCPToolbox(name=name, transport="stdio",
                                    command=spec["command"],
                                    args=tuple(spec.get("args", ())),
                                    env=dict(spec.get("env", {}))))
        else:
            raise ValueError(f"{path}: server {name!r} needs 'command' or 'url'")
    return boxes
