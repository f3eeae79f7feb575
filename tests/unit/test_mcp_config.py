"""Unit: mcp.json loading."""

import json

import pytest

from calfkit_amd.utils.mcp_config import load_mcp_config


def test_load_stdio_and_http(tmp_path):
    f = tmp_path / "mcp.json"
    f.write_text(json.dumps({"mcpServers": {
        "fs": {"command": "mcp-fs", "args": ["--root", "/data"],
               "env": {"X": "1"}},
        "web": {"url": "http://localhost:3000/sse"},
    }}))
    boxes = {b.name: b for b in load_mcp_config(f)}
    assert boxes["fs"].transport == "stdio"
    assert boxes["fs"].args == ("--root", "/data")
    assert boxes["web"].transport == "sse"


def test_bad_config_rejected(tmp_path):
    f = tmp_path / "mcp.json"
    f.write_text(json.dumps({"mcpServers": {"x": {}}}))
    with pytest.raises(ValueError):
        load_mcp_config(f)
    f.write_text(json.dumps({"nope": 1}))
    with pytest.raises(ValueError):
        load_mcp_config(f)
