"""MCP sessions: an SDK-free stdio JSON-RPC client + the SDK adapter.

Capability parity: reference ``adapters/mcp.py``
(/root/reference/src/calfkit/adapters/mcp.py:1): ``OfficialMCPSession``
wraps the official SDK when it exists.  This image is offline (no SDK), so
the stdio transport is implemented DIRECTLY: MCP's stdio framing is
newline-delimited JSON-RPC 2.0 (initialize -> notifications/initialized ->
tools/list / tools/call), which needs no SDK at all.  ``open_session``
dials stdio natively and falls back to the SDK for sse/http transports.
"""

from __future__ import annotations

import asyncio
import json
import logging
from typing import Any

from ..definitions.toolbox import MCPToolbox
from ..mcp import MCPResult, MCPSession, MCPToolInfo

__all__ = ["StdioMCPSession", "SseMCPSession", "OfficialMCPSession",
           "open_session"]

_log = logging.getLogger("calfkit.mcp")

PROTOCOL_VERSION = "2024-11-05"


class StdioMCPSession:
    """A live MCP server over stdio, spoken directly (JSON-RPC 2.0,
    newline-delimited frames per the MCP stdio transport spec)."""

    def __init__(self, proc: asyncio.subprocess.Process):
        self._p = proc
        self._id = 0
        self._pending: dict[int, asyncio.Future] = {}
        self._pump = asyncio.get_running_loop().create_task(self._read_loop())

    # ------------------------------------------------------------- transport
    @classmethod
    async def open(cls, command: str, args: tuple[str, ...] = (),
                   timeout: float = 30.0,
                   env: dict[str, str] | None = None) -> "StdioMCPSession":
        import os

        proc = await asyncio.create_subprocess_exec(
            command, *args,
            stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE,
            env={**os.environ, **env} if env else None)
        self = cls(proc)
        await asyncio.wait_for(self._initialize(), timeout)
        return self

    async def _initialize(self) -> None:
        result = await self._request("initialize", {
            "protocolVersion": PROTOCOL_VERSION,
            "capabilities": {},
            "clientInfo": {"name": "calfkit-amd", "version": "0.2"},
        })
        self.server_info = result.get("serverInfo", {})
        self._notify("notifications/initialized", {})

    async def _read_loop(self) -> None:
        assert self._p.stdout is not None
        try:
            while True:
                line = await self._p.stdout.readline()
                if not line:
                    break
                line = line.strip()
                if not line:
                    continue
                try:
                    msg = json.loads(line)
                except json.JSONDecodeError:
                    _log.warning("mcp: undecodable frame %r", line[:200])
                    continue
                mid = msg.get("id")
                fut = self._pending.pop(mid, None) if mid is not None else None
                if fut is not None and not fut.done():
                    if "error" in msg:
                        fut.set_exception(RuntimeError(
                            f"mcp error {msg['error']}"))
                    else:
                        fut.set_result(msg.get("result", {}))
        except asyncio.CancelledError:
            pass
        finally:
            for fut in self._pending.values():
                if not fut.done():
                    fut.set_exception(ConnectionError("mcp server closed"))

    def _send(self, obj: dict[str, Any]) -> None:
        assert self._p.stdin is not None
        self._p.stdin.write(json.dumps(obj, separators=(",", ":")).encode()
                            + b"\n")

    def _notify(self, method: str, params: dict[str, Any]) -> None:
        self._send({"jsonrpc": "2.0", "method": method, "params": params})

    async def _request(self, method: str, params: dict[str, Any],
                       timeout: float = 60.0) -> dict[str, Any]:
        self._id += 1
        mid = self._id
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[mid] = fut
        self._send({"jsonrpc": "2.0", "id": mid, "method": method,
                    "params": params})
        assert self._p.stdin is not None
        await self._p.stdin.drain()
        return await asyncio.wait_for(fut, timeout)

    # ------------------------------------------------------------------ port
    async def list_tools(self) -> list[MCPToolInfo]:
        result = await self._request("tools/list", {})
        return [MCPToolInfo(name=t["name"],
                            description=t.get("description", ""),
                            input_schema=dict(t.get("inputSchema", {}) or {}))
                for t in result.get("tools", [])]

    async def call_tool(self, name: str, arguments: dict[str, Any]) -> MCPResult:
        result = await self._request("tools/call",
                                     {"name": name, "arguments": arguments})
        parts = [c.get("text", str(c)) for c in result.get("content", [])]
        return MCPResult(content="\n".join(parts),
                         is_error=bool(result.get("isError", False)))

    async def close(self) -> None:
        self._pump.cancel()
        if self._p.stdin is not None:
            try:
                self._p.stdin.close()
            except Exception:
                pass
        try:
            await asyncio.wait_for(self._p.wait(), 5)
        except (asyncio.TimeoutError, ProcessLookupError):
            try:
                self._p.kill()
            except ProcessLookupError:
                pass


class SseMCPSession:
    """A live MCP server over the SSE transport, spoken directly.

    Wire shape (MCP 2024-11-05 SSE transport): GET {url} as an event
    stream; the server's first ``endpoint`` event names the POST target;
    JSON-RPC requests POST there and responses arrive as ``message``
    events on the stream.  httpx is imported in-function (HTTP adapters
    are httpx owners per the layering contract).
    """

    def __init__(self, client: Any, post_url: str):
        self._client = client
        self._post_url = post_url
        self._id = 0
        self._pending: dict[int, asyncio.Future] = {}
        self._pump: asyncio.Task | None = None
        self.server_info: dict[str, Any] = {}

    @classmethod
    async def open(cls, url: str, *, headers: dict[str, str] | None = None,
                   timeout: float = 30.0, transport: Any = None
                   ) -> "SseMCPSession":
        import httpx

        kw: dict[str, Any] = {"timeout": timeout, "base_url": url}
        if transport is not None:
            kw["transport"] = transport
        client = httpx.AsyncClient(headers=headers or {}, **kw)
        endpoint_ready: asyncio.Future = asyncio.get_running_loop().create_future()
        self = cls(client, "")

        async def pump() -> None:
            try:
                async with client.stream("GET", url, headers={
                        "Accept": "text/event-stream"}) as resp:
                    resp.raise_for_status()
                    event, data = "", []
                    async for line in resp.aiter_lines():
                        if line.startswith("event:"):
                            event = line[6:].strip()
                        elif line.startswith("data:"):
                            data.append(line[5:].strip())
                        elif line == "":
                            body = "\n".join(data)
                            if event == "endpoint" and not endpoint_ready.done():
                                endpoint_ready.set_result(body)
                            elif event in ("message", "") and body:
                                self._dispatch(body)
                            event, data = "", []
            except asyncio.CancelledError:
                pass
            except Exception as e:
                if not endpoint_ready.done():
                    endpoint_ready.set_exception(e)
            finally:
                for fut in self._pending.values():
                    if not fut.done():
                        fut.set_exception(ConnectionError("mcp sse closed"))

        self._pump = asyncio.get_running_loop().create_task(pump())
        self._post_url = await asyncio.wait_for(endpoint_ready, timeout)
        await asyncio.wait_for(self._initialize(), timeout)
        return self

    def _dispatch(self, body: str) -> None:
        try:
            msg = json.loads(body)
        except json.JSONDecodeError:
            _log.warning("mcp sse: undecodable frame %r", body[:200])
            return
        mid = msg.get("id")
        fut = self._pending.pop(mid, None) if mid is not None else None
        if fut is not None and not fut.done():
            if "error" in msg:
                fut.set_exception(RuntimeError(f"mcp error {msg['error']}"))
            else:
                fut.set_result(msg.get("result", {}))

    async def _initialize(self) -> None:
        result = await self._request("initialize", {
            "protocolVersion": PROTOCOL_VERSION,
            "capabilities": {},
            "clientInfo": {"name": "calfkit-amd", "version": "0.2"},
        })
        self.server_info = result.get("serverInfo", {})
        await self._client.post(self._post_url, json={
            "jsonrpc": "2.0", "method": "notifications/initialized",
            "params": {}})

    async def _request(self, method: str, params: dict[str, Any],
                       timeout: float = 60.0) -> dict[str, Any]:
        self._id += 1
        mid = self._id
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[mid] = fut
        resp = await self._client.post(self._post_url, json={
            "jsonrpc": "2.0", "id": mid, "method": method, "params": params})
        resp.raise_for_status()
        return await asyncio.wait_for(fut, timeout)

    async def list_tools(self) -> list[MCPToolInfo]:
        result = await self._request("tools/list", {})
        return [MCPToolInfo(name=t["name"],
                            description=t.get("description", ""),
                            input_schema=dict(t.get("inputSchema", {}) or {}))
                for t in result.get("tools", [])]

    async def call_tool(self, name: str, arguments: dict[str, Any]) -> MCPResult:
        result = await self._request("tools/call",
                                     {"name": name, "arguments": arguments})
        parts = [c.get("text", str(c)) for c in result.get("content", [])]
        return MCPResult(content="\n".join(parts),
                         is_error=bool(result.get("isError", False)))

    async def close(self) -> None:
        if self._pump is not None:
            self._pump.cancel()
        try:
            await self._client.aclose()
        except Exception:
            pass


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
    """Dial an MCP server per the toolbox config.

    stdio and sse servers are dialed natively (no SDK needed — the stdio
    framing is newline JSON-RPC and sse is an event stream over httpx);
    the http transport still needs the official SDK.
    """
    if toolbox.transport == "stdio":
        if not toolbox.command:
            raise ValueError(
                f"toolbox {toolbox.name!r}: stdio transport needs command=")
        return await StdioMCPSession.open(toolbox.command, toolbox.args,
                                          env=dict(toolbox.env) or None)
    if toolbox.transport == "sse":
        if not toolbox.url:
            raise ValueError(
                f"toolbox {toolbox.name!r}: sse transport needs url=")
        return await SseMCPSession.open(toolbox.url,
                                        headers=dict(toolbox.headers))
    try:
        import mcp  # noqa: F401
    except ImportError as e:
        raise RuntimeError(
            f"toolbox {toolbox.name!r}: transport {toolbox.transport!r} needs "
            "the official MCP SDK (not installed); use a stdio/sse server or "
            "pass a pre-built session via MCPToolbox(session=...)") from e
    raise NotImplementedError(
        "official MCP streamable-http dialing lands with the SDK available")
