"""MCPToolboxProcessor — proxy for an MCP server connection.

Capability parity: reference ``_processors/mcp.py``
(/root/reference/src/calfkit/_processors/mcp.py:1-2): owns the live session
via the lifecycle pair; MCP toolboxes as mesh nodes with namespaced tool
names per /root/reference/CHANGELOG.md:159,160.
"""

from __future__ import annotations

from typing import Any

from .._hosting import processes
from .._wire import topology
from .._wire.deliveries import Delivery, ToolInvoke
from .._wire.machinery import NodeMachinery
from .._wire.publisher import FaultFunnel
from ..definitions.toolbox import MCPToolbox
from ..errors import DecodeError
from ..mcp import MCPSession
from ..ports import RawDelivery

__all__ = ["MCPToolboxProcessor"]


@processes("toolbox")
class MCPToolboxProcessor:
    def __init__(self, toolbox: MCPToolbox, machinery: NodeMachinery):
        self.toolbox = toolbox
        self.machinery = machinery
        self.node_name = toolbox.name
        self.publisher = machinery.transport.bind(toolbox.name)
        self.funnel = FaultFunnel(self.publisher, "mcp")
        self.session: MCPSession | None = None
        self._tool_infos: list[Any] = []

    def topics(self) -> list[str]:
        return [topology.inbox_topic(self.node_name, self.machinery.namespace)]

    async def start(self) -> None:
        """Open the live session (the lifecycle bracket's startup half)."""
        if self.toolbox.session is not None:
            self.session = self.toolbox.session
        else:
            from ..adapters.mcp import open_session  # sole MCP-SDK importer
            self.session = await open_session(self.toolbox)
        self._tool_infos = list(await self.session.list_tools())

    async def stop(self) -> None:
        if self.session is not None and self.toolbox.session is None:
            await self.session.close()
        self.session = None

    def advert_capabilities(self) -> dict[str, Any]:
        """The toolbox advert carries its (un-namespaced) tool schemas."""
        return {"tools": [
            {"name": t.name, "description": t.description, "parameters": t.input_schema}
            for t in self._tool_infos
        ]}

    async def handle(self, raw: RawDelivery) -> None:
        ledger = self.machinery.ledger(self.node_name, raw.key, "mcp")
        try:
            delivery: Delivery = self.machinery.decoder.decode(raw)
        except DecodeError as e:
            ledger.note(outcome="decode_floor", error=str(e))
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        if not isinstance(delivery, ToolInvoke):
            ledger.note(outcome="ignored")
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        # namespaced names arrive as "<box>.<tool>"
        name = delivery.tool_name
        if name.startswith(self.node_name + "."):
            name = name[len(self.node_name) + 1:]
        try:
            if self.session is None:
                raise RuntimeError(f"toolbox {self.node_name} has no live session")
            result = await self.session.call_tool(name, delivery.arguments)
            await self.publisher.reply_continuation(
                reply=delivery.reply, thread_id=delivery.thread_id,
                marker=delivery.marker, content=result.content,
                is_error=result.is_error)
            ledger.note(outcome="error" if result.is_error else "ok", tool=name)
        except Exception as exc:
            await self.funnel.emit(exc, thread_id=delivery.thread_id,
                                   reply=delivery.reply, marker=delivery.marker,
                                   tool=name)
            ledger.note(outcome="fault", error=str(exc))
        await ledger.flush(self.machinery.bus, self.machinery.namespace)
