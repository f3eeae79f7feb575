"""ToolProcessor — stateless: parse args, invoke, reply on the echoed marker.

Capability parity: reference ``_processors/tool.py``
(/root/reference/src/calfkit/_processors/tool.py:1-2); tool error surfacing
with no hangs per /root/reference/CHANGELOG.md:386; private reply routing per
/root/reference/CHANGELOG.md:393.
"""

from __future__ import annotations

from .._hosting import processes
from .._wire import topology
from .._wire.deliveries import Delivery, ToolInvoke
from .._wire.machinery import NodeMachinery
from .._wire.publisher import FaultFunnel
from ..context import ToolContext
from ..definitions.tools import ToolDef
from ..errors import DecodeError, ToolError
from ..ports import RawDelivery

__all__ = ["ToolProcessor"]


@processes("tool")
class ToolProcessor:
    def __init__(self, tooldef: ToolDef, machinery: NodeMachinery):
        self.tooldef = tooldef
        self.machinery = machinery
        self.node_name = tooldef.name
        self.publisher = machinery.transport.bind(tooldef.name)
        self.funnel = FaultFunnel(self.publisher, "tool")

    def topics(self) -> list[str]:
        return [topology.inbox_topic(self.node_name, self.machinery.namespace)]

    async def start(self) -> None:
        pass

    async def stop(self) -> None:
        pass

    async def handle(self, raw: RawDelivery) -> None:
        ledger = self.machinery.ledger(self.node_name, raw.key, "tool")
        try:
            delivery: Delivery = self.machinery.decoder.decode(raw)
        except DecodeError as e:
            ledger.note(outcome="decode_floor", error=str(e))
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        if not isinstance(delivery, ToolInvoke):
            ledger.note(outcome="ignored", got=type(delivery).__name__)
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return

        ctx = ToolContext(tool_name=delivery.tool_name, call_id=delivery.marker.call_id,
                          caller=delivery.marker.caller, thread_id=delivery.thread_id)
        try:
            result = await self.tooldef.invoke(delivery.arguments, ctx)
            await self.publisher.reply_continuation(
                reply=delivery.reply, thread_id=delivery.thread_id,
                marker=delivery.marker, content=result)
            ledger.note(outcome="ok")
        except ToolError as e:
            # retryable tool error: surfaced to the caller as an error
            # continuation (never a hang, never a node fault)
            await self.publisher.reply_continuation(
                reply=delivery.reply, thread_id=delivery.thread_id,
                marker=delivery.marker, content=str(e), is_error=True)
            ledger.note(outcome="tool_error", error=str(e))
        except Exception as exc:
            await self.funnel.emit(exc, thread_id=delivery.thread_id,
                                   reply=delivery.reply, marker=delivery.marker,
                                   tool=delivery.tool_name)
            ledger.note(outcome="fault", error=str(exc))
        await ledger.flush(self.machinery.bus, self.machinery.namespace)
