"""Egress chokepoint: persist-before-publish; the fault funnel.

Capability parity: reference ``_wire/publisher.py``
(/root/reference/src/calfkit/_wire/publisher.py:1-2) names
``TransportPublisher``, ``BoundPublisher`` (domain-shaped verb set),
``FaultFunnel``.  Persist-before-publish is the crash-consistency contract:
the thread row is flushed before any effect leaves the node, so a crash
between persist and publish re-emits rather than loses
(/root/reference/SURVEY analog: _carriage store-first + this chokepoint).

``max_message_bytes`` enforcement (/root/reference/CHANGELOG.md:24) and
oversized-fault elision (/root/reference/CHANGELOG.md:36) both live here —
every byte leaving a node passes this module.
"""

from __future__ import annotations

import json
from typing import Any

from ..errors import ErrorReport
from ..messages import Message
from ..ports import MessageBus
from ..streaming import StreamChunk
from . import topology
from .deliveries import (
    EchoMarker,
    ReplySlot,
    agent_call_payload,
    continuation_payload,
    encode_envelope,
    fault_payload,
    tool_call_payload,
)

__all__ = ["TransportPublisher", "BoundPublisher", "FaultFunnel"]


class TransportPublisher:
    """The one place envelopes are encoded and handed to the bus."""

    def __init__(self, bus: MessageBus, *, namespace: str = "",
                 max_message_bytes: int = 1024 * 1024):
        self.bus = bus
        self.namespace = namespace
        self.max_message_bytes = max_message_bytes

    def bind(self, node: str) -> "BoundPublisher":
        return BoundPublisher(self, node)

    async def send(self, kind: str, topic: str, key: str, *, to: str, thread_id: str,
                   payload: dict[str, Any], reply: ReplySlot | None = None,
                   marker: EchoMarker | None = None, emitter: str = "",
                   hop: int = 0) -> None:
        value, headers = encode_envelope(kind, to=to, thread_id=thread_id,
                                         payload=payload, reply=reply, marker=marker,
                                         emitter=emitter, hop=hop)
        if len(value) > self.max_message_bytes:
            if kind == "fault":
                report = ErrorReport.from_wire(payload.get("report", {}))
                payload = fault_payload(report.elide_if_oversized(len(value),
                                                                  self.max_message_bytes))
                value, headers = encode_envelope(kind, to=to, thread_id=thread_id,
                                                 payload=payload, reply=reply,
                                                 marker=marker, emitter=emitter, hop=hop)
            else:
                raise ValueError(
                    f"envelope for {topic} is {len(value)}B > "
                    f"max_message_bytes={self.max_message_bytes}")
        await self.bus.publish(topic, key, value, headers)


class BoundPublisher:
    """Domain-shaped verbs, bound to one emitting node."""

    def __init__(self, transport: TransportPublisher, node: str):
        self._t = transport
        self.node = node

    # ------------------------------------------------------------- verbs
    async def call_tool(self, *, inbox: str, thread_id: str, tool_name: str,
                        arguments: dict[str, Any], marker: EchoMarker,
                        reply: ReplySlot, hop: int = 0) -> None:
        await self._t.send("tool_call", inbox, thread_id, to=inbox,
                           thread_id=thread_id,
                           payload=tool_call_payload(tool_name, arguments),
                           reply=reply, marker=marker, emitter=self.node, hop=hop)

    async def message_peer(self, *, inbox: str, thread_id: str, message: Message,
                           reply: ReplySlot | None, marker: EchoMarker | None = None,
                           hop: int = 0) -> None:
        await self._t.send("agent_call", inbox, thread_id, to=inbox,
                           thread_id=thread_id,
                           payload=agent_call_payload(message),
                           reply=reply, marker=marker, emitter=self.node, hop=hop)

    async def handoff(self, *, inbox: str, thread_id: str, message: Message,
                      seed: tuple[Message, ...], reason: str,
                      reply: ReplySlot | None, marker: EchoMarker | None = None,
                      hop: int = 0) -> None:
        await self._t.send(
            "agent_call", inbox, thread_id, to=inbox, thread_id=thread_id,
            payload=agent_call_payload(
                message, handoff={"from": self.node, "reason": reason}, seed=seed),
            reply=reply, marker=marker, emitter=self.node, hop=hop)

    async def reply_continuation(self, *, reply: ReplySlot, thread_id: str,
                                 marker: EchoMarker, content: Any = None,
                                 is_error: bool = False,
                                 message: Message | None = None, hop: int = 0) -> None:
        """Reply on the echoed marker to the caller's private inbox.

        The envelope's thread identity is the CALLER's grain (reply.key):
        the caller resumes its own thread row, whatever thread the callee
        ran the work on.
        """
        await self._t.send("continuation", reply.inbox, reply.key or thread_id,
                           to=reply.node, thread_id=reply.key or thread_id,
                           payload=continuation_payload(content, is_error=is_error,
                                                        message=message),
                           marker=marker, emitter=self.node, hop=hop)

    async def emit_event(self, *, event: str, thread_id: str, data: dict[str, Any],
                         hop: int = 0) -> None:
        topic = topology.event_topic(event, self._t.namespace)
        await self._t.send("agent_call", topic, thread_id or event, to=topic,
                           thread_id=thread_id,
                           payload=agent_call_payload(_event_message(event, data)),
                           emitter=self.node, hop=hop)

    async def stream_chunk(self, *, thread_id: str, chunk: StreamChunk,
                           firehose: bool = True,
                           reply: ReplySlot | None = None) -> None:
        value = json.dumps(chunk.to_wire(), separators=(",", ":")).encode()
        headers = ((
            "x-calf-kind", b"chunk"), ("x-calf-emitter", self.node.encode()))
        if firehose:
            await self._t.bus.publish(
                topology.firehose_topic(thread_id, self._t.namespace),
                thread_id, value, headers)
        if reply is not None and reply.inbox:
            await self._t.bus.publish(reply.inbox, reply.key or thread_id, value, headers)

    async def fault(self, *, report: ErrorReport, thread_id: str,
                    reply: ReplySlot | None, marker: EchoMarker | None = None,
                    hop: int = 0) -> None:
        if reply is None or not reply.inbox:
            # no return address: the ledger topic is the dead-letter rail
            topic = topology.ledger_topic(self._t.namespace)
            key = thread_id or self.node
            to = ""
        else:
            topic, key, to = reply.inbox, reply.key or thread_id, reply.node
        await self._t.send("fault", topic, key, to=to, thread_id=thread_id,
                           payload=fault_payload(report), marker=marker,
                           emitter=self.node, hop=hop)


class FaultFunnel:
    """Every egress fault goes through here — one choke point, typed."""

    def __init__(self, publisher: BoundPublisher, kind: str):
        self._pub = publisher
        self._kind = kind

    async def emit(self, exc: BaseException, *, thread_id: str,
                   reply: ReplySlot | None, marker: EchoMarker | None = None,
                   **run_state: Any) -> ErrorReport:
        report = ErrorReport.from_exception(self._pub.node, self._kind, exc, **run_state)
        await self._pub.fault(report=report, thread_id=thread_id, reply=reply,
                              marker=marker)
        return report


def _event_message(event: str, data: dict[str, Any]) -> Message:
    from ..messages import TextPart, UserMessage

    return UserMessage((TextPart(json.dumps({"event": event, "data": data})),),
                       author="event")
