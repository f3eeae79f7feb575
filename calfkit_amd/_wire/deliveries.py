"""Decoded inbound model — all kind-to-content knowledge lives here.

Capability parity: reference ``_wire/deliveries.py``
(/root/reference/src/calfkit/_wire/deliveries.py:1-3): the ``Delivery`` union
(Call/Continuation/Fault) and ``DeliveryDecoder``; decode floor catching
JSON/Unicode/validation errors per /root/reference/CHANGELOG.md:168; envelope
layering (routing / state / typed payload) per /root/reference/CHANGELOG.md:506;
``Envelope.reply`` slot + ``x-calf-kind`` header per
/root/reference/CHANGELOG.md:194; echo-marker call identity per
/root/reference/CHANGELOG.md:46.

Wire shape (JSON; the byte-compatibility contract of this build):

    headers: x-calf-kind: agent_call|tool_call|continuation|fault|event|chunk
             x-calf-emitter: <node>   x-calf-hop: <int>
    value: {"routing": {"to": ..., "thread_id": ..., "reply": {...}|null},
            "state":   {"marker": {...}|null, "hop": int},
            "payload": {<kind-specific>}}
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Union

from ..errors import DecodeError, ErrorReport
from ..messages import Message, message_from_wire, message_to_wire
from ..ports import RawDelivery

__all__ = [
    "KIND_HEADER",
    "EMITTER_HEADER",
    "HOP_HEADER",
    "ReplySlot",
    "EchoMarker",
    "Call",
    "ToolInvoke",
    "Continuation",
    "Fault",
    "Delivery",
    "DeliveryDecoder",
    "encode_envelope",
]

KIND_HEADER = "x-calf-kind"
EMITTER_HEADER = "x-calf-emitter"
HOP_HEADER = "x-calf-hop"


@dataclass(frozen=True)
class ReplySlot:
    """Per-delivery return address (the ``Envelope.reply`` slot)."""

    inbox: str
    key: str = ""
    node: str = ""

    def to_wire(self) -> dict[str, Any]:
        return {"inbox": self.inbox, "key": self.key, "node": self.node}

    @classmethod
    def from_wire(cls, d: dict[str, Any] | None) -> "ReplySlot | None":
        if not d:
            return None
        return cls(inbox=d.get("inbox", ""), key=d.get("key", ""), node=d.get("node", ""))


@dataclass(frozen=True)
class EchoMarker:
    """Typed call-identity carriage, echoed verbatim on the reply."""

    call_id: str
    frame_id: str = ""
    caller: str = ""

    def to_wire(self) -> dict[str, Any]:
        return {"call_id": self.call_id, "frame_id": self.frame_id, "caller": self.caller}

    @classmethod
    def from_wire(cls, d: dict[str, Any] | None) -> "EchoMarker | None":
        if not d:
            return None
        return cls(call_id=d.get("call_id", ""), frame_id=d.get("frame_id", ""),
                   caller=d.get("caller", ""))


@dataclass(frozen=True)
class Call:
    """A conversational call to an agent node."""

    thread_id: str
    message: Message
    reply: ReplySlot | None = None
    #: present when this call is a handoff seed
    handoff: dict[str, Any] | None = None
    #: seed history accompanying a handoff
    seed: tuple[Message, ...] = ()
    metadata: dict[str, Any] = field(default_factory=dict)


@dataclass(frozen=True)
class ToolInvoke:
    """A tool invocation delivered to a tool/toolbox node."""

    thread_id: str
    tool_name: str
    arguments: dict[str, Any]
    marker: EchoMarker
    reply: ReplySlot
    metadata: dict[str, Any] = field(default_factory=dict)


@dataclass(frozen=True)
class Continuation:
    """A reply completing (part of) a pending frame, echoing the marker."""

    thread_id: str
    marker: EchoMarker
    content: Any = None
    is_error: bool = False
    #: full message form when a peer agent answered
    message: Message | None = None


@dataclass(frozen=True)
class Fault:
    """A fault-rail delivery."""

    thread_id: str
    report: ErrorReport
    marker: EchoMarker | None = None


Delivery = Union[Call, ToolInvoke, Continuation, Fault]


def encode_envelope(kind: str, *, to: str, thread_id: str, payload: dict[str, Any],
                    reply: ReplySlot | None = None, marker: EchoMarker | None = None,
                    emitter: str = "", hop: int = 0) -> tuple[bytes, tuple[tuple[str, bytes], ...]]:
    """Encode one envelope; returns (value, headers)."""
    envelope = {
        "routing": {"to": to, "thread_id": thread_id,
                     "reply": reply.to_wire() if reply else None},
        "state": {"marker": marker.to_wire() if marker else None, "hop": hop},
        "payload": payload,
    }
    headers = (
        (KIND_HEADER, kind.encode()),
        (EMITTER_HEADER, emitter.encode()),
        (HOP_HEADER, str(hop).encode()),
    )
    return json.dumps(envelope, separators=(",", ":")).encode(), headers


class DeliveryDecoder:
    """bytes -> Delivery, with the decode floor.

    Route dispatch is header-based (x-calf-kind) with an envelope-payload
    fallback for header-stripping transports
    (/root/reference/CHANGELOG.md:285).
    """

    def decode(self, raw: RawDelivery) -> Delivery:
        try:
            return self._decode(raw)
        except (json.JSONDecodeError, UnicodeDecodeError, KeyError, TypeError,
                ValueError) as e:
            raise DecodeError(f"undecodable delivery on {raw.topic}: {e}") from e

    def _decode(self, raw: RawDelivery) -> Delivery:
        kind_hdr = raw.header(KIND_HEADER)
        envelope = json.loads(raw.value.decode("utf-8"))
        routing = envelope.get("routing", {})
        state = envelope.get("state", {})
        payload = envelope.get("payload", {})
        kind = kind_hdr.decode() if kind_hdr else payload.get("type", "")
        thread_id = routing.get("thread_id", "") or raw.key
        reply = ReplySlot.from_wire(routing.get("reply"))
        marker = EchoMarker.from_wire(state.get("marker"))

        if kind == "agent_call":
            metadata = dict(payload.get("metadata", {}))
            if marker is not None and "marker" not in metadata:
                metadata["marker"] = marker.to_wire()
            return Call(
                thread_id=thread_id,
                message=message_from_wire(payload["message"]),
                reply=reply,
                handoff=payload.get("handoff"),
                seed=tuple(message_from_wire(m) for m in payload.get("seed", ())),
                metadata=metadata,
            )
        if kind == "tool_call":
            if marker is None or reply is None:
                raise DecodeError("tool_call without marker/reply")
            return ToolInvoke(
                thread_id=thread_id,
                tool_name=payload["tool_name"],
                arguments=dict(payload.get("arguments", {})),
                marker=marker,
                reply=reply,
                metadata=dict(payload.get("metadata", {})),
            )
        if kind == "continuation":
            if marker is None:
                raise DecodeError("continuation without marker")
            msg = payload.get("message")
            return Continuation(
                thread_id=thread_id,
                marker=marker,
                content=payload.get("content"),
                is_error=bool(payload.get("is_error", False)),
                message=message_from_wire(msg) if msg else None,
            )
        if kind == "fault":
            return Fault(
                thread_id=thread_id,
                report=ErrorReport.from_wire(payload.get("report", {})),
                marker=marker,
            )
        raise DecodeError(f"unknown delivery kind {kind!r}")


# -- payload builders (the encode side of the kind-to-content knowledge) --

def agent_call_payload(message: Message, *, handoff: dict[str, Any] | None = None,
                       seed: tuple[Message, ...] = (),
                       metadata: dict[str, Any] | None = None) -> dict[str, Any]:
    p: dict[str, Any] = {"type": "agent_call", "message": message_to_wire(message)}
    if handoff:
        p["handoff"] = handoff
    if seed:
        p["seed"] = [message_to_wire(m) for m in seed]
    if metadata:
        p["metadata"] = metadata
    return p


def tool_call_payload(tool_name: str, arguments: dict[str, Any],
                      metadata: dict[str, Any] | None = None) -> dict[str, Any]:
    p: dict[str, Any] = {"type": "tool_call", "tool_name": tool_name,
                          "arguments": arguments}
    if metadata:
        p["metadata"] = metadata
    return p


def continuation_payload(content: Any = None, *, is_error: bool = False,
                         message: Message | None = None) -> dict[str, Any]:
    p: dict[str, Any] = {"type": "continuation", "content": content,
                          "is_error": is_error}
    if message is not None:
        p["message"] = message_to_wire(message)
    return p


def fault_payload(report: ErrorReport) -> dict[str, Any]:
    return {"type": "fault", "report": report.to_wire()}
