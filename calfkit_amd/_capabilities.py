"""Side-load capability seam: the closed effect vocabulary and the
messaging + handoff capabilities built on it.

Capability parity: reference ``_capabilities.py``
(/root/reference/src/calfkit/_capabilities.py:1-3): closed effect vocabulary
``Dispatch``/``Transfer``/``Reject``, ``SideloadedCapability``, the
``@sideloads`` registry; peer messaging per /root/reference/CHANGELOG.md:154,
handoff (incl. tool-based handoff with field descriptions) per
/root/reference/CHANGELOG.md:153,49,157.

Design: a capability contributes pseudo-tools to the step surface
(mode="capability") and *realizes* calls on them into effects.  The effect
vocabulary is closed — the drive maps effects, the publisher realizes them;
nothing else escapes an agent hop.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Union

from .messages import Message, TextPart, ToolCallPart, UserMessage
from .ports import Advert
from .schemas import ToolBinding, ToolSchema

__all__ = [
    "Dispatch",
    "Transfer",
    "Reject",
    "Effect",
    "SideloadedCapability",
    "sideloads",
    "SIDELOAD_REGISTRY",
    "HandoffRequest",
    "MessagingCapability",
    "HandoffCapability",
]


# ------------------------------------------------------ effect vocabulary

@dataclass(frozen=True)
class Dispatch:
    """Send a message to a peer agent's inbox."""

    target: str
    target_inbox: str
    message: Message
    call_id: str
    #: when True the peer's answer comes back as a continuation of call_id
    reply_expected: bool = True


@dataclass(frozen=True)
class Transfer:
    """Hand the conversation off to another agent."""

    target: str
    target_inbox: str
    reason: str
    call_id: str


@dataclass(frozen=True)
class Reject:
    """The capability refused the call; surfaced to the model as an error."""

    call_id: str
    tool_name: str
    reason: str


Effect = Union[Dispatch, Transfer, Reject]


@dataclass(frozen=True)
class HandoffRequest:
    """Typed argument model of the handoff tool — the field descriptions ARE
    the model-facing schema (reference /root/reference/CHANGELOG.md:157,
    "HandoffRequest field descriptions")."""

    to: str
    reason: str = ""

    FIELD_DESCRIPTIONS = {
        "to": "agent to hand off to",
        "reason": "why the conversation is being handed off",
    }

    @classmethod
    def from_arguments(cls, arguments: dict) -> "HandoffRequest":
        return cls(to=str(arguments.get("to", "")),
                   reason=str(arguments.get("reason", "")))


# ----------------------------------------------------------- capabilities

SIDELOAD_REGISTRY: dict[str, type["SideloadedCapability"]] = {}


def sideloads(name: str):
    """Register a capability implementation under a stable name."""

    def deco(cls: type["SideloadedCapability"]) -> type["SideloadedCapability"]:
        SIDELOAD_REGISTRY[name] = cls
        cls.capability_name = name
        return cls

    return deco


class SideloadedCapability:
    """A capability that rides the tool seam without being a mesh node."""

    capability_name = "?"

    def contribute(self, peers: dict[str, Advert]) -> list[ToolBinding]:
        """Pseudo-tools offered to the model this hop."""
        return []

    def realize(self, call: ToolCallPart, peers: dict[str, Advert],
                source_agent: str) -> Effect | None:
        """Map a call on one of this capability's tools into an effect.

        Returns None when the call is not this capability's.
        """
        return None


def _peer_menu(peers: dict[str, Advert]) -> str:
    lines = [f"- {name}: {a.description or '(no description)'}"
             for name, a in sorted(peers.items())]
    return "\n".join(lines)


@sideloads("messaging")
class MessagingCapability(SideloadedCapability):
    """Peer messaging: the model sends a message to a named peer agent.

    ``await_reply=True`` turns the dispatch into a pending peer call — the
    peer's answer continues this thread.
    """

    TOOL = "send_message"

    def contribute(self, peers: dict[str, Advert]) -> list[ToolBinding]:
        if not peers:
            return []
        schema = ToolSchema(
            name=self.TOOL,
            description=("Send a message to a peer agent and optionally await "
                          "its reply. Known peers:\n" + _peer_menu(peers)),
            parameters={
                "type": "object",
                "properties": {
                    "to": {"type": "string", "enum": sorted(peers),
                            "description": "peer agent name"},
                    "message": {"type": "string", "description": "what to tell the peer"},
                    "await_reply": {"type": "boolean", "default": True,
                                     "description": "wait for the peer's answer"},
                },
                "required": ["to", "message"],
            })
        return [ToolBinding(schema=schema, mode="capability")]

    def realize(self, call: ToolCallPart, peers: dict[str, Advert],
                source_agent: str) -> Effect | None:
        if call.tool_name != self.TOOL:
            return None
        target = str(call.arguments.get("to", ""))
        advert = peers.get(target)
        if advert is None:
            return Reject(call_id=call.call_id, tool_name=call.tool_name,
                          reason=f"unknown peer {target!r}; known: {sorted(peers)}")
        text = str(call.arguments.get("message", ""))
        return Dispatch(
            target=target,
            target_inbox=advert.inbox,
            message=UserMessage((TextPart(text),), author=source_agent),
            call_id=call.call_id,
            reply_expected=bool(call.arguments.get("await_reply", True)),
        )


@sideloads("handoff")
class HandoffCapability(SideloadedCapability):
    """Tool-based handoff: transfer the conversation to a peer agent."""

    TOOL = "handoff"

    def contribute(self, peers: dict[str, Advert]) -> list[ToolBinding]:
        if not peers:
            return []
        schema = ToolSchema(
            name=self.TOOL,
            description=("Hand this conversation off to another agent better "
                          "suited to continue it. Known agents:\n" + _peer_menu(peers)),
            parameters={
                "type": "object",
                "properties": {
                    "to": {"type": "string", "enum": sorted(peers),
                            "description": HandoffRequest.FIELD_DESCRIPTIONS["to"]},
                    "reason": {"type": "string",
                                "description": HandoffRequest.FIELD_DESCRIPTIONS["reason"]},
                },
                "required": ["to"],
            })
        return [ToolBinding(schema=schema, mode="capability")]

    def realize(self, call: ToolCallPart, peers: dict[str, Advert],
                source_agent: str) -> Effect | None:
        if call.tool_name != self.TOOL:
            return None
        req = HandoffRequest.from_arguments(call.arguments)
        advert = peers.get(req.to)
        if advert is None:
            return Reject(call_id=call.call_id, tool_name=call.tool_name,
                          reason=f"unknown agent {req.to!r}; known: {sorted(peers)}")
        return Transfer(target=req.to, target_inbox=advert.inbox,
                        reason=req.reason, call_id=call.call_id)
