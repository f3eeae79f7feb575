"""Owned message vocabulary: parts, messages, converters.

Capability parity: reference ``messages.py``
(/root/reference/src/calfkit/messages.py:1-2) commits to parts, message types,
a ``Message`` union, ``ProviderFacts`` and value-to-parts converters.  The
concrete shapes here are original: frozen dataclasses with a symmetric
``to_wire``/``message_from_wire`` JSON round-trip (the wire stays JSON for
Kafka byte-compatibility).
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Union

__all__ = [
    "TextPart",
    "ToolCallPart",
    "ToolReturnPart",
    "Part",
    "SystemMessage",
    "UserMessage",
    "AssistantMessage",
    "ToolMessage",
    "Message",
    "ProviderFacts",
    "Usage",
    "value_to_parts",
    "parts_to_text",
    "message_to_wire",
    "message_from_wire",
    "thread_to_wire",
    "thread_from_wire",
]


# ----------------------------------------------------------------- parts

@dataclass(frozen=True)
class TextPart:
    text: str
    kind: str = "text"


@dataclass(frozen=True)
class ToolCallPart:
    """The model asked for a tool. ``call_id`` is the echo-marker identity."""

    call_id: str
    tool_name: str
    arguments: dict[str, Any] = field(default_factory=dict)
    kind: str = "tool_call"


@dataclass(frozen=True)
class ToolReturnPart:
    call_id: str
    tool_name: str
    content: Any = None
    is_error: bool = False
    kind: str = "tool_return"


Part = Union[TextPart, ToolCallPart, ToolReturnPart]


# ----------------------------------------------------------------- facts

@dataclass(frozen=True)
class Usage:
    prompt_tokens: int = 0
    completion_tokens: int = 0

    @property
    def total_tokens(self) -> int:
        return self.prompt_tokens + self.completion_tokens


@dataclass(frozen=True)
class ProviderFacts:
    """What actually happened at the provider — never a real-time reflection.

    The MI355X engine also records per-phase GPU timings here so the
    StepLedger can flush them with the step record.
    """

    model_name: str = ""
    usage: Usage = field(default_factory=Usage)
    finish_reason: str = "stop"  # "stop" | "tool_calls" | "length" | "error"
    prefill_ms: float = 0.0
    decode_ms: float = 0.0


# -------------------------------------------------------------- messages

@dataclass(frozen=True)
class SystemMessage:
    parts: tuple[Part, ...]
    role: str = "system"


@dataclass(frozen=True)
class UserMessage:
    parts: tuple[Part, ...]
    #: who said it — matters for multi-agent channel projection
    author: str = "user"
    role: str = "user"


@dataclass(frozen=True)
class AssistantMessage:
    parts: tuple[Part, ...]
    author: str = "assistant"
    facts: ProviderFacts | None = None
    role: str = "assistant"

    @property
    def tool_calls(self) -> tuple[ToolCallPart, ...]:
        return tuple(p for p in self.parts if isinstance(p, ToolCallPart))

    @property
    def text(self) -> str:
        return parts_to_text(self.parts)


@dataclass(frozen=True)
class ToolMessage:
    parts: tuple[Part, ...]
    role: str = "tool"


Message = Union[SystemMessage, UserMessage, AssistantMessage, ToolMessage]

_ROLE_TO_CLS: dict[str, type] = {
    "system": SystemMessage,
    "user": UserMessage,
    "assistant": AssistantMessage,
    "tool": ToolMessage,
}


# ------------------------------------------------------------ converters

def value_to_parts(value: Any) -> tuple[Part, ...]:
    """Convert an arbitrary user value into message parts."""
    if value is None:
        return ()
    if isinstance(value, (TextPart, ToolCallPart, ToolReturnPart)):
        return (value,)
    if isinstance(value, str):
        return (TextPart(value),)
    if isinstance(value, (list, tuple)):
        out: list[Part] = []
        for v in value:
            out.extend(value_to_parts(v))
        return tuple(out)
    # structured values ride as JSON text
    return (TextPart(json.dumps(value, default=str)),)


def parts_to_text(parts: tuple[Part, ...]) -> str:
    return "".join(p.text for p in parts if isinstance(p, TextPart))


# ------------------------------------------------------------- wire form

def _part_to_wire(p: Part) -> dict[str, Any]:
    if isinstance(p, TextPart):
        return {"kind": "text", "text": p.text}
    if isinstance(p, ToolCallPart):
        return {"kind": "tool_call", "call_id": p.call_id, "tool_name": p.tool_name,
                "arguments": p.arguments}
    if isinstance(p, ToolReturnPart):
        return {"kind": "tool_return", "call_id": p.call_id, "tool_name": p.tool_name,
                "content": p.content, "is_error": p.is_error}
    raise TypeError(f"unknown part {p!r}")


def _part_from_wire(d: dict[str, Any]) -> Part:
    k = d.get("kind")
    if k == "text":
        return TextPart(d.get("text", ""))
    if k == "tool_call":
        return ToolCallPart(d.get("call_id", ""), d.get("tool_name", ""),
                            dict(d.get("arguments", {})))
    if k == "tool_return":
        return ToolReturnPart(d.get("call_id", ""), d.get("tool_name", ""),
                              d.get("content"), bool(d.get("is_error", False)))
    raise ValueError(f"unknown part kind {k!r}")


def message_to_wire(m: Message) -> dict[str, Any]:
    d: dict[str, Any] = {"role": m.role, "parts": [_part_to_wire(p) for p in m.parts]}
    if isinstance(m, (UserMessage, AssistantMessage)):
        d["author"] = m.author
    if isinstance(m, AssistantMessage) and m.facts is not None:
        f = m.facts
        d["facts"] = {
            "model_name": f.model_name,
            "usage": {"prompt_tokens": f.usage.prompt_tokens,
                       "completion_tokens": f.usage.completion_tokens},
            "finish_reason": f.finish_reason,
            "prefill_ms": f.prefill_ms,
            "decode_ms": f.decode_ms,
        }
    return d


def message_from_wire(d: dict[str, Any]) -> Message:
    role = d.get("role")
    cls = _ROLE_TO_CLS.get(role or "")
    if cls is None:
        raise ValueError(f"unknown message role {role!r}")
    parts = tuple(_part_from_wire(p) for p in d.get("parts", ()))
    if cls is SystemMessage:
        return SystemMessage(parts)
    if cls is UserMessage:
        return UserMessage(parts, author=d.get("author", "user"))
    if cls is ToolMessage:
        return ToolMessage(parts)
    facts = None
    if "facts" in d:
        f = d["facts"]
        u = f.get("usage", {})
        facts = ProviderFacts(
            model_name=f.get("model_name", ""),
            usage=Usage(int(u.get("prompt_tokens", 0)), int(u.get("completion_tokens", 0))),
            finish_reason=f.get("finish_reason", "stop"),
            prefill_ms=float(f.get("prefill_ms", 0.0)),
            decode_ms=float(f.get("decode_ms", 0.0)),
        )
    return AssistantMessage(parts, author=d.get("author", "assistant"), facts=facts)


def thread_to_wire(messages: tuple[Message, ...] | list[Message]) -> list[dict[str, Any]]:
    return [message_to_wire(m) for m in messages]


def thread_from_wire(items: list[dict[str, Any]]) -> tuple[Message, ...]:
    return tuple(message_from_wire(d) for d in items)
