"""Turn policy: end-shape + correction machinery.

Capability parity: reference ``_policy.py``
(/root/reference/src/calfkit/_policy.py:1-2) names ``TurnPolicy`` and
``scan_open_calls``; structured output with ``final_result``-prefixed unions
per /root/reference/CHANGELOG.md:487,170.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass
from typing import Any

from .messages import (
    AssistantMessage,
    Message,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    UserMessage,
)
from .schemas import OutputSpec

__all__ = ["scan_open_calls", "TurnPolicy", "EndShape"]


def scan_open_calls(messages: tuple[Message, ...]) -> dict[str, ToolCallPart]:
    """Every tool call in the history with no matching return yet."""
    open_calls: dict[str, ToolCallPart] = {}
    for m in messages:
        if isinstance(m, AssistantMessage):
            for p in m.parts:
                if isinstance(p, ToolCallPart):
                    open_calls[p.call_id] = p
        elif isinstance(m, ToolMessage):
            for p in m.parts:
                if isinstance(p, ToolReturnPart):
                    open_calls.pop(p.call_id, None)
    return open_calls


@dataclass(frozen=True)
class EndShape:
    """Verdict on whether a model output may end the turn."""

    ok: bool
    #: parsed structured output when an OutputSpec was in force
    output: Any = None
    #: correction message to send back to the model when not ok
    correction: Message | None = None


_JSON_BLOCK = re.compile(r"```(?:json)?\s*(\{.*?\})\s*```", re.DOTALL)


@dataclass(frozen=True)
class TurnPolicy:
    """End-shape rules for one agent."""

    max_steps: int = 16
    max_corrections: int = 2

    def check_end_shape(self, message: AssistantMessage,
                        output: OutputSpec | None) -> EndShape:
        """A turn may end when nothing is pending and the final shape holds."""
        if message.tool_calls:
            return EndShape(ok=False)  # calls pending — not an end at all
        if output is None:
            return EndShape(ok=True)
        text = message.text.strip()
        parsed, err = self._parse_output(text, output)
        if err is None:
            return EndShape(ok=True, output=parsed)
        correction = UserMessage(
            (TextPart(
                "Your final answer must be a single JSON object matching the "
                f"{output.name!r} schema ({json.dumps(output.json_schema)}), "
                f"optionally wrapped as {{\"final_result\": ...}}. Problem: {err}. "
                "Reply with only the JSON object."),),
            author="system")
        return EndShape(ok=False, correction=correction)

    @staticmethod
    def _parse_output(text: str, output: OutputSpec) -> tuple[Any, str | None]:
        candidate = text
        m = _JSON_BLOCK.search(text)
        if m:
            candidate = m.group(1)
        else:
            # take the outermost {...} if the model chattered around it
            start, end = candidate.find("{"), candidate.rfind("}")
            if start >= 0 and end > start:
                candidate = candidate[start:end + 1]
        try:
            value = json.loads(candidate)
        except json.JSONDecodeError as e:
            return None, f"not valid JSON ({e})"
        # final_result-prefixed union unwrap
        if isinstance(value, dict) and set(value.keys()) == {"final_result"}:
            value = value["final_result"]
        err = _check_schema(value, output.json_schema)
        return (value, None) if err is None else (None, err)


def _check_schema(value: Any, schema: dict[str, Any], path: str = "$") -> str | None:
    """Minimal structural JSON-schema check (type/properties/required/items)."""
    t = schema.get("type")
    if t == "object":
        if not isinstance(value, dict):
            return f"{path}: expected object"
        for req in schema.get("required", ()):
            if req not in value:
                return f"{path}.{req}: required field missing"
        props = schema.get("properties", {})
        for k, v in value.items():
            if k in props:
                err = _check_schema(v, props[k], f"{path}.{k}")
                if err:
                    return err
        return None
    if t == "array":
        if not isinstance(value, list):
            return f"{path}: expected array"
        item = schema.get("items")
        if item:
            for i, v in enumerate(value):
                err = _check_schema(v, item, f"{path}[{i}]")
                if err:
                    return err
        return None
    if t == "string":
        return None if isinstance(value, str) else f"{path}: expected string"
    if t == "integer":
        return None if isinstance(value, int) and not isinstance(value, bool) \
            else f"{path}: expected integer"
    if t == "number":
        return None if isinstance(value, (int, float)) and not isinstance(value, bool) \
            else f"{path}: expected number"
    if t == "boolean":
        return None if isinstance(value, bool) else f"{path}: expected boolean"
    if "anyOf" in schema:
        for sub in schema["anyOf"]:
            if _check_schema(value, sub, path) is None:
                return None
        return f"{path}: matched no variant"
    if "enum" in schema:
        return None if value in schema["enum"] else f"{path}: not in enum"
    return None
