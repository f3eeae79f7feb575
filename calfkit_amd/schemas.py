"""Model-facing vocabulary: tool schemas, settings, the engine request.

Capability parity: reference ``schemas.py``
(/root/reference/src/calfkit/schemas.py:1-6) names ``ToolSchema``,
``ToolBinding``, ``ModelSettings``, ``EngineRequest`` and sits below the ports
because ``Provider.complete()`` is typed on ``EngineRequest``.  Schema
derivation from Python signatures lives in ``definitions/tools.py``; this
module only owns the shapes.
"""

from __future__ import annotations

from dataclasses import dataclass, field, replace
from typing import Any

from .messages import Message

__all__ = ["ToolSchema", "ToolBinding", "ModelSettings", "EngineRequest", "OutputSpec"]


@dataclass(frozen=True)
class ToolSchema:
    """A JSON-schema description of one callable tool, as the model sees it."""

    name: str
    description: str = ""
    parameters: dict[str, Any] = field(default_factory=lambda: {"type": "object", "properties": {}})
    strict: bool = False

    def to_wire(self) -> dict[str, Any]:
        return {"name": self.name, "description": self.description,
                "parameters": self.parameters, "strict": self.strict}

    @classmethod
    def from_wire(cls, d: dict[str, Any]) -> "ToolSchema":
        return cls(name=d["name"], description=d.get("description", ""),
                   parameters=dict(d.get("parameters", {})), strict=bool(d.get("strict", False)))


@dataclass(frozen=True)
class ToolBinding:
    """A schema bound to *where its call must be published*.

    ``target_inbox`` is empty for capability-provided pseudo-tools that are
    realized in-process (messaging, handoff).
    """

    schema: ToolSchema
    target_inbox: str = ""
    #: "mesh" = publish to a node inbox; "capability" = side-loaded effect
    mode: str = "mesh"


@dataclass(frozen=True)
class ModelSettings:
    max_tokens: int = 1024
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0  # 0 = disabled
    seed: int | None = None
    stop: tuple[str, ...] = ()
    #: parallel tool-call fan-out permitted
    parallel_tool_calls: bool = True

    def merged(self, other: "ModelSettings | None") -> "ModelSettings":
        """Overlay non-default fields of ``other`` on this instance."""
        if other is None:
            return self
        base = ModelSettings()
        changes = {
            f: getattr(other, f)
            for f in ("max_tokens", "temperature", "top_p", "top_k", "seed", "stop",
                       "parallel_tool_calls")
            if getattr(other, f) != getattr(base, f)
        }
        return replace(self, **changes) if changes else self


@dataclass(frozen=True)
class OutputSpec:
    """Structured-output request: the model must end with this JSON shape.

    ``final_result`` union prefixing per /root/reference/CHANGELOG.md:487,170.
    """

    name: str
    json_schema: dict[str, Any]
    description: str = ""


@dataclass(frozen=True)
class EngineRequest:
    """Everything a provider needs for one completion step — wire-free."""

    messages: tuple[Message, ...]
    tools: tuple[ToolSchema, ...] = ()
    settings: ModelSettings = field(default_factory=ModelSettings)
    output: OutputSpec | None = None
    #: routing hint for multi-model providers; engine ignores unknown names
    model_name: str = ""
    #: tool choice presented to the model: "" = auto, "required" = must
    #: call some tool, or one advertised tool's name (ToolMode, reference
    #: selections v10.1)
    tool_choice: str = ""
