"""Framework-internal identity family — born at the parse boundary.

Capability parity: reference ``definitions/refs.py``
(/root/reference/src/calfkit/definitions/refs.py:1-3) names ``AgentRef``,
``ToolRef``, ``ToolboxRef``, ``EventRef``, ``ThreadRef``.
"""

from __future__ import annotations

import re
from dataclasses import dataclass

__all__ = ["AgentRef", "ToolRef", "ToolboxRef", "EventRef", "ThreadRef", "validate_node_name"]

_NAME_RE = re.compile(r"^[a-zA-Z][a-zA-Z0-9_\-]{0,127}$")


def validate_node_name(name: str) -> str:
    if not _NAME_RE.match(name):
        raise ValueError(
            f"invalid node name {name!r}: must match [a-zA-Z][a-zA-Z0-9_-]*, <=128 chars")
    return name


@dataclass(frozen=True)
class AgentRef:
    name: str

    def __post_init__(self) -> None:
        validate_node_name(self.name)


@dataclass(frozen=True)
class ToolRef:
    name: str

    def __post_init__(self) -> None:
        validate_node_name(self.name)


@dataclass(frozen=True)
class ToolboxRef:
    name: str

    def __post_init__(self) -> None:
        validate_node_name(self.name)


@dataclass(frozen=True)
class EventRef:
    """A named event stream on the mesh event plane."""

    name: str


@dataclass(frozen=True)
class ThreadRef:
    """Identity of a conversation thread (the grain key of agent traffic)."""

    thread_id: str
