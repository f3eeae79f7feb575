"""Public seam-facing context objects users annotate.

Capability parity: reference ``context.py``
(/root/reference/src/calfkit/context.py:1-3) names ``TurnContext`` (agent
plane), ``ToolContext`` (tool plane), ``EventStamp``; ToolContext injection
for ``@agent_tool`` is /root/reference/CHANGELOG.md:563.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable

__all__ = ["EventStamp", "TurnContext", "ToolContext"]


@dataclass(frozen=True)
class EventStamp:
    """Provenance of one event: which node emitted it, when, in which hop."""

    node: str
    ts: float = field(default_factory=time.time)
    hop: int = 0
    seq: int = 0


EmitFn = Callable[[str, Any], Awaitable[None]]


async def _no_emit(event: str, payload: Any) -> None:  # pragma: no cover
    return None


@dataclass
class TurnContext:
    """Handed to agent hooks and instruction functions for one turn."""

    agent_name: str
    thread_id: str
    turn: int = 0
    deps: Any = None
    metadata: dict[str, Any] = field(default_factory=dict)
    #: fire-and-forget event emission onto the mesh event plane
    emit: EmitFn = _no_emit


@dataclass
class ToolContext:
    """Injected into a tool function that annotates a ``ToolContext`` param."""

    tool_name: str
    call_id: str
    caller: str = ""
    thread_id: str = ""
    metadata: dict[str, Any] = field(default_factory=dict)
    emit: EmitFn = _no_emit
