"""Stream chunks — a real-time reflection of facts, never the facts.

Capability parity: reference ``streaming.py``
(/root/reference/src/calfkit/streaming.py:1-2) names ``StreamChunk`` and
``ChunkSink``; intermediate step-event streaming to the caller firehose is
/root/reference/CHANGELOG.md:124.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Protocol, runtime_checkable

__all__ = ["StreamChunk", "ChunkSink", "NullSink", "CollectingSink"]


@dataclass(frozen=True)
class StreamChunk:
    """One increment of an in-progress step.

    ``event`` ∈ {"text_delta", "tool_call", "step_started", "step_finished",
    "handoff", "fault"}.  ``seq`` orders chunks within a step.
    """

    event: str
    seq: int = 0
    text: str = ""
    data: dict[str, Any] = field(default_factory=dict)
    node: str = ""
    thread_id: str = ""

    def to_wire(self) -> dict[str, Any]:
        return {"event": self.event, "seq": self.seq, "text": self.text,
                "data": self.data, "node": self.node, "thread_id": self.thread_id}

    @classmethod
    def from_wire(cls, d: dict[str, Any]) -> "StreamChunk":
        return cls(event=d.get("event", "?"), seq=int(d.get("seq", 0)),
                   text=d.get("text", ""), data=dict(d.get("data", {})),
                   node=d.get("node", ""), thread_id=d.get("thread_id", ""))


@runtime_checkable
class ChunkSink(Protocol):
    """Where stream chunks go. Implementations must tolerate bursts."""

    def emit(self, chunk: StreamChunk) -> None: ...


class NullSink:
    def emit(self, chunk: StreamChunk) -> None:  # pragma: no cover - trivial
        pass


class CollectingSink:
    """Test/diagnostic sink that keeps everything."""

    def __init__(self) -> None:
        self.chunks: list[StreamChunk] = []

    def emit(self, chunk: StreamChunk) -> None:
        self.chunks.append(chunk)

    @property
    def text(self) -> str:
        return "".join(c.text for c in self.chunks if c.event == "text_delta")
