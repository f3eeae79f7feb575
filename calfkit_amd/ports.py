"""The four driven ports and their vocabulary. Tests fake here.

Capability parity: reference ``ports.py``
(/root/reference/src/calfkit/ports.py:1-4) names ``MessageBus``,
``ThreadStore``, ``Provider``, ``Directory`` and the port vocabulary
``RawDelivery``, ``GrainKey``, ``ThreadRecord``, ``ThreadStatus``,
``AppendDisposition``, ``Advert``, ``DirectoryView``.  The shapes here are
original asyncio protocols; key-ordered lane semantics (one consumer task per
grain key) are part of the MessageBus contract itself
(/root/reference/CHANGELOG.md:31).
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Awaitable, Callable, Protocol, runtime_checkable

from .messages import Message
from .schemas import EngineRequest
from .streaming import ChunkSink

__all__ = [
    "GrainKey",
    "RawDelivery",
    "DeliveryHandler",
    "MessageBus",
    "ThreadStatus",
    "AppendDisposition",
    "ThreadRecord",
    "ThreadStore",
    "Provider",
    "Advert",
    "DirectoryView",
    "Directory",
]

#: The ordering key: all deliveries sharing a grain key are consumed serially
#: in one lane; concurrency exists only across keys. For agent traffic the
#: grain key is the thread id.
GrainKey = str


@dataclass(frozen=True)
class RawDelivery:
    """One consumed record, before decoding."""

    topic: str
    key: GrainKey
    value: bytes
    headers: tuple[tuple[str, bytes], ...] = ()
    offset: int = -1
    partition: int = 0

    def header(self, name: str) -> bytes | None:
        for k, v in self.headers:
            if k == name:
                return v
        return None


DeliveryHandler = Callable[[RawDelivery], Awaitable[None]]


@runtime_checkable
class MessageBus(Protocol):
    """The transport port.

    Contract: ``publish`` is durable once awaited; ``subscribe`` registers a
    handler for a topic and the bus guarantees per-key serialized dispatch
    (key-ordered lanes) with concurrency across keys.
    """

    async def publish(self, topic: str, key: GrainKey, value: bytes,
                      headers: tuple[tuple[str, bytes], ...] = ()) -> None: ...

    async def subscribe(self, topic: str, handler: DeliveryHandler, *,
                        group: str = "") -> None: ...

    async def ensure_topics(self, topics: list[str]) -> None: ...

    async def close(self) -> None: ...


class ThreadStatus(enum.Enum):
    OPEN = "open"
    PENDING = "pending"  # waiting on tool / peer continuations
    CLOSED = "closed"
    DEAD = "dead"  # poisoned; deliveries for it are discarded


class AppendDisposition(enum.Enum):
    """What the store did with an append."""

    APPENDED = "appended"
    DUPLICATE = "duplicate"  # idempotent replay — already present
    CONFLICT = "conflict"  # lost an optimistic-concurrency race


@dataclass
class ThreadRecord:
    """The durable state of one conversation thread at one node."""

    thread_id: str
    messages: tuple[Message, ...] = ()
    status: ThreadStatus = ThreadStatus.OPEN
    #: monotonically increasing per-append revision
    revision: int = 0
    #: open call state: call_id -> wire dict (pending tool/peer batches)
    pending: dict[str, Any] = field(default_factory=dict)
    #: the frame id of the in-flight parallel batch, if any
    frame_id: str | None = None
    metadata: dict[str, Any] = field(default_factory=dict)


@runtime_checkable
class ThreadStore(Protocol):
    """Durable thread persistence. Store-first: the row is the truth."""

    async def load(self, node: str, thread_id: str) -> ThreadRecord | None: ...

    async def append(self, node: str, thread_id: str, messages: list[Message], *,
                     expected_revision: int | None = None,
                     status: ThreadStatus | None = None,
                     pending: dict[str, Any] | None = None,
                     frame_id: str | None = None,
                     metadata: dict[str, Any] | None = None) -> AppendDisposition: ...

    async def set_status(self, node: str, thread_id: str, status: ThreadStatus) -> None: ...


@runtime_checkable
class Provider(Protocol):
    """The model port. ``complete`` runs one step; chunks go to the sink."""

    async def complete(self, request: EngineRequest, sink: ChunkSink) -> Message: ...


@dataclass(frozen=True)
class Advert:
    """One node's capability advertisement on the control plane."""

    node: str
    kind: str  # "agent" | "tool" | "toolbox"
    description: str = ""
    #: tool parameter schemas / agent handoff descriptions, wire-shaped
    capabilities: dict[str, Any] = field(default_factory=dict)
    inbox: str = ""
    worker: str = ""
    ts: float = 0.0


@dataclass
class DirectoryView:
    """A materialized, point-in-time view of the mesh."""

    adverts: dict[str, Advert] = field(default_factory=dict)

    def agents(self) -> list[Advert]:
        return [a for a in self.adverts.values() if a.kind == "agent"]

    def tools(self) -> list[Advert]:
        return [a for a in self.adverts.values() if a.kind == "tool"]

    def toolboxes(self) -> list[Advert]:
        return [a for a in self.adverts.values() if a.kind == "toolbox"]

    def get(self, node: str) -> Advert | None:
        return self.adverts.get(node)


@runtime_checkable
class Directory(Protocol):
    """The discovery port: advertise capabilities, observe the mesh."""

    async def advertise(self, advert: Advert) -> None: ...

    async def withdraw(self, node: str) -> None: ...

    async def view(self) -> DirectoryView: ...

    async def watch(self) -> AsyncIterator[DirectoryView]: ...
