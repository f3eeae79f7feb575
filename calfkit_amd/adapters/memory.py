"""Zero-broker adapters — same contracts, no infrastructure.

Capability parity: reference ``adapters/memory.py``
(/root/reference/src/calfkit/adapters/memory.py:1-3): ``InMemoryBus``,
``InMemoryDirectory``, ``InProcessThreadStore``; key-ordered lane semantics
per /root/reference/CHANGELOG.md:31.

The whole mesh (N workers + client) runs in one process on these — the
default test lane (/root/reference/CONTRIBUTING.md:86-88).
"""

from __future__ import annotations

import asyncio
import copy
import time
from collections import defaultdict
from typing import Any, AsyncIterator

from ..messages import Message
from ..ports import (
    Advert,
    AppendDisposition,
    DeliveryHandler,
    DirectoryView,
    GrainKey,
    RawDelivery,
    ThreadRecord,
    ThreadStatus,
)

__all__ = ["InMemoryBus", "InMemoryDirectory", "InProcessThreadStore"]


class _Lane:
    """One key's serialized consumption lane."""

    def __init__(self) -> None:
        self.queue: asyncio.Queue = asyncio.Queue()
        self.task: asyncio.Task | None = None


class InMemoryBus:
    """Topic/key bus with per-key ordered dispatch, all in one loop.

    Handlers for a topic each get every record (fan-out like consumer groups
    with distinct group ids; same ``group`` => one of them gets it,
    round-robin, matching Kafka consumer-group semantics).
    """

    def __init__(self) -> None:
        self._handlers: dict[str, list[tuple[str, DeliveryHandler]]] = defaultdict(list)
        self._lanes: dict[tuple[str, GrainKey], _Lane] = {}
        self._offsets: dict[str, int] = defaultdict(int)
        self._rr: dict[tuple[str, str], int] = defaultdict(int)
        self._closed = False
        self._inflight = 0
        self._idle = asyncio.Event()
        self._idle.set()
        self._log: list[RawDelivery] = []

    # ------------------------------------------------------------------ port
    async def publish(self, topic: str, key: GrainKey, value: bytes,
                      headers: tuple[tuple[str, bytes], ...] = ()) -> None:
        if self._closed:
            raise RuntimeError("bus closed")
        off = self._offsets[topic]
        self._offsets[topic] = off + 1
        raw = RawDelivery(topic=topic, key=key, value=value, headers=headers,
                          offset=off)
        self._log.append(raw)
        targets: list[DeliveryHandler] = []
        groups: dict[str, list[DeliveryHandler]] = defaultdict(list)
        for group, handler in self._handlers.get(topic, ()):  # fan-out by group
            if group:
                groups[group].append(handler)
            else:
                targets.append(handler)
        for group, members in groups.items():
            i = self._rr[(topic, group)] % len(members)
            self._rr[(topic, group)] += 1
            targets.append(members[i])
        for handler in targets:
            self._enqueue(topic, key, raw, handler)

    async def subscribe(self, topic: str, handler: DeliveryHandler, *,
                        group: str = "") -> None:
        self._handlers[topic].append((group, handler))

    async def ensure_topics(self, topics: list[str]) -> None:
        for t in topics:
            self._offsets.setdefault(t, 0)

    async def close(self) -> None:
        self._closed = True
        for lane in self._lanes.values():
            if lane.task is not None:
                lane.task.cancel()
        self._lanes.clear()

    # ------------------------------------------------------------- internals
    def _enqueue(self, topic: str, key: GrainKey, raw: RawDelivery,
                 handler: DeliveryHandler) -> None:
        lane = self._lanes.get((topic, key))
        if lane is None:
            lane = _Lane()
            self._lanes[(topic, key)] = lane
            lane.task = asyncio.get_running_loop().create_task(self._run_lane(lane))
        self._inflight += 1
        self._idle.clear()
        lane.queue.put_nowait((raw, handler))

    async def _run_lane(self, lane: _Lane) -> None:
        while True:
            raw, handler = await lane.queue.get()
            try:
                await handler(raw)
            except Exception:
                import logging
                logging.getLogger("calfkit.membus").exception(
                    "handler failed on %s", raw.topic)
            finally:
                self._inflight -= 1
                if self._inflight == 0:
                    self._idle.set()

    # ------------------------------------------------------------- test aids
    async def quiesce(self, timeout: float = 10.0) -> None:
        """Wait until every published record has been handled."""
        deadline = time.monotonic() + timeout
        while True:
            await asyncio.wait_for(self._idle.wait(),
                                   max(0.01, deadline - time.monotonic()))
            # handlers may publish more; settle
            await asyncio.sleep(0)
            if self._inflight == 0:
                return

    def records(self, topic: str) -> list[RawDelivery]:
        return [r for r in self._log if r.topic == topic]


class InMemoryDirectory:
    def __init__(self) -> None:
        self._adverts: dict[str, Advert] = {}
        self._watchers: list[asyncio.Queue] = []

    async def advertise(self, advert: Advert) -> None:
        advert = Advert(node=advert.node, kind=advert.kind,
                        description=advert.description,
                        capabilities=advert.capabilities, inbox=advert.inbox,
                        worker=advert.worker, ts=time.time())
        self._adverts[advert.node] = advert
        await self._notify()

    async def withdraw(self, node: str) -> None:
        self._adverts.pop(node, None)
        await self._notify()

    async def view(self) -> DirectoryView:
        return DirectoryView(adverts=dict(self._adverts))

    async def watch(self) -> AsyncIterator[DirectoryView]:
        q: asyncio.Queue = asyncio.Queue()
        self._watchers.append(q)
        yield await self.view()
        try:
            while True:
                await q.get()
                yield await self.view()
        finally:
            self._watchers.remove(q)

    async def _notify(self) -> None:
        for q in self._watchers:
            q.put_nowait(None)


class InProcessThreadStore:
    """Dict-backed ThreadStore with optimistic concurrency."""

    def __init__(self) -> None:
        self._rows: dict[tuple[str, str], ThreadRecord] = {}
        self._lock = asyncio.Lock()

    async def load(self, node: str, thread_id: str) -> ThreadRecord | None:
        rec = self._rows.get((node, thread_id))
        if rec is None:
            return None
        return ThreadRecord(thread_id=rec.thread_id, messages=rec.messages,
                            status=rec.status, revision=rec.revision,
                            pending=copy.deepcopy(rec.pending),
                            frame_id=rec.frame_id,
                            metadata=copy.deepcopy(rec.metadata))

    async def append(self, node: str, thread_id: str, messages: list[Message], *,
                     expected_revision: int | None = None,
                     status: ThreadStatus | None = None,
                     pending: dict[str, Any] | None = None,
                     frame_id: str | None = None,
                     metadata: dict[str, Any] | None = None) -> AppendDisposition:
        async with self._lock:
            rec = self._rows.get((node, thread_id))
            if rec is None:
                rec = ThreadRecord(thread_id=thread_id)
                self._rows[(node, thread_id)] = rec
            if expected_revision is not None and rec.revision != expected_revision:
                return AppendDisposition.CONFLICT
            rec.messages = rec.messages + tuple(messages)
            rec.revision += 1
            if status is not None:
                rec.status = status
            if pending is not None:
                rec.pending = copy.deepcopy(pending)
            rec.frame_id = frame_id
            if metadata is not None:
                rec.metadata = copy.deepcopy(metadata)
            return AppendDisposition.APPENDED

    async def set_status(self, node: str, thread_id: str, status: ThreadStatus) -> None:
        rec = self._rows.get((node, thread_id))
        if rec is not None:
            rec.status = status
