"""MeshBus + MeshDirectory — the broker-backed MessageBus/Directory ports.

Key-ordered lanes live client-side (the broker guarantees per-key sticky
group assignment; the client serializes handler execution per (topic, key)
exactly like the in-memory bus — /root/reference/CHANGELOG.md:31 semantics).
"""

from __future__ import annotations

import asyncio
import logging
import uuid
from collections import defaultdict
from typing import Any, AsyncIterator

from ...ports import Advert, DeliveryHandler, DirectoryView, GrainKey, RawDelivery
from .protocol import (
    b64d,
    b64e,
    encode_frame,
    headers_from_wire,
    headers_to_wire,
    read_frame,
)

_log = logging.getLogger("calfkit.mesh.client")

__all__ = ["MeshBus", "MeshDirectory", "connect_mesh"]


class _Lane:
    def __init__(self) -> None:
        self.queue: asyncio.Queue = asyncio.Queue()
        self.task: asyncio.Task | None = None


class _MeshConnection:
    """One TCP connection shared by the bus and directory adapters."""

    def __init__(self, host: str, port: int):
        self.host, self.port = host, port
        self.reader: asyncio.StreamReader | None = None
        self.writer: asyncio.StreamWriter | None = None
        self._reader_task: asyncio.Task | None = None
        self._handlers: dict[str, list[DeliveryHandler]] = defaultdict(list)
        self._lanes: dict[tuple[str, GrainKey], _Lane] = {}
        self._view_futures: dict[str, asyncio.Future] = {}
        self._dir_queue: list[asyncio.Queue] = []
        self._adverts: dict[str, Advert] = {}
        self._watching = False
        self._closed = False

    async def open(self) -> None:
        self.reader, self.writer = await asyncio.open_connection(self.host,
                                                                 self.port)
        self._reader_task = asyncio.get_running_loop().create_task(self._pump())

    def send(self, obj: dict[str, Any]) -> None:
        assert self.writer is not None
        self.writer.write(encode_frame(obj))

    async def _pump(self) -> None:
        assert self.reader is not None
        while not self._closed:
            frame = await read_frame(self.reader)
            if frame is None:
                break
            op = frame.get("op")
            if op == "record":
                self._dispatch(frame)
            elif op == "view":
                fut = self._view_futures.pop(frame.get("req_id", ""), None)
                if fut is not None and not fut.done():
                    fut.set_result(frame.get("adverts", []))
            elif op == "dirupdate":
                self._adverts = {
                    a["node"]: _advert_from_wire(a)
                    for a in frame.get("adverts", [])}
                for q in self._dir_queue:
                    q.put_nowait(None)

    def _dispatch(self, frame: dict[str, Any]) -> None:
        topic = frame["topic"]
        raw = RawDelivery(
            topic=topic, key=frame.get("key", ""),
            value=b64d(frame["value_b64"]),
            headers=headers_from_wire(frame.get("headers", [])),
            offset=int(frame.get("offset", -1)))
        handlers = self._handlers.get(topic, [])
        for h in handlers:
            lane_key = (topic, raw.key)
            lane = self._lanes.get(lane_key)
            if lane is None:
                lane = _Lane()
                self._lanes[lane_key] = lane
                lane.task = asyncio.get_running_loop().create_task(
                    self._run_lane(lane))
            lane.queue.put_nowait((raw, h))

    async def _run_lane(self, lane: _Lane) -> None:
        while True:
            raw, handler = await lane.queue.get()
            try:
                await handler(raw)
            except Exception:
                _log.exception("mesh handler failed on %s", raw.topic)

    async def close(self) -> None:
        self._closed = True
        if self._reader_task is not None:
            self._reader_task.cancel()
        for lane in self._lanes.values():
            if lane.task is not None:
                lane.task.cancel()
        if self.writer is not None:
            try:
                self.writer.close()
            except Exception:
                pass


def _advert_from_wire(a: dict[str, Any]) -> Advert:
    return Advert(node=a.get("node", ""), kind=a.get("kind", ""),
                  description=a.get("description", ""),
                  capabilities=dict(a.get("capabilities", {})),
                  inbox=a.get("inbox", ""), worker=a.get("worker", ""),
                  ts=float(a.get("ts", 0.0)))


class MeshBus:
    """MessageBus over one mesh connection."""

    def __init__(self, conn: _MeshConnection):
        self._c = conn

    async def publish(self, topic: str, key: GrainKey, value: bytes,
                      headers: tuple[tuple[str, bytes], ...] = ()) -> None:
        self._c.send({"op": "publish", "topic": topic, "key": key,
                      "value_b64": b64e(value),
                      "headers": headers_to_wire(headers)})
        assert self._c.writer is not None
        await self._c.writer.drain()

    async def subscribe(self, topic: str, handler: DeliveryHandler, *,
                        group: str = "") -> None:
        self._c._handlers[topic].append(handler)
        self._c.send({"op": "subscribe", "topic": topic, "group": group})
        assert self._c.writer is not None
        await self._c.writer.drain()

    async def ensure_topics(self, topics: list[str]) -> None:
        self._c.send({"op": "ensure", "topics": list(topics)})
        await self._c.writer.drain()

    async def close(self) -> None:
        await self._c.close()


class MeshDirectory:
    """Directory over the same mesh connection (compacted advert table)."""

    def __init__(self, conn: _MeshConnection):
        self._c = conn

    async def advertise(self, advert: Advert) -> None:
        self._c.send({"op": "advertise", "advert": {
            "node": advert.node, "kind": advert.kind,
            "description": advert.description,
            "capabilities": advert.capabilities, "inbox": advert.inbox,
            "worker": advert.worker}})
        await self._c.writer.drain()

    async def withdraw(self, node: str) -> None:
        self._c.send({"op": "withdraw", "node": node})
        await self._c.writer.drain()

    async def view(self) -> DirectoryView:
        req_id = uuid.uuid4().hex
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._c._view_futures[req_id] = fut
        self._c.send({"op": "view", "req_id": req_id})
        await self._c.writer.drain()
        adverts = await asyncio.wait_for(fut, 30)
        return DirectoryView(adverts={a["node"]: _advert_from_wire(a)
                                      for a in adverts})

    async def watch(self) -> AsyncIterator[DirectoryView]:
        q: asyncio.Queue = asyncio.Queue()
        self._c._dir_queue.append(q)
        if not self._c._watching:
            self._c._watching = True
            self._c.send({"op": "watch"})
            await self._c.writer.drain()
        try:
            yield await self.view()
            while True:
                await q.get()
                yield DirectoryView(adverts=dict(self._c._adverts))
        finally:
            self._c._dir_queue.remove(q)


async def connect_mesh(bootstrap: str, *, namespace: str = "") -> tuple[MeshBus, MeshDirectory]:
    """Dial ``host:port`` and return the (bus, directory) adapter pair."""
    host, _, port = bootstrap.rpartition(":")
    conn = _MeshConnection(host or "127.0.0.1", int(port))
    await conn.open()
    return MeshBus(conn), MeshDirectory(conn)
