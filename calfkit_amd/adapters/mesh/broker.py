"""MeshBroker — the bundled zero-setup broker.

Semantics the core depends on:
- topics with per-key ordering (order preserved per topic; consumers get
  key-ordered lanes client-side);
- consumer groups: records fan out to every distinct group; within a group,
  one member gets each record, keyed round-robin BY KEY so a thread's
  deliveries stay on one member (the Kafka partition-assignment analog);
- a compacted directory table with live watch pushes (the control plane);
- explicit topic provisioning (no reliance on auto-create).

Run standalone via ``ck dev mesh start`` or in-process for tests.
"""

from __future__ import annotations

import asyncio
import logging
import time
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Any

from .protocol import encode_frame, read_frame

_log = logging.getLogger("calfkit.mesh.broker")

__all__ = ["MeshBroker"]


@dataclass
class _Subscription:
    topic: str
    group: str
    conn: "_Conn"


@dataclass
class _Conn:
    writer: asyncio.StreamWriter
    id: int
    alive: bool = True
    subs: list[_Subscription] = field(default_factory=list)
    #: nodes this connection advertised (liveness: they die with it)
    adverts: set[str] = field(default_factory=set)

    def send(self, obj: dict[str, Any]) -> None:
        if not self.alive:
            return
        try:
            self.writer.write(encode_frame(obj))
        except (ConnectionResetError, RuntimeError):
            self.alive = False


class MeshBroker:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.host = host
        self.port = port
        self._server: asyncio.AbstractServer | None = None
        self._topics: dict[str, int] = {}  # topic -> next offset
        self._subs: dict[str, list[_Subscription]] = defaultdict(list)
        #: per (topic, group): key -> member index assignment
        self._assign: dict[tuple[str, str], dict[str, int]] = defaultdict(dict)
        self._rr: dict[tuple[str, str], int] = defaultdict(int)
        self._adverts: dict[str, dict[str, Any]] = {}
        self._watchers: list[_Conn] = []
        self._next_conn = 0

    # ------------------------------------------------------------- lifecycle
    async def start(self) -> "MeshBroker":
        self._server = await asyncio.start_server(self._serve, self.host,
                                                  self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        _log.info("mesh broker listening on %s:%d", self.host, self.port)
        return self

    @property
    def address(self) -> str:
        return f"{self.host}:{self.port}"

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()

    async def serve_forever(self) -> None:
        assert self._server is not None
        async with self._server:
            await self._server.serve_forever()

    # -------------------------------------------------------------- serving
    async def _serve(self, reader: asyncio.StreamReader,
                     writer: asyncio.StreamWriter) -> None:
        self._next_conn += 1
        conn = _Conn(writer=writer, id=self._next_conn)
        try:
            while True:
                frame = await read_frame(reader)
                if frame is None:
                    break
                await self._handle(conn, frame)
        except Exception:
            _log.exception("mesh connection %d failed", conn.id)
        finally:
            conn.alive = False
            self._drop_conn(conn)
            try:
                writer.close()
            except Exception:
                pass

    def _drop_conn(self, conn: _Conn) -> None:
        for sub in conn.subs:
            members = self._subs.get(sub.topic, [])
            if sub in members:
                members.remove(sub)
            # rebalance: drop key assignments pointing at the dead member
            self._assign.pop((sub.topic, sub.group), None)
        if conn in self._watchers:
            self._watchers.remove(conn)
        # liveness: a dead worker's adverts leave the directory with it
        dropped = False
        for node in conn.adverts:
            if self._adverts.pop(node, None) is not None:
                dropped = True
        if dropped:
            self._push_directory()

    async def _handle(self, conn: _Conn, f: dict[str, Any]) -> None:
        op = f.get("op")
        if op == "publish":
            self._publish(f)
        elif op == "subscribe":
            sub = _Subscription(topic=f["topic"], group=f.get("group", ""),
                                conn=conn)
            conn.subs.append(sub)
            self._subs[f["topic"]].append(sub)
            self._topics.setdefault(f["topic"], 0)
        elif op == "ensure":
            for t in f.get("topics", ()):  # explicit provisioning
                self._topics.setdefault(t, 0)
        elif op == "advertise":
            advert = dict(f["advert"])
            advert["ts"] = time.time()
            self._adverts[advert["node"]] = advert
            conn.adverts.add(advert["node"])
            self._push_directory()
        elif op == "withdraw":
            self._adverts.pop(f.get("node", ""), None)
            self._push_directory()
        elif op == "view":
            conn.send({"op": "view", "req_id": f.get("req_id"),
                       "adverts": list(self._adverts.values())})
        elif op == "watch":
            self._watchers.append(conn)
            conn.send({"op": "dirupdate", "adverts": list(self._adverts.values())})
        elif op == "ping":
            conn.send({"op": "pong", "req_id": f.get("req_id")})
        else:
            _log.warning("unknown op %r from conn %d", op, conn.id)

    def _publish(self, f: dict[str, Any]) -> None:
        topic, key = f["topic"], f.get("key", "")
        offset = self._topics.get(topic, 0)
        self._topics[topic] = offset + 1
        record = {"op": "record", "topic": topic, "key": key,
                  "value_b64": f["value_b64"], "headers": f.get("headers", []),
                  "offset": offset}
        # fan out once per consumer group; sticky key->member inside a group
        by_group: dict[str, list[_Subscription]] = defaultdict(list)
        for sub in self._subs.get(topic, []):
            if sub.conn.alive:
                by_group[sub.group].append(sub)
        for group, members in by_group.items():
            if not members:
                continue
            if group == "":
                for sub in members:  # ungrouped: broadcast
                    sub.conn.send(record)
                continue
            assign = self._assign[(topic, group)]
            idx = assign.get(key)
            if idx is None or idx >= len(members):
                idx = self._rr[(topic, group)] % len(members)
                self._rr[(topic, group)] += 1
                assign[key] = idx
            members[idx].conn.send(record)

    def _push_directory(self) -> None:
        update = {"op": "dirupdate", "adverts": list(self._adverts.values())}
        for w in list(self._watchers):
            if w.alive:
                w.send(update)
            else:
                self._watchers.remove(w)
