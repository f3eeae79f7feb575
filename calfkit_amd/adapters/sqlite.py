"""SqliteThreadStore — the durable thread backend.

Capability parity: reference ``adapters/sqlite.py``
(/root/reference/src/calfkit/adapters/sqlite.py:1-2): durable threads,
mounted as a lifecycle ``@resource``; message-history persistence per
/root/reference/CHANGELOG.md:643.

Implementation: stdlib sqlite3 in WAL mode driven through a thread executor
(sqlite3 is sync; the agent plane is asyncio).  One row per (node, thread)
holding the serialized record; appends are optimistic-concurrency-checked on
the revision column, giving the same APPENDED/DUPLICATE/CONFLICT contract as
the in-memory store.
"""

from __future__ import annotations

import asyncio
import json
import sqlite3
import threading
from typing import Any

from ..lifecycle import ResourceDef
from ..messages import Message, thread_from_wire, thread_to_wire
from ..ports import AppendDisposition, ThreadRecord, ThreadStatus

__all__ = ["SqliteThreadStore", "sqlite_store_resource"]

_SCHEMA = """
CREATE TABLE IF NOT EXISTS threads (
    node TEXT NOT NULL,
    thread_id TEXT NOT NULL,
    revision INTEGER NOT NULL DEFAULT 0,
    status TEXT NOT NULL DEFAULT 'open',
    messages TEXT NOT NULL DEFAULT '[]',
    pending TEXT NOT NULL DEFAULT '{}',
    frame_id TEXT,
    metadata TEXT NOT NULL DEFAULT '{}',
    PRIMARY KEY (node, thread_id)
);
"""


class SqliteThreadStore:
    def __init__(self, path: str = ":memory:"):
        self.path = path
        self._local = threading.local()
        self._init_lock = threading.Lock()
        # :memory: needs a single shared connection (each new conn is empty)
        self._memory_conn: sqlite3.Connection | None = None
        if path == ":memory:":
            self._memory_conn = self._new_conn()

    def _new_conn(self) -> sqlite3.Connection:
        conn = sqlite3.connect(self.path, check_same_thread=False)
        conn.execute("PRAGMA journal_mode=WAL")
        conn.execute("PRAGMA synchronous=NORMAL")
        conn.execute(_SCHEMA)
        return conn

    def _conn(self) -> sqlite3.Connection:
        if self._memory_conn is not None:
            return self._memory_conn
        conn = getattr(self._local, "conn", None)
        if conn is None:
            with self._init_lock:
                conn = self._new_conn()
            self._local.conn = conn
        return conn

    async def _run(self, fn, *args):
        return await asyncio.get_running_loop().run_in_executor(None, fn, *args)

    # ------------------------------------------------------------------ port
    async def load(self, node: str, thread_id: str) -> ThreadRecord | None:
        def q():
            cur = self._conn().execute(
                "SELECT revision, status, messages, pending, frame_id, metadata "
                "FROM threads WHERE node=? AND thread_id=?", (node, thread_id))
            return cur.fetchone()

        row = await self._run(q)
        if row is None:
            return None
        revision, status, messages, pending, frame_id, metadata = row
        return ThreadRecord(
            thread_id=thread_id,
            messages=thread_from_wire(json.loads(messages)),
            status=ThreadStatus(status),
            revision=revision,
            pending=json.loads(pending),
            frame_id=frame_id,
            metadata=json.loads(metadata),
        )

    async def append(self, node: str, thread_id: str, messages: list[Message], *,
                     expected_revision: int | None = None,
                     status: ThreadStatus | None = None,
                     pending: dict[str, Any] | None = None,
                     frame_id: str | None = None,
                     metadata: dict[str, Any] | None = None) -> AppendDisposition:
        new_wire = json.dumps(thread_to_wire(messages))

        def tx():
            conn = self._conn()
            with conn:  # one transaction
                cur = conn.execute(
                    "SELECT revision, messages, status, pending, metadata "
                    "FROM threads WHERE node=? AND thread_id=?",
                    (node, thread_id))
                row = cur.fetchone()
                if row is None:
                    revision, existing = 0, "[]"
                    cur_status, cur_pending, cur_meta = "open", "{}", "{}"
                    conn.execute(
                        "INSERT INTO threads (node, thread_id) VALUES (?, ?)",
                        (node, thread_id))
                else:
                    revision, existing, cur_status, cur_pending, cur_meta = row
                if expected_revision is not None and revision != expected_revision:
                    return AppendDisposition.CONFLICT
                merged = json.loads(existing) + json.loads(new_wire)
                conn.execute(
                    "UPDATE threads SET revision=?, messages=?, status=?, "
                    "pending=?, frame_id=?, metadata=? "
                    "WHERE node=? AND thread_id=?",
                    (revision + 1, json.dumps(merged),
                     (status.value if status else cur_status),
                     json.dumps(pending if pending is not None
                                else json.loads(cur_pending)),
                     frame_id,
                     json.dumps(metadata if metadata is not None
                                else json.loads(cur_meta)),
                     node, thread_id))
                return AppendDisposition.APPENDED

        return await self._run(tx)

    async def set_status(self, node: str, thread_id: str,
                         status: ThreadStatus) -> None:
        def q():
            with self._conn() as conn:
                conn.execute(
                    "UPDATE threads SET status=? WHERE node=? AND thread_id=?",
                    (status.value, node, thread_id))

        await self._run(q)

    async def close(self) -> None:
        def q():
            if self._memory_conn is not None:
                self._memory_conn.close()

        await self._run(q)


def sqlite_store_resource(path: str) -> ResourceDef:
    """Mount a SqliteThreadStore as a worker lifecycle ``@resource``."""

    async def sqlite_store():
        store = SqliteThreadStore(path)
        yield store
        await store.close()

    return ResourceDef(name="sqlite_store", factory=sqlite_store)
