"""ThreadCarriage: durable rows + working copy; the admission table.

Capability parity: reference ``_carriage.py``
(/root/reference/src/calfkit/_carriage.py:1-3): "store-first, mirror on
appended"; admission verdicts ``Proceed/Absorbed/Dead/Discard``; the
durable-agent premise (any replica resumes any thread between hops).

Semantics here (original):
- ``admit`` is the poison/duplicate/stale defense, decided from the durable
  record BEFORE any work happens.
- the working copy mirrors the store only after an append reports APPENDED —
  never optimistically.
- pending parallel-tool batches are keyed by ``frame_id``
  (/root/reference/CHANGELOG.md:372); a continuation for an unknown call is
  Absorbed (stale duplicate), for a dead thread Discarded.
"""

from __future__ import annotations

import enum
from typing import Any

from .messages import Message
from .ports import AppendDisposition, ThreadRecord, ThreadStatus, ThreadStore

__all__ = ["Admission", "ThreadCarriage"]


class Admission(enum.Enum):
    PROCEED = "proceed"  # run the pipeline for this delivery
    ABSORBED = "absorbed"  # recorded/ignored without a step (duplicate, early)
    DEAD = "dead"  # thread poisoned — fault the delivery
    DISCARD = "discard"  # drop silently (stale echo, replay)


class ThreadCarriage:
    """Owns one (node, thread) row across a hop."""

    def __init__(self, store: ThreadStore, node: str, thread_id: str):
        self._store = store
        self.node = node
        self.thread_id = thread_id
        self.record: ThreadRecord = ThreadRecord(thread_id=thread_id)
        self._appended_this_hop: list[Message] = []

    # ------------------------------------------------------------ hydrate
    async def hydrate(self) -> ThreadRecord:
        """Load durable rows into the working copy. Store-first."""
        rec = await self._store.load(self.node, self.thread_id)
        if rec is None:
            rec = ThreadRecord(thread_id=self.thread_id)
        self.record = rec
        return rec

    # ------------------------------------------------------------- admit
    def admit(self, kind: str, call_id: str | None = None) -> Admission:
        """The admission table. ``kind`` ∈ {"call", "continuation", "fault"}.

        Decided purely from the hydrated record.
        """
        rec = self.record
        if rec.status == ThreadStatus.DEAD:
            return Admission.DEAD
        if kind == "continuation":
            if not rec.pending:
                return Admission.DISCARD  # stale echo of a finished batch
            if call_id is None or call_id not in rec.pending:
                return Admission.ABSORBED  # duplicate reply inside a live batch
            return Admission.PROCEED
        if kind == "fault":
            # a fault continuation follows the same identity rules
            if call_id is not None and rec.pending and call_id in rec.pending:
                return Admission.PROCEED
            return Admission.ABSORBED
        # fresh call
        if rec.status == ThreadStatus.PENDING:
            # a new inbound while a batch is in flight: record it, step later
            return Admission.ABSORBED
        return Admission.PROCEED

    # ----------------------------------------------------------- mutation
    @property
    def messages(self) -> tuple[Message, ...]:
        return self.record.messages

    def working_extend(self, messages: list[Message]) -> None:
        """Stage messages on the working copy for this hop's flush."""
        self._appended_this_hop.extend(messages)

    @property
    def staged(self) -> list[Message]:
        return list(self._appended_this_hop)

    def working_view(self) -> tuple[Message, ...]:
        return self.record.messages + tuple(self._appended_this_hop)

    async def flush(self, *, status: ThreadStatus | None = None,
                    pending: dict[str, Any] | None = None,
                    frame_id: str | None = None) -> AppendDisposition:
        """Persist staged messages + state. Mirrors only on APPENDED."""
        disposition = await self._store.append(
            self.node, self.thread_id, self._appended_this_hop,
            expected_revision=self.record.revision,
            status=status, pending=pending, frame_id=frame_id,
            metadata=dict(self.record.metadata))
        if disposition == AppendDisposition.APPENDED:
            self.record.messages = self.record.messages + tuple(self._appended_this_hop)
            self.record.revision += 1
            if status is not None:
                self.record.status = status
            if pending is not None:
                self.record.pending = dict(pending)
            self.record.frame_id = frame_id
            self._appended_this_hop = []
        return disposition

    async def mark_dead(self) -> None:
        await self._store.set_status(self.node, self.thread_id, ThreadStatus.DEAD)
        self.record.status = ThreadStatus.DEAD
