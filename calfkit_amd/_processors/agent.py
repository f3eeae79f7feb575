"""AgentProcessor — the agent driver loop:
decode → admit → hydrate → surface → hooks → drive → realize.

Capability parity: reference ``_processors/agent.py``
(/root/reference/src/calfkit/_processors/agent.py:1-2) spells the pipeline
verbatim; re-entrancy across hops (pending frame -> Continuation -> pipeline
again) per /root/reference/SURVEY §3.2; durable fan-out with per-frame
aggregation per /root/reference/CHANGELOG.md:192,196.
"""

from __future__ import annotations

import asyncio
from typing import Any

from .._capabilities import Dispatch, Transfer
from .._carriage import Admission, ThreadCarriage
from .._drive import StepResolution, drive_step
from .._engine import Engine
from .._policy import TurnPolicy
from .._surface import build_step_surface
from .._hosting import processes
from .._wire.deliveries import (
    Call,
    Continuation,
    Delivery,
    EchoMarker,
    Fault,
    ReplySlot,
)
from .._wire.machinery import NodeMachinery
from .._wire.publisher import FaultFunnel
from .._wire import topology
from ..context import TurnContext
from ..definitions.agent import Agent
from ..errors import DecodeError
from ..messages import (
    AssistantMessage,
    ToolMessage,
    ToolReturnPart,
    parts_to_text,
)
from .._projection import project_for_model, project_for_transfer
from ..ports import RawDelivery, ThreadStatus
from ..streaming import ChunkSink, StreamChunk

__all__ = ["AgentProcessor"]

MAX_HOPS = 64


class _BusSink(ChunkSink):
    """Streams chunks to the firehose as they happen (real-time reflection).

    Thread-safe: the GPU engine may emit from its scheduler thread.
    """

    def __init__(self, processor: "AgentProcessor", thread_id: str,
                 reply: ReplySlot | None, loop: asyncio.AbstractEventLoop):
        self._p = processor
        self._thread_id = thread_id
        self._reply = reply
        self._loop = loop
        self._tasks: set[asyncio.Task] = set()

    def emit(self, chunk: StreamChunk) -> None:
        chunk = StreamChunk(event=chunk.event, seq=chunk.seq, text=chunk.text,
                            data=chunk.data, node=self._p.node_name,
                            thread_id=self._thread_id)
        try:
            running = asyncio.get_running_loop()
        except RuntimeError:
            running = None
        try:
            if running is self._loop:
                self._track(self._loop.create_task(self._send(chunk)))
            else:
                self._loop.call_soon_threadsafe(
                    lambda: self._track(self._loop.create_task(self._send(chunk))))
        except RuntimeError:
            pass  # loop already closed (process teardown) — drop the chunk

    def _track(self, t: asyncio.Task) -> None:
        self._tasks.add(t)
        t.add_done_callback(self._tasks.discard)

    async def _send(self, chunk: StreamChunk) -> None:
        try:
            await self._p.publisher.stream_chunk(thread_id=self._thread_id,
                                                 chunk=chunk, reply=None)
        except Exception:
            pass  # the reflection never takes the hop down

    async def drain(self) -> None:
        if self._tasks:
            await asyncio.gather(*list(self._tasks), return_exceptions=True)


@processes("agent")
class AgentProcessor:
    def __init__(self, agent: Agent, machinery: NodeMachinery):
        self.agent = agent
        self.machinery = machinery
        self.node_name = agent.name
        self.publisher = machinery.transport.bind(agent.name)
        self.funnel = FaultFunnel(self.publisher, "agent")
        provider = agent.model.resolve_client() if agent.model else None
        if provider is None:
            raise TypeError(f"agent {agent.name!r} has no model client")
        self.engine = Engine(agent, provider)
        self.policy = TurnPolicy(max_steps=agent.max_steps)

    # --------------------------------------------------------------- hosting
    def topics(self) -> list[str]:
        return topology.derive_topics(
            self.node_name, tuple(e.name for e in self.agent.listen_refs),
            self.machinery.namespace)

    async def start(self) -> None:
        pass

    async def stop(self) -> None:
        pass

    # ------------------------------------------------------------- pipeline
    async def handle(self, raw: RawDelivery) -> None:
        ledger = self.machinery.ledger(self.node_name, raw.key, "agent")
        # 1. decode
        try:
            delivery = self.machinery.decoder.decode(raw)
        except DecodeError as e:
            ledger.note(outcome="decode_floor", error=str(e))
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return

        thread_id = delivery.thread_id
        ledger.thread_id = thread_id
        carriage = ThreadCarriage(self.machinery.store, self.node_name, thread_id)
        await carriage.hydrate()

        # 2. admit
        kind, call_id = _admission_key(delivery)
        admission = carriage.admit(kind, call_id)
        ledger.note(admission=admission.value, delivery_kind=kind)
        if admission == Admission.DISCARD:
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        if admission == Admission.DEAD:
            await self.funnel.emit(
                RuntimeError(f"thread {thread_id} is dead"), thread_id=thread_id,
                reply=getattr(delivery, "reply", None),
                marker=getattr(delivery, "marker", None))
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        if admission == Admission.ABSORBED:
            await self._absorb(carriage, delivery, ledger)
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return

        hop = int(carriage.record.metadata.get("hop", 0)) + 1
        if hop > MAX_HOPS:
            await carriage.mark_dead()
            await self.funnel.emit(
                RuntimeError(f"thread {thread_id} exceeded {MAX_HOPS} hops"),
                thread_id=thread_id, reply=_stored_reply(carriage))
            await ledger.flush(self.machinery.bus, self.machinery.namespace)
            return
        carriage.record.metadata["hop"] = hop

        try:
            await self._proceed(carriage, delivery, ledger, hop)
        except Exception as exc:  # the hop-level fault floor
            # echo the ORIGINAL caller's marker so the fault correlates at
            # the client / calling agent
            marker = getattr(delivery, "marker", None)
            pm = carriage.record.metadata.get("peer_marker")
            if marker is None and pm:
                marker = EchoMarker.from_wire(pm)
            await self.funnel.emit(exc, thread_id=thread_id,
                                   reply=_stored_reply(carriage) or
                                   getattr(delivery, "reply", None),
                                   marker=marker)
            ledger.note(outcome="fault", error=str(exc))
        await ledger.flush(self.machinery.bus, self.machinery.namespace)

    # ---------------------------------------------------------------- hops
    async def _proceed(self, carriage: ThreadCarriage, delivery: Delivery,
                       ledger: Any, hop: int) -> None:
        thread_id = carriage.thread_id
        if isinstance(delivery, Call):
            if delivery.reply is not None:
                carriage.record.metadata["reply"] = delivery.reply.to_wire()
            if delivery.handoff is not None:
                carriage.record.metadata["handoff_from"] = delivery.handoff.get("from", "")
            if delivery.seed and not carriage.messages:
                carriage.working_extend(list(delivery.seed))
            if delivery.metadata.get("marker"):
                carriage.record.metadata["peer_marker"] = delivery.metadata["marker"]
            if delivery.metadata.get("model_settings"):
                carriage.record.metadata["model_settings"] = \
                    delivery.metadata["model_settings"]
            carriage.working_extend([delivery.message])
            await self._step_and_realize(carriage, ledger, hop,
                                         handoff=delivery.handoff)
            return

        if isinstance(delivery, (Continuation, Fault)):
            complete = await self._record_continuation(carriage, delivery)
            if not complete:
                await carriage.flush(status=ThreadStatus.PENDING,
                                     pending=carriage.record.pending,
                                     frame_id=carriage.record.frame_id)
                ledger.note(outcome="frame_partial")
                return
            # frame complete: synthesize the tool message, clear, step again
            returns = [
                ToolReturnPart(call_id=cid, tool_name=p["tool_name"],
                               content=p.get("content"),
                               is_error=bool(p.get("is_error", False)))
                for cid, p in carriage.record.pending.items()
            ]
            carriage.working_extend([ToolMessage(tuple(returns))])
            carriage.record.pending = {}
            await self._step_and_realize(carriage, ledger, hop)
            return

        raise TypeError(f"agent node got {type(delivery).__name__}")

    async def _record_continuation(self, carriage: ThreadCarriage,
                                   delivery: Continuation | Fault) -> bool:
        """Record one arrival into the pending frame; True when complete."""
        pending = carriage.record.pending
        marker = delivery.marker
        assert marker is not None
        entry = pending.get(marker.call_id)
        if entry is None:
            return False
        if isinstance(delivery, Fault):
            ctx = self._ctx(carriage)
            part = await self.agent.hook_set.run_on_tool_error(
                ctx, delivery.report, marker.call_id, entry.get("tool_name", "?"))
            entry["content"] = part.content
            entry["is_error"] = part.is_error
        else:
            content = delivery.content
            if delivery.message is not None:
                content = parts_to_text(delivery.message.parts) or content
            entry["content"] = content
            entry["is_error"] = delivery.is_error
        entry["done"] = True
        return all(e.get("done") for e in pending.values())

    async def _absorb(self, carriage: ThreadCarriage, delivery: Delivery,
                      ledger: Any) -> None:
        """Record without stepping (early message mid-frame, duplicate reply)."""
        if isinstance(delivery, Call):
            carriage.working_extend([delivery.message])
            await carriage.flush(status=carriage.record.status,
                                 pending=carriage.record.pending,
                                 frame_id=carriage.record.frame_id)
            ledger.note(outcome="absorbed_call")
        else:
            ledger.note(outcome="absorbed_duplicate")

    # ------------------------------------------------------- step + realize
    async def _step_and_realize(self, carriage: ThreadCarriage, ledger: Any,
                                hop: int, handoff: dict[str, Any] | None = None) -> None:
        thread_id = carriage.thread_id
        ctx = self._ctx(carriage)
        if handoff is not None:
            ctx.metadata["handoff_incoming"] = handoff.get("reason", "") or \
                f"from {handoff.get('from', '?')}"
            await self.agent.hook_set.run_on_handoff(
                ctx, self.node_name, handoff.get("reason", ""))

        # 3. surface
        view = await self.machinery.directory.view()
        surface = build_step_surface(self.agent, view, strict=False)

        # 4. hooks
        if not await self.agent.hook_set.run_on_delivery(ctx, None):
            ledger.note(outcome="dropped_by_hook")
            return
        amendment = await self.agent.hook_set.run_before_step(ctx)
        if amendment.inject:
            carriage.working_extend(list(amendment.inject))

        # 5. drive (pure)
        system = self.agent.render_system_prompt(ctx)
        if amendment.extra_instructions:
            system = f"{system}\n\n{amendment.extra_instructions}"
        history = project_for_model(carriage.working_view(),
                                    agent_name=self.node_name, system_prompt=system)
        loop = asyncio.get_running_loop()
        sink = _BusSink(self, thread_id, _stored_reply(carriage), loop)
        sink.emit(StreamChunk(event="step_started", data={"hop": hop}))
        settings_override = amendment.settings
        if settings_override is None:
            raw_ms = carriage.record.metadata.get("model_settings")
            if raw_ms:
                from ..schemas import ModelSettings

                known = {k: v for k, v in raw_ms.items()
                         if k in ModelSettings.__dataclass_fields__}
                if "stop" in known:
                    known["stop"] = tuple(known["stop"])
                settings_override = ModelSettings(**known)
        resolution = await drive_step(self.engine, surface, self.policy, history,
                                      sink, ctx, settings_override=settings_override)
        ledger.note(steps=len([m for m in resolution.new_messages
                               if isinstance(m, AssistantMessage)]),
                    pending_tools=len(resolution.pending_tool_calls),
                    pending_peers=len(resolution.pending_peer_calls))
        facts = next((m.facts for m in reversed(resolution.new_messages)
                      if isinstance(m, AssistantMessage) and m.facts), None)
        if facts is not None:
            ledger.note(prefill_ms=facts.prefill_ms, decode_ms=facts.decode_ms,
                        prompt_tokens=facts.usage.prompt_tokens,
                        completion_tokens=facts.usage.completion_tokens)

        # 6. realize — persist BEFORE publish
        carriage.working_extend(resolution.new_messages)
        await self._realize(carriage, resolution, ledger, hop, sink)
        sink.emit(StreamChunk(event="step_finished",
                              data={"finished": resolution.finished}))
        await sink.drain()

    async def _realize(self, carriage: ThreadCarriage, resolution: StepResolution,
                       ledger: Any, hop: int, sink: _BusSink) -> None:
        thread_id = carriage.thread_id
        reply = _stored_reply(carriage)

        transfer = next((e for e in resolution.effects if isinstance(e, Transfer)), None)
        if transfer is not None:
            # handoff: persist CLOSED, then seed the target with the projection;
            # the original caller's echo marker travels so the target's final
            # answer correlates at the caller ("follows handoffs across turns")
            peer_marker = carriage.record.metadata.pop("peer_marker", None)
            await carriage.flush(status=ThreadStatus.CLOSED, pending={}, frame_id=None)
            seed = project_for_transfer(carriage.messages, source=self.node_name,
                                        target=transfer.target, reason=transfer.reason)
            notice = seed[-1]
            await self.publisher.handoff(
                inbox=transfer.target_inbox, thread_id=thread_id,
                message=notice, seed=seed[:-1], reason=transfer.reason,
                reply=reply,
                marker=EchoMarker.from_wire(peer_marker) if peer_marker else None,
                hop=hop)
            await self.agent.hook_set.run_on_handoff(
                self._ctx(carriage), transfer.target, transfer.reason)
            sink.emit(StreamChunk(event="handoff",
                                  data={"to": transfer.target, "reason": transfer.reason}))
            ledger.note(outcome="handoff", target=transfer.target)
            return

        if resolution.finished:
            peer_marker = carriage.record.metadata.pop("peer_marker", None)
            await carriage.flush(status=ThreadStatus.OPEN, pending={}, frame_id=None)
            await self._publish_dispatches(resolution, thread_id, hop)
            final = resolution.final
            if final is not None:
                if reply is not None and reply.inbox:
                    await self.publisher.reply_continuation(
                        reply=reply, thread_id=thread_id,
                        marker=EchoMarker.from_wire(peer_marker) if peer_marker
                        else EchoMarker(call_id=f"turn-{hop}", caller=self.node_name),
                        content=resolution.output if resolution.output is not None
                        else parts_to_text(final.parts),
                        message=final, hop=hop)
            ledger.note(outcome="finished")
            return

        # pending frame: persist the batch, then fan out
        pending: dict[str, Any] = {}
        for p in resolution.pending_tool_calls:
            pending[p.call_id] = {"tool_name": p.tool_name, "done": False}
        for p in resolution.pending_peer_calls:
            pending[p.call_id] = {"tool_name": "send_message", "done": False,
                                   "peer": p.target}
        await carriage.flush(status=ThreadStatus.PENDING, pending=pending,
                             frame_id=resolution.frame_id)

        my_reply = ReplySlot(
            inbox=topology.inbox_topic(self.node_name, self.machinery.namespace),
            key=thread_id, node=self.node_name)
        for p in resolution.pending_tool_calls:
            await self.publisher.call_tool(
                inbox=p.target_inbox, thread_id=thread_id, tool_name=p.tool_name,
                arguments=p.arguments,
                marker=EchoMarker(call_id=p.call_id,
                                  frame_id=resolution.frame_id or "",
                                  caller=self.node_name),
                reply=my_reply, hop=hop)
        await self._publish_dispatches(resolution, thread_id, hop, reply=my_reply)
        ledger.note(outcome="pending_frame", frame=resolution.frame_id)

    async def _publish_dispatches(self, resolution: StepResolution, thread_id: str,
                                  hop: int, reply: ReplySlot | None = None) -> None:
        for e in resolution.effects:
            if not isinstance(e, Dispatch):
                continue
            # a peer call shares the thread's grain key lineage but runs on the
            # peer's own thread row; the reply correlates by echo marker
            peer_thread = f"{thread_id}:{e.target}"
            metadata = {}
            if e.reply_expected and reply is not None:
                metadata["marker"] = EchoMarker(call_id=e.call_id, caller=self.node_name).to_wire()
            from .._wire.deliveries import agent_call_payload
            await self.machinery.transport.send(
                "agent_call", e.target_inbox, peer_thread, to=e.target,
                thread_id=peer_thread,
                payload=agent_call_payload(e.message, metadata=metadata),
                reply=reply if e.reply_expected else None,
                marker=EchoMarker(call_id=e.call_id, caller=self.node_name)
                if e.reply_expected else None,
                emitter=self.node_name, hop=hop)

    # ---------------------------------------------------------------- misc
    def _ctx(self, carriage: ThreadCarriage) -> TurnContext:
        async def emit(event: str, payload: Any) -> None:
            await self.publisher.emit_event(event=event, thread_id=carriage.thread_id,
                                            data={"payload": payload})

        return TurnContext(agent_name=self.node_name, thread_id=carriage.thread_id,
                           turn=int(carriage.record.metadata.get("hop", 0)),
                           metadata=carriage.record.metadata, emit=emit)


def _admission_key(delivery: Delivery) -> tuple[str, str | None]:
    if isinstance(delivery, Call):
        return "call", None
    if isinstance(delivery, Continuation):
        return "continuation", delivery.marker.call_id
    if isinstance(delivery, Fault):
        return "fault", delivery.marker.call_id if delivery.marker else None
    return "call", None


def _stored_reply(carriage: ThreadCarriage) -> ReplySlot | None:
    return ReplySlot.from_wire(carriage.record.metadata.get("reply"))
