"""Client — the caller surface. Holds no node machinery.

Capability parity: reference ``client/``
(/root/reference/src/calfkit/client/__init__.py:1-2); gateway +
send/start/execute + firehose per /root/reference/CHANGELOG.md:161;
``reply_to`` return address per /root/reference/CHANGELOG.md:223-227;
fire-and-forget emit per /root/reference/CHANGELOG.md:336-340; client-side
output-type deserialization per /root/reference/CHANGELOG.md:487; caller-side
mesh view per /root/reference/CHANGELOG.md:111; shared correlation-based
reply dispatcher per /root/reference/CHANGELOG.md:486,572.
"""

from __future__ import annotations

import asyncio
import json
import uuid
from dataclasses import dataclass
from typing import Any, AsyncIterator

from .._wire import topology
from .._wire.deliveries import (
    EchoMarker,
    KIND_HEADER,
    ReplySlot,
    agent_call_payload,
    encode_envelope,
)
from ..definitions.tools import _coerce
from ..errors import NodeFaultError, ErrorReport, TimeoutError_
from ..messages import (
    AssistantMessage,
    Message,
    TextPart,
    UserMessage,
    message_from_wire,
    parts_to_text,
    value_to_parts,
)
from ..ports import Directory, DirectoryView, MessageBus, RawDelivery
from ..streaming import StreamChunk

__all__ = ["Client", "TurnResult"]


@dataclass
class TurnResult:
    """What ``execute`` resolves to."""

    message: AssistantMessage | None
    content: Any = None
    thread_id: str = ""
    output: Any = None

    @property
    def text(self) -> str:
        if self.message is not None:
            return parts_to_text(self.message.parts)
        return str(self.content) if self.content is not None else ""


class Client:
    """The agent gateway for callers."""

    def __init__(self, bus: MessageBus, directory: Directory, *,
                 namespace: str = "", client_id: str = "",
                 max_message_bytes: int = 1024 * 1024):
        self.bus = bus
        self.directory = directory
        self.namespace = namespace
        self.client_id = client_id or f"client-{uuid.uuid4().hex[:8]}"
        self.max_message_bytes = max_message_bytes
        self.reply_inbox = topology.reply_topic(self.client_id, namespace)
        self._futures: dict[str, asyncio.Future] = {}
        self._firehoses: dict[str, list[asyncio.Queue]] = {}
        self._started = False

    # ------------------------------------------------------------- connect
    @classmethod
    async def connect(cls, bus: MessageBus | None = None,
                      directory: Directory | None = None, *,
                      bootstrap: str = "", namespace: str = "",
                      client_id: str = "", **kw: Any) -> "Client":
        """Connect to a mesh.

        With no arguments an in-memory mesh is assumed (pass the worker's bus
        and directory).  With ``bootstrap`` a broker-backed mesh is dialed
        via the mesh adapter (lazy import).
        """
        if bus is None or directory is None:
            if not bootstrap:
                raise ValueError("Client.connect needs (bus, directory) or bootstrap=")
            from ..adapters.mesh import connect_mesh  # lazy adapter import
            bus, directory = await connect_mesh(bootstrap, namespace=namespace)
        client = cls(bus, directory, namespace=namespace, client_id=client_id, **kw)
        await client._start()
        return client

    async def _start(self) -> None:
        if self._started:
            return
        await self.bus.ensure_topics([self.reply_inbox])
        await self.bus.subscribe(self.reply_inbox, self._on_reply)
        self._started = True

    async def close(self) -> None:
        for fut in self._futures.values():
            if not fut.done():
                fut.cancel()
        self._futures.clear()

    # ------------------------------------------------------------ dispatcher
    async def _on_reply(self, raw: RawDelivery) -> None:
        kind = raw.header(KIND_HEADER)
        kind = kind.decode() if kind else ""
        if kind == "chunk":
            chunk = StreamChunk.from_wire(json.loads(raw.value))
            for q in self._firehoses.get(chunk.thread_id, ()):  # reply-inbox chunks
                q.put_nowait(chunk)
            return
        envelope = json.loads(raw.value.decode())
        marker = (envelope.get("state") or {}).get("marker") or {}
        call_id = marker.get("call_id", "")
        fut = self._futures.pop(call_id, None)
        payload = envelope.get("payload", {})
        if fut is None or fut.done():
            return
        if kind == "fault" or payload.get("type") == "fault":
            fut.set_exception(NodeFaultError(
                ErrorReport.from_wire(payload.get("report", {}))))
            return
        msg = payload.get("message")
        fut.set_result(TurnResult(
            message=message_from_wire(msg) if msg else None,
            content=payload.get("content"),
            thread_id=(envelope.get("routing") or {}).get("thread_id", ""),
        ))

    # ----------------------------------------------------------------- verbs
    async def send(self, agent: str, content: Any, *, thread_id: str | None = None,
                   reply_to: str = "") -> str:
        """One-way message to an agent inbox. Returns the thread id."""
        thread_id = thread_id or uuid.uuid4().hex
        message = self._as_message(content)
        reply = None
        if reply_to:
            reply = ReplySlot(inbox=topology.reply_topic(reply_to, self.namespace),
                              key=thread_id, node=reply_to)
        await self._publish_call(agent, thread_id, message, reply, None)
        return thread_id

    async def start(self, agent: str, content: Any, *,
                    thread_id: str | None = None) -> str:
        """Begin a thread with replies routed here, without blocking."""
        thread_id = thread_id or uuid.uuid4().hex
        message = self._as_message(content)
        reply = ReplySlot(inbox=self.reply_inbox, key=thread_id, node=self.client_id)
        await self._publish_call(agent, thread_id, message, reply, None)
        return thread_id

    async def execute(self, agent: str, content: Any, *,
                      thread_id: str | None = None, timeout: float = 300.0,
                      output_type: Any = None,
                      model_settings: Any = None) -> TurnResult:
        """Request/response: publish, await the reply on the private inbox.

        ``model_settings`` overrides the agent's settings for this one turn
        (runtime model_settings, /root/reference/CHANGELOG.md:406).
        """
        thread_id = thread_id or uuid.uuid4().hex
        call_id = uuid.uuid4().hex
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._futures[call_id] = fut
        message = self._as_message(content)
        reply = ReplySlot(inbox=self.reply_inbox, key=thread_id, node=self.client_id)
        marker = EchoMarker(call_id=call_id, caller=self.client_id)
        await self._publish_call(agent, thread_id, message, reply, marker,
                                 model_settings=model_settings)
        try:
            result: TurnResult = await asyncio.wait_for(fut, timeout)
        except asyncio.TimeoutError:
            self._futures.pop(call_id, None)
            raise TimeoutError_(f"no reply from {agent!r} within {timeout}s")
        result.thread_id = result.thread_id or thread_id
        if output_type is not None:
            result.output = self._deserialize_output(result, output_type)
        return result

    async def emit(self, event: str, data: Any, *, thread_id: str = "") -> None:
        """Caller-side fire-and-forget event emission."""
        payload = agent_call_payload(UserMessage(
            (TextPart(json.dumps({"event": event, "data": data})),), author="event"))
        value, headers = encode_envelope("agent_call",
                                         to=topology.event_topic(event, self.namespace),
                                         thread_id=thread_id, payload=payload,
                                         emitter=self.client_id)
        await self.bus.publish(topology.event_topic(event, self.namespace),
                               thread_id or event, value, headers)

    async def firehose(self, thread_id: str) -> AsyncIterator[StreamChunk]:
        """Subscribe to the intermediate step events of one thread."""
        q: asyncio.Queue = asyncio.Queue()
        self._firehoses.setdefault(thread_id, []).append(q)
        topic = topology.firehose_topic(thread_id, self.namespace)

        async def pump(raw: RawDelivery) -> None:
            q.put_nowait(StreamChunk.from_wire(json.loads(raw.value)))

        await self.bus.subscribe(topic, pump)
        try:
            while True:
                chunk = await q.get()
                yield chunk
        finally:
            self._firehoses[thread_id].remove(q)

    async def mesh(self) -> DirectoryView:
        """Caller-side live view of the mesh."""
        return await self.directory.view()

    # ------------------------------------------------------------- internals
    async def _publish_call(self, agent: str, thread_id: str, message: Message,
                            reply: ReplySlot | None, marker: EchoMarker | None,
                            model_settings: Any = None) -> None:
        inbox = topology.inbox_topic(agent, self.namespace)
        metadata = None
        if model_settings is not None:
            from dataclasses import asdict

            metadata = {"model_settings": asdict(model_settings)}
        value, headers = encode_envelope(
            "agent_call", to=agent, thread_id=thread_id,
            payload=agent_call_payload(message, metadata=metadata),
            reply=reply, marker=marker,
            emitter=self.client_id)
        if len(value) > self.max_message_bytes:
            raise ValueError(f"message {len(value)}B exceeds max_message_bytes")
        await self.bus.publish(inbox, thread_id, value, headers)

    @staticmethod
    def _as_message(content: Any) -> Message:
        if isinstance(content, (UserMessage,)):
            return content
        return UserMessage(value_to_parts(content))

    @staticmethod
    def _deserialize_output(result: TurnResult, output_type: Any) -> Any:
        value = result.content
        if value is None and result.message is not None:
            try:
                value = json.loads(parts_to_text(result.message.parts))
            except json.JSONDecodeError:
                return None
        if isinstance(value, dict) and set(value) == {"final_result"}:
            value = value["final_result"]
        try:
            return _coerce(value, output_type)
        except (TypeError, ValueError):
            return value
