"""Kafka adapters: ``KafkaBus`` + ``KafkaDirectory`` — the sole aiokafka
importer.

Capability parity: reference ``adapters/kafka``
(/root/reference/src/calfkit/adapters/kafka/__init__.py:1-3): key-ordered
lane machinery as private internals, compacted control-plane topics + ktable
materialization + heartbeats, opt-in producer idempotence
(/root/reference/CHANGELOG.md:90), explicit topic provisioning with no
reliance on broker auto-create (/root/reference/CHANGELOG.md:297-301,329).

Byte compatibility: record values and headers pass through VERBATIM — the
JSON envelope + ``x-calf-kind`` headers from ``calfkit_amd._wire`` are the
wire format on Kafka exactly as on the bundled mesh broker.

The aiokafka SDK is not installed in this offline image; the adapter logic
is exercised by tests/unit/test_kafka_adapter.py against a faked aiokafka
module with the same surface, and runs unchanged against the real SDK.
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, AsyncIterator

from ...ports import Advert, DeliveryHandler, DirectoryView, GrainKey, RawDelivery
from ..._wire import topology

_log = logging.getLogger("calfkit.kafka")

__all__ = ["KafkaBus", "KafkaDirectory", "HEARTBEAT_S", "ADVERT_TTL_S"]

HEARTBEAT_S = 10.0
ADVERT_TTL_S = 3 * HEARTBEAT_S


def _aiokafka():
    try:
        import aiokafka  # the one allowed import site
    except ImportError as e:  # pragma: no cover - exercised only offline
        raise RuntimeError(
            "KafkaBus needs the aiokafka package; in broker-free setups use "
            "calfkit_amd.adapters.mesh (same envelope format)") from e
    return aiokafka


class _Lane:
    def __init__(self, maxsize: int = 0) -> None:
        self.queue: asyncio.Queue = asyncio.Queue(maxsize=maxsize)
        self.task: asyncio.Task | None = None


class _OffsetLedger:
    """Per-partition low-watermark of HANDLED offsets.

    Offsets are committed only after the lane handler completes, so a worker
    crash re-delivers in-flight work (at-least-once) instead of silently
    dropping it the way auto-commit did.  Handlers run
    out of order across lanes, so completion is tracked as a contiguity
    watermark per (topic, partition).
    """

    def __init__(self) -> None:
        self._next: dict[tuple[str, int], int] = {}
        self._done: dict[tuple[str, int], set[int]] = {}

    def expect(self, tp: tuple[str, int], offset: int) -> None:
        """Called in consume order (= offset order within a partition)."""
        self._next.setdefault(tp, offset)

    def complete(self, tp: tuple[str, int], offset: int) -> int | None:
        """Mark handled; returns the new commit watermark when it advanced
        (the offset AFTER the last contiguously handled record)."""
        done = self._done.setdefault(tp, set())
        done.add(offset)
        nxt = self._next[tp]
        advanced = False
        while nxt in done:
            done.remove(nxt)
            nxt += 1
            advanced = True
        if advanced:
            self._next[tp] = nxt
            return nxt
        return None


class KafkaBus:
    """MessageBus over Kafka with client-side key-ordered lanes.

    A consumer-group member receives a key's records in partition order;
    the lane layer additionally serializes HANDLER execution per key while
    letting different keys run concurrently — the reference's key-ordered
    serialized consumption (/root/reference/CHANGELOG.md:31).
    """

    def __init__(self, bootstrap: str, *, client_id: str = "calfkit",
                 enable_idempotence: bool = False,
                 max_message_bytes: int = 1024 * 1024,
                 provision_topics: bool = True,
                 partitions: int = 16,
                 lane_queue_max: int = 256):
        self.bootstrap = bootstrap
        self.client_id = client_id
        self.enable_idempotence = enable_idempotence
        self.max_message_bytes = max_message_bytes
        self.provision_topics = provision_topics
        self.partitions = partitions
        #: bound on queued-but-unhandled deliveries per lane; when a lane
        #: backs up the consume loop awaits on the full queue, which pauses
        #: fetching (client-side backpressure instead of unbounded memory)
        self.lane_queue_max = lane_queue_max
        self._producer: Any = None
        self._consumers: list[Any] = []
        self._tasks: list[asyncio.Task] = []
        self._lanes: dict[tuple[str, GrainKey], _Lane] = {}
        self._closed = False

    async def _ensure_producer(self) -> Any:
        if self._producer is None:
            ak = _aiokafka()
            self._producer = ak.AIOKafkaProducer(
                bootstrap_servers=self.bootstrap, client_id=self.client_id,
                enable_idempotence=self.enable_idempotence,
                max_request_size=self.max_message_bytes)
            await self._producer.start()
        return self._producer

    # ------------------------------------------------------------------ port
    async def publish(self, topic: str, key: GrainKey, value: bytes,
                      headers: tuple[tuple[str, bytes], ...] = ()) -> None:
        producer = await self._ensure_producer()
        await producer.send_and_wait(topic, value=value, key=key.encode(),
                                     headers=list(headers))

    async def subscribe(self, topic: str, handler: DeliveryHandler, *,
                        group: str = "") -> None:
        ak = _aiokafka()
        consumer = ak.AIOKafkaConsumer(
            topic, bootstrap_servers=self.bootstrap,
            group_id=group or None,
            client_id=self.client_id,
            auto_offset_reset="latest",
            # offsets are committed AFTER the lane handler completes (see
            # _OffsetLedger); auto-commit would ack queued-but-unhandled
            # deliveries and drop them on a crash
            enable_auto_commit=False)
        await consumer.start()
        self._consumers.append(consumer)
        self._tasks.append(asyncio.get_running_loop().create_task(
            self._consume(consumer, handler,
                          _OffsetLedger() if group else None)))

    async def _consume(self, consumer: Any, handler: DeliveryHandler,
                       ledger: "_OffsetLedger | None") -> None:
        try:
            async for msg in consumer:
                raw = RawDelivery(
                    topic=msg.topic,
                    key=(msg.key or b"").decode(),
                    value=msg.value or b"",
                    headers=tuple(msg.headers or ()),
                    offset=msg.offset,
                    partition=msg.partition)
                if ledger is not None:
                    ledger.expect((raw.topic, raw.partition), raw.offset)
                await self._enqueue(raw, handler, consumer, ledger)
        except asyncio.CancelledError:
            pass
        except Exception:
            _log.exception("kafka consume loop died")

    async def _enqueue(self, raw: RawDelivery, handler: DeliveryHandler,
                       consumer: Any, ledger: "_OffsetLedger | None") -> None:
        lane_key = (raw.topic, raw.key)
        lane = self._lanes.get(lane_key)
        if lane is None:
            lane = _Lane(maxsize=self.lane_queue_max)
            self._lanes[lane_key] = lane
            lane.task = asyncio.get_running_loop().create_task(
                self._run_lane(lane))
        # blocks when the lane is full -> the consume loop stops fetching
        await lane.queue.put((raw, handler, consumer, ledger))

    async def _run_lane(self, lane: _Lane) -> None:
        while True:
            raw, handler, consumer, ledger = await lane.queue.get()
            try:
                await handler(raw)
            except Exception:
                # handler faults are the fault rail's business (in-band
                # ErrorReport); the delivery still counts as handled
                _log.exception("kafka handler failed on %s", raw.topic)
            if ledger is not None:
                watermark = ledger.complete((raw.topic, raw.partition),
                                            raw.offset)
                if watermark is not None:
                    try:
                        ak = _aiokafka()
                        tp = ak.TopicPartition(raw.topic, raw.partition)
                        await consumer.commit({tp: watermark})
                    except Exception:
                        _log.exception("offset commit failed on %s", raw.topic)

    async def ensure_topics(self, topics: list[str]) -> None:
        """Provision at startup — never hang on no-auto-create brokers."""
        if not self.provision_topics or not topics:
            return
        ak = _aiokafka()
        admin = ak.admin.AIOKafkaAdminClient(bootstrap_servers=self.bootstrap)
        await admin.start()
        try:
            news = [ak.admin.NewTopic(name=t, num_partitions=self.partitions,
                                      replication_factor=1,
                                      topic_configs=_topic_config(t))
                    for t in topics]
            try:
                await admin.create_topics(news, validate_only=False)
            except Exception as e:  # TopicAlreadyExistsError and friends
                if "exist" not in type(e).__name__.lower() and \
                        "exist" not in str(e).lower():
                    raise
        finally:
            await admin.close()

    async def close(self) -> None:
        self._closed = True
        for t in self._tasks:
            t.cancel()
        for lane in self._lanes.values():
            if lane.task is not None:
                lane.task.cancel()
        for c in self._consumers:
            await c.stop()
        if self._producer is not None:
            await self._producer.stop()


def _topic_config(topic: str) -> dict[str, str]:
    if ".directory" in topic:
        return {"cleanup.policy": "compact"}
    return {}


class KafkaDirectory:
    """Directory over a compacted control-plane topic.

    Adverts are keyed by node and re-published as heartbeats; the local view
    is a ktable materialized by a background consumer from the earliest
    offset; entries older than ADVERT_TTL_S drop out of the view (liveness).
    """

    def __init__(self, bootstrap: str, *, namespace: str = "",
                 client_id: str = "calfkit-dir"):
        self.bootstrap = bootstrap
        self.topic = topology.directory_topic(namespace)
        self.client_id = client_id
        self._producer: Any = None
        self._table: dict[str, Advert] = {}
        self._own: dict[str, Advert] = {}
        self._materializer: asyncio.Task | None = None
        self._heartbeat: asyncio.Task | None = None
        self._caught_up = asyncio.Event()
        self._watchers: list[asyncio.Queue] = []

    async def _ensure(self) -> None:
        if self._producer is None:
            ak = _aiokafka()
            self._producer = ak.AIOKafkaProducer(
                bootstrap_servers=self.bootstrap, client_id=self.client_id)
            await self._producer.start()
        if self._materializer is None:
            self._materializer = asyncio.get_running_loop().create_task(
                self._materialize())
        if self._heartbeat is None:
            self._heartbeat = asyncio.get_running_loop().create_task(
                self._heartbeat_loop())

    async def _materialize(self) -> None:
        ak = _aiokafka()
        consumer = ak.AIOKafkaConsumer(
            self.topic, bootstrap_servers=self.bootstrap,
            group_id=None, auto_offset_reset="earliest")
        await consumer.start()
        try:
            self._caught_up.set()  # live view builds as records arrive
            async for msg in consumer:
                node = (msg.key or b"").decode()
                if not node:
                    continue
                if not msg.value:  # tombstone
                    self._table.pop(node, None)
                else:
                    d = json.loads(msg.value)
                    self._table[node] = Advert(
                        node=node, kind=d.get("kind", ""),
                        description=d.get("description", ""),
                        capabilities=d.get("capabilities", {}),
                        inbox=d.get("inbox", ""), worker=d.get("worker", ""),
                        ts=float(d.get("ts", 0.0)))
                for q in self._watchers:
                    q.put_nowait(None)
        except asyncio.CancelledError:
            pass
        finally:
            await consumer.stop()

    async def _heartbeat_loop(self) -> None:
        try:
            while True:
                await asyncio.sleep(HEARTBEAT_S)
                for advert in list(self._own.values()):
                    await self._publish(advert)
        except asyncio.CancelledError:
            pass

    async def _publish(self, advert: Advert) -> None:
        payload = json.dumps({
            "kind": advert.kind, "description": advert.description,
            "capabilities": advert.capabilities, "inbox": advert.inbox,
            "worker": advert.worker, "ts": time.time()}).encode()
        await self._producer.send_and_wait(self.topic, value=payload,
                                           key=advert.node.encode())

    # ------------------------------------------------------------------ port
    async def advertise(self, advert: Advert) -> None:
        await self._ensure()
        self._own[advert.node] = advert
        await self._publish(advert)

    async def withdraw(self, node: str) -> None:
        await self._ensure()
        self._own.pop(node, None)
        await self._producer.send_and_wait(self.topic, value=None,
                                           key=node.encode())

    async def view(self) -> DirectoryView:
        await self._ensure()
        await self._caught_up.wait()
        now = time.time()
        live = {n: a for n, a in self._table.items()
                if now - a.ts < ADVERT_TTL_S}
        return DirectoryView(adverts=live)

    async def watch(self) -> AsyncIterator[DirectoryView]:
        await self._ensure()
        q: asyncio.Queue = asyncio.Queue()
        self._watchers.append(q)
        try:
            yield await self.view()
            while True:
                await q.get()
                yield await self.view()
        finally:
            self._watchers.remove(q)

    async def close(self) -> None:
        for t in (self._materializer, self._heartbeat):
            if t is not None:
                t.cancel()
        if self._producer is not None:
            await self._producer.stop()
