"""Unit: KafkaBus/KafkaDirectory logic against a faked aiokafka SDK.

The real SDK is absent offline; this fake implements the aiokafka surface
the adapter touches (producer send_and_wait, consumer async-iteration with
groups, admin create_topics) over an in-memory broker, so the adapter's
lanes / provisioning / compaction / heartbeat logic is tested for real.
An opt-in real-broker lane (CALF_TEST_KAFKA_BOOTSTRAP) mirrors the
reference's taxonomy when a broker exists.
"""

from __future__ import annotations

import asyncio
import sys
import types
from collections import defaultdict

import pytest


# ------------------------------------------------------- the faked aiokafka

class _Msg:
    def __init__(self, topic, key, value, headers, offset, partition=0):
        self.topic, self.key, self.value = topic, key, value
        self.headers, self.offset, self.partition = headers, offset, partition


class _FakeCluster:
    def __init__(self):
        self.logs: dict[str, list[_Msg]] = defaultdict(list)
        self.consumers: list["_FakeConsumer"] = []
        self.topics: dict[str, dict] = {}
        self.group_rr: dict[tuple[str, str], int] = defaultdict(int)

    def produce(self, topic, key, value, headers):
        log = self.logs[topic]
        msg = _Msg(topic, key, value, tuple(headers or ()), len(log))
        log.append(msg)
        # deliver: every group gets one member (round-robin by key hash);
        # ungrouped consumers all get it
        by_group = defaultdict(list)
        for c in self.consumers:
            if topic in c._topics and c._started:
                by_group[c._group].append(c)
        for group, members in by_group.items():
            if group is None:
                for m in members:
                    m._queue.put_nowait(msg)
            else:
                i = hash(key) % len(members)
                members[i]._queue.put_nowait(msg)


CLUSTER = _FakeCluster()


class _FakeProducer:
    def __init__(self, bootstrap_servers=None, **kw):
        self.kw = kw
        self._started = False

    async def start(self):
        self._started = True

    async def stop(self):
        self._started = False

    async def send_and_wait(self, topic, value=None, key=None, headers=None):
        assert self._started
        CLUSTER.produce(topic, key, value, headers)


class _TopicPartition:
    def __init__(self, topic, partition):
        self.topic, self.partition = topic, partition

    def __hash__(self):
        return hash((self.topic, self.partition))

    def __eq__(self, other):
        return (self.topic, self.partition) == (other.topic, other.partition)


class _FakeConsumer:
    def __init__(self, *topics, bootstrap_servers=None, group_id=None,
                 auto_offset_reset="latest", **kw):
        self._topics = topics
        self._group = group_id
        self._queue: asyncio.Queue = asyncio.Queue()
        self._started = False
        self._reset = auto_offset_reset
        self.auto_commit = kw.get("enable_auto_commit", True)
        self.committed: dict[_TopicPartition, int] = {}

    async def commit(self, offsets):
        assert self._group, "commit() without a consumer group"
        self.committed.update(offsets)

    async def start(self):
        self._started = True
        CLUSTER.consumers.append(self)
        if self._reset == "earliest":
            for t in self._topics:
                for msg in CLUSTER.logs.get(t, []):
                    self._queue.put_nowait(msg)

    async def stop(self):
        self._started = False
        if self in CLUSTER.consumers:
            CLUSTER.consumers.remove(self)

    def __aiter__(self):
        return self

    async def __anext__(self):
        if not self._started:
            raise StopAsyncIteration
        return await self._queue.get()


class _NewTopic:
    def __init__(self, name, num_partitions, replication_factor,
                 topic_configs=None):
        self.name = name
        self.num_partitions = num_partitions
        self.topic_configs = topic_configs or {}


class _FakeAdmin:
    def __init__(self, bootstrap_servers=None):
        pass

    async def start(self):
        pass

    async def close(self):
        pass

    async def create_topics(self, topics, validate_only=False):
        for t in topics:
            if t.name in CLUSTER.topics:
                raise RuntimeError(f"TopicAlreadyExists: {t.name}")
            CLUSTER.topics[t.name] = t.topic_configs


@pytest.fixture(autouse=True)
def fake_aiokafka(monkeypatch):
    global CLUSTER
    CLUSTER = _FakeCluster()
    mod = types.ModuleType("aiokafka")
    mod.AIOKafkaProducer = _FakeProducer
    mod.AIOKafkaConsumer = _FakeConsumer
    mod.TopicPartition = _TopicPartition
    admin = types.ModuleType("aiokafka.admin")
    admin.AIOKafkaAdminClient = _FakeAdmin
    admin.NewTopic = _NewTopic
    mod.admin = admin
    monkeypatch.setitem(sys.modules, "aiokafka", mod)
    monkeypatch.setitem(sys.modules, "aiokafka.admin", admin)
    yield mod


async def test_publish_subscribe_round_trip_preserves_bytes():
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus("b:9092")
    seen = []

    async def handler(raw):
        seen.append(raw)

    await bus.subscribe("calf.inbox.x", handler, group="x")
    value = b'{"routing":{}}'
    headers = (("x-calf-kind", b"agent_call"),)
    await bus.publish("calf.inbox.x", "t1", value, headers)
    await asyncio.sleep(0.05)
    assert len(seen) == 1
    assert seen[0].value == value  # byte-compatible pass-through
    assert seen[0].headers == headers
    assert seen[0].key == "t1"
    await bus.close()


async def test_lane_serialization_per_key():
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus("b:9092")
    order = []
    gate = asyncio.Event()

    async def handler(raw):
        if raw.key == "slow" and not order:
            await gate.wait()
        order.append((raw.key, raw.offset))

    await bus.subscribe("t", handler, group="g")
    await bus.publish("t", "slow", b"1")
    await bus.publish("t", "slow", b"2")
    await bus.publish("t", "fast", b"3")
    await asyncio.sleep(0.05)
    # the fast lane ran while the slow lane was blocked
    assert ("fast", 2) in order and len([o for o in order if o[0] == "slow"]) == 0
    gate.set()
    await asyncio.sleep(0.05)
    assert [o for o in order if o[0] == "slow"] == [("slow", 0), ("slow", 1)]
    await bus.close()


async def test_offsets_commit_only_after_handler_completes():
    """at-least-once: a delivery whose handler is still running must not be
    committed; completion advances a contiguous low-watermark per partition
    (auto-commit silently dropped in-flight agent calls)."""
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus("b:9092")
    gate = asyncio.Event()
    handled = []

    async def handler(raw):
        if raw.key == "slow":
            await gate.wait()
        handled.append(raw.offset)

    await bus.subscribe("t", handler, group="g")
    consumer = bus._consumers[0]
    assert consumer.auto_commit is False
    await bus.publish("t", "slow", b"0")   # offset 0, blocked in handler
    await bus.publish("t", "fast", b"1")   # offset 1, completes immediately
    await asyncio.sleep(0.05)
    # offset 1 handled, but 0 still in flight -> nothing committable yet
    assert handled == [1]
    assert consumer.committed == {}
    gate.set()
    await asyncio.sleep(0.05)
    # both handled -> watermark jumps past the contiguous prefix
    from calfkit_amd.adapters.kafka import _aiokafka
    tp = _aiokafka().TopicPartition("t", 0)
    assert consumer.committed == {tp: 2}
    await bus.close()


async def test_lane_backpressure_bounds_queue_and_pauses_consumption():
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus("b:9092", lane_queue_max=2)
    gate = asyncio.Event()
    handled = []

    async def handler(raw):
        await gate.wait()
        handled.append(raw.offset)

    await bus.subscribe("t", handler, group="g")
    for i in range(8):
        await bus.publish("t", "k", b"x")
    await asyncio.sleep(0.05)
    # one in the handler + at most lane_queue_max queued; the rest stay in
    # the consumer (fetch paused), not in unbounded lane memory
    lane = bus._lanes[("t", "k")]
    assert lane.queue.qsize() <= 2
    gate.set()
    await asyncio.sleep(0.1)
    assert handled == list(range(8))  # order preserved end-to-end
    await bus.close()


async def test_topic_provisioning_compacts_directory():
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus("b:9092")
    await bus.ensure_topics(["calf.inbox.a", "calf.directory"])
    assert CLUSTER.topics["calf.directory"] == {"cleanup.policy": "compact"}
    assert CLUSTER.topics["calf.inbox.a"] == {}
    # idempotent: second call with an existing topic does not raise
    await bus.ensure_topics(["calf.inbox.a"])
    await bus.close()


async def test_directory_materializes_and_tombstones():
    from calfkit_amd.adapters.kafka import KafkaDirectory
    from calfkit_amd.ports import Advert

    d1 = KafkaDirectory("b:9092")
    await d1.advertise(Advert(node="agent-a", kind="agent", inbox="i"))
    await asyncio.sleep(0.05)
    view = await d1.view()
    assert "agent-a" in view.adverts
    # a second directory client materializes from the compacted log
    d2 = KafkaDirectory("b:9092")
    view2 = await d2.view()
    await asyncio.sleep(0.05)
    view2 = await d2.view()
    assert "agent-a" in view2.adverts
    await d1.withdraw("agent-a")
    await asyncio.sleep(0.05)
    view = await d1.view()
    assert "agent-a" not in view.adverts
    await d1.close()
    await d2.close()
