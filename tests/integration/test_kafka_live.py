"""Real-broker lane: the Kafka adapters against a LIVE broker.

Parity with the reference's ``-m kafka`` lane
(/root/reference/CONTRIBUTING.md:100-135): point
``CALF_TEST_KAFKA_BOOTSTRAP`` at a running Kafka/Redpanda and run
``pytest -m kafka``.  Without the env var (or without the aiokafka SDK —
this image is offline) the lane SKIPS cleanly rather than failing.

Every topic/group derives from a per-run namespace so runs sharing a
broker don't collide.
"""

import asyncio
import os
import uuid

import pytest

BOOTSTRAP = os.environ.get("CALF_TEST_KAFKA_BOOTSTRAP", "")

pytestmark = [
    pytest.mark.kafka,
    pytest.mark.skipif(not BOOTSTRAP,
                       reason="no CALF_TEST_KAFKA_BOOTSTRAP broker"),
    pytest.mark.skipif(
        __import__("importlib").util.find_spec("aiokafka") is None,
        reason="aiokafka SDK not installed"),
]


@pytest.fixture()
def namespace() -> str:
    return f"calftest-{uuid.uuid4().hex[:8]}"


async def test_bus_round_trip(namespace):
    from calfkit_amd.adapters.kafka import KafkaBus

    bus = KafkaBus(BOOTSTRAP, client_id=namespace)
    topic = f"{namespace}.input"
    try:
        await bus.ensure_topics([topic])
        got: asyncio.Queue = asyncio.Queue()

        async def handler(raw):
            got.put_nowait(raw)

        await bus.subscribe(topic, handler, group=f"{namespace}.g")
        await asyncio.sleep(1.0)  # consumer group join
        await bus.publish(topic, "k1", b'{"x":1}',
                          headers=(("x-calf-kind", b"probe"),))
        raw = await asyncio.wait_for(got.get(), 15)
        assert raw.key == "k1" and raw.value == b'{"x":1}'
        assert dict(raw.headers).get("x-calf-kind") == b"probe"
    finally:
        await bus.close()


async def test_directory_advertise_and_view(namespace):
    from calfkit_amd.adapters.kafka import KafkaBus, KafkaDirectory
    from calfkit_amd.ports import Advert

    bus = KafkaBus(BOOTSTRAP, client_id=namespace)
    d = KafkaDirectory(BOOTSTRAP, namespace=namespace)
    try:
        await bus.ensure_topics([d.topic])
        await d.advertise(Advert(node=f"{namespace}-agent", kind="agent",
                                 description="live-lane probe",
                                 capabilities={}, inbox="x", worker="w1"))
        for _ in range(50):
            view = await d.view()
            if view.get(f"{namespace}-agent"):
                break
            await asyncio.sleep(0.2)
        assert view.get(f"{namespace}-agent") is not None
    finally:
        await d.close()
        await bus.close()


async def test_agent_turn_over_kafka(namespace):
    """A full Client.execute -> Worker(Agent) turn with Kafka as the bus."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.kafka import KafkaBus, KafkaDirectory
    from calfkit_amd.adapters.memory import InProcessThreadStore
    from calfkit_amd.testing import FakeProvider

    bus_w = KafkaBus(BOOTSTRAP, client_id=f"{namespace}-w")
    dir_w = KafkaDirectory(BOOTSTRAP, namespace=namespace,
                           client_id=f"{namespace}-wd")
    agent = Agent(name=f"{namespace}-echoer",
                  model_client=FakeProvider(["pong"]))
    try:
        async with Worker(agent, bus=bus_w, store=InProcessThreadStore(),
                          directory=dir_w):
            bus_c = KafkaBus(BOOTSTRAP, client_id=f"{namespace}-c")
            dir_c = KafkaDirectory(BOOTSTRAP, namespace=namespace,
                                   client_id=f"{namespace}-cd")
            try:
                client = await Client.connect(bus_c, dir_c)
                result = await client.execute(f"{namespace}-echoer", "ping",
                                              timeout=30)
                assert result.text == "pong"
            finally:
                await dir_c.close()
                await bus_c.close()
    finally:
        await dir_w.close()
        await bus_w.close()
