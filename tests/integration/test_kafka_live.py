"""Live-broker lane: the bus/directory adapters against a REAL broker.

Parity with the reference's ``-m kafka`` lane
(/root/reference/CONTRIBUTING.md:100-135), with the offline twist from
VERDICT r01 #6: the same tests run against the bundled mesh broker —
which speaks the SAME envelope bytes (tests/unit/test_wire_golden.py
freezes them) — so the live-lane logic executes in every run even though
aiokafka + a Kafka broker only exist when ``CALF_TEST_KAFKA_BOOTSTRAP``
points at one (then the kafka lane stops skipping).

Every topic/group derives from a per-run namespace so runs sharing a
broker don't collide.
"""

import asyncio
import importlib.util
import os
import uuid

import pytest

BOOTSTRAP = os.environ.get("CALF_TEST_KAFKA_BOOTSTRAP", "")
HAVE_KAFKA = bool(BOOTSTRAP) and importlib.util.find_spec("aiokafka") is not None

pytestmark = pytest.mark.kafka


class _Lane:
    """One live transport: factories + teardown for bus/directory pairs."""

    def __init__(self, name):
        self.name = name
        self.broker = None
        self._closers = []

    async def start(self):
        if self.name == "mesh":
            from calfkit_amd.adapters.mesh import MeshBroker

            self.broker = await MeshBroker().start()
        return self

    async def bus(self, client_id: str):
        if self.name == "kafka":
            from calfkit_amd.adapters.kafka import KafkaBus

            b = KafkaBus(BOOTSTRAP, client_id=client_id)
        else:
            from calfkit_amd.adapters.mesh import connect_mesh

            b, _ = await connect_mesh(self.broker.address)
        self._closers.append(b.close)
        return b

    async def pair(self, namespace: str, client_id: str):
        """(bus, directory) sharing a connection/bootstrap."""
        if self.name == "kafka":
            from calfkit_amd.adapters.kafka import KafkaBus, KafkaDirectory

            b = KafkaBus(BOOTSTRAP, client_id=client_id)
            d = KafkaDirectory(BOOTSTRAP, namespace=namespace,
                               client_id=f"{client_id}-d")
            self._closers += [d.close, b.close]
            return b, d
        from calfkit_amd.adapters.mesh import connect_mesh

        b, d = await connect_mesh(self.broker.address)
        self._closers.append(b.close)
        return b, d

    async def stop(self):
        for c in reversed(self._closers):
            try:
                await c()
            except Exception:
                pass
        if self.broker is not None:
            await self.broker.stop()


@pytest.fixture(params=["mesh", "kafka"])
def lane_name(request):
    # sync fixture (the conftest async runner doesn't support async
    # fixtures); tests start/stop the lane inside their own loop
    if request.param == "kafka" and not HAVE_KAFKA:
        pytest.skip("no CALF_TEST_KAFKA_BOOTSTRAP broker / aiokafka SDK")
    return request.param


@pytest.fixture()
def namespace() -> str:
    return f"calftest-{uuid.uuid4().hex[:8]}"


async def test_bus_round_trip(lane_name, namespace):
    live = await _Lane(lane_name).start()
    bus = await live.bus(namespace)
    topic = f"{namespace}.input"
    await bus.ensure_topics([topic])
    got: asyncio.Queue = asyncio.Queue()

    async def handler(raw):
        got.put_nowait(raw)

    await bus.subscribe(topic, handler, group=f"{namespace}.g")
    await asyncio.sleep(1.0 if live.name == "kafka" else 0.05)
    await bus.publish(topic, "k1", b'{"x":1}',
                      headers=(("x-calf-kind", b"probe"),))
    try:
        raw = await asyncio.wait_for(got.get(), 15)
        assert raw.key == "k1" and raw.value == b'{"x":1}'
        assert dict(raw.headers).get("x-calf-kind") == b"probe"
    finally:
        await live.stop()


async def test_directory_advertise_and_view(lane_name, namespace):
    from calfkit_amd.ports import Advert

    live = await _Lane(lane_name).start()
    _, d = await live.pair(namespace, namespace)
    await d.advertise(Advert(node=f"{namespace}-agent", kind="agent",
                             description="live-lane probe",
                             capabilities={}, inbox="x", worker="w1"))
    try:
        view = None
        for _ in range(50):
            view = await d.view()
            if view.get(f"{namespace}-agent"):
                break
            await asyncio.sleep(0.2)
        assert view.get(f"{namespace}-agent") is not None
    finally:
        await live.stop()


async def test_agent_turn_over_live_bus(lane_name, namespace):
    """A full Client.execute -> Worker(Agent) turn over the live bus."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import InProcessThreadStore
    from calfkit_amd.testing import FakeProvider

    live = await _Lane(lane_name).start()
    try:
        bus_w, dir_w = await live.pair(namespace, f"{namespace}-w")
        agent = Agent(name=f"{namespace}-echoer",
                      model_client=FakeProvider(["pong"]))
        async with Worker(agent, bus=bus_w, store=InProcessThreadStore(),
                          directory=dir_w):
            bus_c, dir_c = await live.pair(namespace, f"{namespace}-c")
            client = await Client.connect(bus_c, dir_c)
            result = await client.execute(f"{namespace}-echoer", "ping",
                                          timeout=30)
            assert result.text == "pong"
    finally:
        await live.stop()
