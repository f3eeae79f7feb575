"""Integration: the bundled mesh broker carrying a real multi-worker mesh.

The real-broker lane of the reference's test taxonomy
(/root/reference/CONTRIBUTING.md:108-135): produce/consume round-trips,
capability discovery over the wire, topic provisioning, group semantics.
"""

import asyncio

import pytest

from calfkit_amd import Agent, Client, Worker, agent_tool
from calfkit_amd.adapters.memory import InProcessThreadStore
from calfkit_amd.adapters.mesh import MeshBroker, connect_mesh
from calfkit_amd.testing import FakeProvider, call_tool


@agent_tool
def lookup(key: str) -> str:
    """Look up a key."""
    return f"value-of-{key}"


async def test_round_trip_over_broker():
    broker = await MeshBroker().start()
    try:
        bus_w, dir_w = await connect_mesh(broker.address)
        provider = FakeProvider([call_tool("lookup", key="k1"), "found it"])
        agent = Agent(name="seeker", tools=[lookup], model_client=provider)
        async with Worker(agent, lookup, bus=bus_w, store=InProcessThreadStore(),
                          directory=dir_w):
            bus_c, dir_c = await connect_mesh(broker.address)
            client = await Client.connect(bus_c, dir_c)
            result = await client.execute("seeker", "find k1", timeout=10)
            assert result.text == "found it"
            view = await client.mesh()
            assert {a.node for a in view.tools()} == {"lookup"}
    finally:
        await broker.stop()


async def test_two_workers_one_broker():
    """Agents hosted by SEPARATE workers discover and message each other."""
    broker = await MeshBroker().start()
    try:
        bus1, dir1 = await connect_mesh(broker.address)
        bus2, dir2 = await connect_mesh(broker.address)
        from calfkit_amd.definitions.selections import Messaging

        asker = Agent(name="asker2", model_client=FakeProvider([
            call_tool("send_message", to="oracle2", message="answer?"),
            "oracle said 42",
        ]), peers=[Messaging(discover=True)])
        oracle = Agent(name="oracle2", model_client=FakeProvider(["42"]))
        async with Worker(asker, bus=bus1, store=InProcessThreadStore(),
                          directory=dir1, name="w1"):
            async with Worker(oracle, bus=bus2, store=InProcessThreadStore(),
                              directory=dir2, name="w2"):
                bus_c, dir_c = await connect_mesh(broker.address)
                client = await Client.connect(bus_c, dir_c)
                result = await client.execute("asker2", "go ask", timeout=10)
                assert result.text == "oracle said 42"
                view = await client.mesh()
                workers = {a.worker for a in view.agents()}
                assert workers == {"w1", "w2"}
    finally:
        await broker.stop()


async def test_consumer_group_replicas_share_by_key():
    """Two replicas of one node split thread keys (scaling semantics)."""
    broker = await MeshBroker().start()
    try:
        seen: dict[str, list[str]] = {"a": [], "b": []}

        async def handler_for(name):
            async def h(raw):
                seen[name].append(raw.key)
            return h

        bus_a, _ = await connect_mesh(broker.address)
        bus_b, _ = await connect_mesh(broker.address)
        await bus_a.subscribe("calf.inbox.x", await handler_for("a"), group="x")
        await bus_b.subscribe("calf.inbox.x", await handler_for("b"), group="x")
        bus_p, _ = await connect_mesh(broker.address)
        for i in range(10):
            key = f"t{i % 4}"
            await bus_p.publish("calf.inbox.x", key, b"{}")
        await asyncio.sleep(0.3)
        total = len(seen["a"]) + len(seen["b"])
        assert total == 10
        # sticky per key: each key only ever lands on one member
        overlap = set(seen["a"]) & set(seen["b"])
        assert not overlap, (seen, overlap)
    finally:
        await broker.stop()


async def test_dead_worker_adverts_leave_the_directory():
    """Liveness: a worker connection dying withdraws its adverts."""
    broker = await MeshBroker().start()
    try:
        bus_w, dir_w = await connect_mesh(broker.address)
        agent = Agent(name="mortal", model_client=FakeProvider(["hi"]))
        worker = Worker(agent, bus=bus_w, store=InProcessThreadStore(),
                        directory=dir_w)
        await worker.start()
        bus_c, dir_c = await connect_mesh(broker.address)
        client = await Client.connect(bus_c, dir_c)
        assert (await client.mesh()).get("mortal") is not None
        # kill the worker's CONNECTION without a clean withdraw
        await bus_w.close()
        await asyncio.sleep(0.2)
        assert (await client.mesh()).get("mortal") is None
    finally:
        await broker.stop()


async def test_client_connect_bootstrap_string():
    """The README path: Client.connect(bootstrap='host:port') dials the
    broker without pre-built adapters."""
    broker = await MeshBroker().start()
    try:
        bus_w, dir_w = await connect_mesh(broker.address)
        agent = Agent(name="dialee", model_client=FakeProvider(["hello"]))
        async with Worker(agent, bus=bus_w, store=InProcessThreadStore(),
                          directory=dir_w):
            client = await Client.connect(bootstrap=broker.address)
            result = await client.execute("dialee", "hi", timeout=10)
            assert result.text == "hello"
    finally:
        await broker.stop()
