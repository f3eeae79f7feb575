"""Unit: client verbs beyond execute — send, start, reply_to routing."""

import asyncio

from calfkit_amd import Agent, Client, Worker
from calfkit_amd.adapters.memory import (
    InMemoryBus, InMemoryDirectory, InProcessThreadStore)
from calfkit_amd.testing import EchoProvider, FakeProvider


async def test_send_is_fire_and_forget():
    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    provider = FakeProvider(["noted"])
    agent = Agent(name="sink", model_client=provider)
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        tid = await client.send("sink", "just so you know")
        assert tid
        await bus.quiesce()
        assert provider.requests  # the agent stepped, nobody awaited


async def test_start_routes_replies_to_private_inbox():
    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    agent = Agent(name="starter", model_client=EchoProvider())
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        tid = await client.start("starter", "begin")
        await bus.quiesce()
        # the final landed on the client's private reply topic
        recs = bus.records(client.reply_inbox)
        assert any(r.key == tid for r in recs)


async def test_send_with_reply_to_third_party():
    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    agent = Agent(name="relay", model_client=EchoProvider())
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        tid = await client.send("relay", "answer elsewhere",
                                reply_to="observer-7")
        await bus.quiesce()
        recs = bus.records("calf.reply.observer-7")
        assert any(r.key == tid for r in recs)
