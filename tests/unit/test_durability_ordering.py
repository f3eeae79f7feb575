"""The crash-consistency contract: a hop's thread row is PERSISTED before
any effect leaves the node (persist-before-publish), so a crash between
the two re-emits rather than loses."""

import asyncio

from calfkit_amd import Agent, Client, Worker, agent_tool
from calfkit_amd.adapters.memory import (
    InMemoryBus, InMemoryDirectory, InProcessThreadStore)
from calfkit_amd.testing import FakeProvider, call_tool


@agent_tool
def probe(x: str) -> str:
    """A probe tool."""
    return f"probed {x}"


class OrderSpy:
    """Wraps store+bus, recording the interleaving of appends and publishes."""

    def __init__(self, store, bus):
        self.events = []
        self._store = store
        self._bus = bus

    def wrap(self):
        store, bus, ev = self._store, self._bus, self.events
        orig_append, orig_publish = store.append, bus.publish

        async def append(node, thread_id, msgs, **kw):
            r = await orig_append(node, thread_id, msgs, **kw)
            ev.append(("persist", node, thread_id))
            return r

        async def publish(topic, key, value, headers=()):
            if ".inbox." in topic or ".reply." in topic:  # effects only
                ev.append(("publish", topic, key))
            return await orig_publish(topic, key, value, headers)

        store.append = append
        bus.publish = publish


async def test_persist_happens_before_every_effect():
    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    spy = OrderSpy(store, bus)
    provider = FakeProvider([call_tool("probe", x="a"), "done"])
    agent = Agent(name="durable2", tools=[probe], model_client=provider)
    async with Worker(agent, probe, bus=bus, store=store, directory=directory):
        spy.wrap()
        client = await Client.connect(bus, directory)
        r = await client.execute("durable2", "go", timeout=10)
        assert r.text == "done"

    # for the agent node: every publish of a tool call / reply must come
    # AFTER a persist of the same thread at that hop
    last_persist_idx = {}
    for i, e in enumerate(spy.events):
        if e[0] == "persist" and e[1] == "durable2":
            last_persist_idx[e[2]] = i
    tool_call_pubs = [i for i, e in enumerate(spy.events)
                      if e[0] == "publish" and "inbox.probe" in e[1]]
    assert tool_call_pubs, spy.events
    for i in tool_call_pubs:
        thread = spy.events[i][2]
        assert thread in last_persist_idx
        persists_before = [j for j, e in enumerate(spy.events[:i])
                           if e[0] == "persist" and e[1] == "durable2"]
        assert persists_before, \
            f"tool call published before any persist: {spy.events[:i+1]}"


async def test_pending_frame_survives_in_the_store():
    """Mid-frame state is durable: a fresh carriage sees the open batch."""
    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    slow_gate = asyncio.Event()

    @agent_tool
    async def slow(x: str) -> str:
        """Slow tool."""
        await slow_gate.wait()
        return "finally"

    provider = FakeProvider([call_tool("slow", x="1"), "after"])
    agent = Agent(name="pending_agent", tools=[slow], model_client=provider)
    async with Worker(agent, slow, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        task = asyncio.create_task(
            client.execute("pending_agent", "go", thread_id="TP", timeout=10))
        for _ in range(100):
            rec = await store.load("pending_agent", "TP")
            if rec is not None and rec.pending:
                break
            await asyncio.sleep(0.01)
        assert rec is not None and rec.pending, "pending frame never persisted"
        assert rec.frame_id
        from calfkit_amd.ports import ThreadStatus

        assert rec.status == ThreadStatus.PENDING
        slow_gate.set()
        r = await task
        assert r.text == "after"
        rec = await store.load("pending_agent", "TP")
        assert not rec.pending and rec.status == ThreadStatus.OPEN
