"""Unit: durable SQLite thread store — same contract as the memory store,
plus real durability across store instances (crash/resume premise)."""

import os

from calfkit_amd.adapters.sqlite import SqliteThreadStore
from calfkit_amd.messages import TextPart, UserMessage
from calfkit_amd.ports import AppendDisposition, ThreadStatus


async def test_append_load_round_trip(tmp_path):
    db = str(tmp_path / "threads.db")
    store = SqliteThreadStore(db)
    d = await store.append("n", "t1", [UserMessage((TextPart("hello"),))],
                           expected_revision=0, status=ThreadStatus.PENDING,
                           pending={"c1": {"done": False}}, frame_id="f1",
                           metadata={"hop": 2})
    assert d == AppendDisposition.APPENDED
    rec = await store.load("n", "t1")
    assert rec.revision == 1
    assert rec.messages[0].parts[0].text == "hello"
    assert rec.status == ThreadStatus.PENDING
    assert rec.pending == {"c1": {"done": False}}
    assert rec.frame_id == "f1" and rec.metadata == {"hop": 2}


async def test_conflict_on_stale_revision(tmp_path):
    store = SqliteThreadStore(str(tmp_path / "x.db"))
    await store.append("n", "t", [UserMessage((TextPart("a"),))])
    d = await store.append("n", "t", [UserMessage((TextPart("b"),))],
                           expected_revision=0)
    assert d == AppendDisposition.CONFLICT
    rec = await store.load("n", "t")
    assert len(rec.messages) == 1


async def test_durability_across_instances(tmp_path):
    db = str(tmp_path / "durable.db")
    s1 = SqliteThreadStore(db)
    await s1.append("n", "t", [UserMessage((TextPart("persisted"),))])
    await s1.close()
    s2 = SqliteThreadStore(db)  # a different replica resumes the thread
    rec = await s2.load("n", "t")
    assert rec is not None and rec.messages[0].parts[0].text == "persisted"


async def test_mesh_survives_worker_restart(tmp_path):
    """The durable-agent premise: a new Worker resumes mid-thread state."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import InMemoryBus, InMemoryDirectory
    from calfkit_amd.testing import FunctionModel

    db = str(tmp_path / "mesh.db")
    bus, directory = InMemoryBus(), InMemoryDirectory()
    provider = FunctionModel(
        lambda req: f"{sum(1 for m in req.messages if m.role == 'user')} user msgs")

    store1 = SqliteThreadStore(db)
    agent = Agent(name="durable", model_client=provider)
    async with Worker(agent, bus=bus, store=store1, directory=directory):
        client = await Client.connect(bus, directory)
        r1 = await client.execute("durable", "first", thread_id="T", timeout=5)
        assert r1.text == "1 user msgs"
    await store1.close()

    # a fresh worker (fresh store instance, same file) continues the thread
    bus2, directory2 = InMemoryBus(), InMemoryDirectory()
    store2 = SqliteThreadStore(db)
    agent2 = Agent(name="durable", model_client=provider)
    async with Worker(agent2, bus=bus2, store=store2, directory=directory2):
        client2 = await Client.connect(bus2, directory2)
        r2 = await client2.execute("durable", "second", thread_id="T", timeout=5)
        assert r2.text == "2 user msgs"
