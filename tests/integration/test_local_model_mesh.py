"""End-to-end: agent mesh with the in-process LocalModelClient (TINY, CPU).

This is BASELINE config 2's shape with the tiny model: a single Agent node
whose provider is the CDNA4 engine (torch-reference path on CPU), full wire
round-trip through the in-memory mesh.
"""

import pytest

from calfkit_amd import Agent, Client, Worker
from calfkit_amd.adapters.memory import (
    InMemoryBus,
    InMemoryDirectory,
    InProcessThreadStore,
)
from calfkit_amd.engine.local import LocalModelClient
from calfkit_amd.models.config import TINY
from calfkit_amd.schemas import ModelSettings


async def test_agent_turn_on_local_engine():
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    provider = LocalModelClient(TINY, device="cpu", max_batch=4,
                                max_model_len=256, use_graphs=False)
    agent = Agent(name="localist", model_client=provider,
                  model_settings=ModelSettings(max_tokens=6, temperature=0.0))
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("localist", "hello from the mesh",
                                      timeout=300)
        # random-init weights -> synthetic token text, non-empty
        assert result.text.strip()
        assert result.message.facts is not None
        assert result.message.facts.usage.completion_tokens == 6
        assert result.message.facts.decode_ms > 0
        # a second turn reuses the shared engine and the thread history
        r2 = await client.execute("localist", "again", thread_id=result.thread_id,
                                  timeout=300)
        assert r2.text.strip()


async def test_stop_sequence_truncates_and_sets_finish():
    """Client-side stop sequences: the same deterministic (greedy) turn
    with a stop string drawn from its own output is truncated at the stop
    and reports finish_reason 'stop'."""
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    provider = LocalModelClient(TINY, device="cpu", max_batch=4,
                                max_model_len=256, use_graphs=False)
    agent = Agent(name="stopper", model_client=provider,
                  model_settings=ModelSettings(max_tokens=8, temperature=0.0))
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        full = await client.execute("stopper", "say things", timeout=300)
        words = full.text.split()
        assert len(words) >= 3
        stop_word = words[2]
        trunc = await client.execute(
            "stopper", "say things", timeout=300,
            model_settings=ModelSettings(max_tokens=8, temperature=0.0,
                                         stop=[stop_word]))
        assert stop_word not in trunc.text
        assert trunc.text.strip() == " ".join(words[:2])
        assert trunc.message.facts.finish_reason == "stop"
