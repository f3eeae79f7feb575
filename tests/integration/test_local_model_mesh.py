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


async def _tool_call_round_trip(device: str, use_graphs: bool):
    """VERDICT r01 #3 end-to-end: the ENGINE's sampled output becomes a
    ToolCallPart that fans out over the mesh to a real tool node, whose
    reply continues the turn to a final text answer — zero
    ToolCallDirector, zero FakeProvider."""
    from calfkit_amd import agent_tool

    calls = []

    @agent_tool
    async def get_weather(city: str, days: int) -> str:
        """Weather forecast for a city."""
        calls.append((city, days))
        return f"sunny in {city} for {days} days"

    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    provider = LocalModelClient(TINY, device=device, max_batch=4,
                                max_model_len=256, use_graphs=use_graphs)
    agent = Agent(name="forecaster", model_client=provider,
                  tools=[get_weather],
                  model_settings=ModelSettings(max_tokens=24,
                                               temperature=0.7, seed=11))
    async with Worker(agent, get_weather, bus=bus, store=store,
                      directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("forecaster", "what is the weather",
                                      timeout=300)
        # the tool really ran, with schema-typed args sampled on the engine
        assert len(calls) == 1
        city, days = calls[0]
        assert isinstance(city, str) and city.startswith("t")
        assert isinstance(days, int)
        # and the agent produced a closing answer after the tool reply
        assert result.text.strip()


async def test_engine_tool_call_round_trip_cpu():
    await _tool_call_round_trip("cpu", False)


@pytest.mark.gpu
async def test_engine_tool_call_round_trip_gpu():
    await _tool_call_round_trip("cuda", True)


async def test_structured_output_from_the_engine():
    """Engine-driven structured output (final_result union): the policy's
    end-shape check passes on JSON materialized from the sampled stream,
    and the client deserializes the typed result."""
    from dataclasses import dataclass

    from calfkit_amd import Agent, Client, Worker

    @dataclass
    class Weather:
        city: str
        temp_c: int

    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    provider = LocalModelClient(TINY, device="cpu", max_batch=4,
                                max_model_len=256, use_graphs=False)
    agent = Agent(name="typed-engine", model_client=provider,
                  output_type=Weather,
                  model_settings=ModelSettings(max_tokens=12,
                                               temperature=0.6, seed=9))
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("typed-engine", "weather please",
                                      timeout=300, output_type=Weather)
        assert isinstance(result.output, Weather)
        assert isinstance(result.output.temp_c, int)
        assert result.output.city.startswith("t")


async def _handoff_round(device: str, use_graphs: bool):
    """The GPU/engine grammar can drive the HANDOFF capability: 'handoff'
    is advertised like any tool, the sampled token picks the peer from the
    schema's enum (so it always names a REAL agent), and the conversation
    transfers across nodes — no FakeProvider anywhere on the front agent."""
    from calfkit_amd.definitions.selections import Handoff
    from calfkit_amd.testing import FakeProvider

    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    front = Agent(name="front",
                  model_client=LocalModelClient(TINY, device=device,
                                                max_batch=4,
                                                max_model_len=256,
                                                use_graphs=use_graphs),
                  peers=[Handoff(names=("specialist",))],
                  model_settings=ModelSettings(max_tokens=16,
                                               temperature=0.7, seed=21))
    specialist = Agent(name="specialist",
                       model_client=FakeProvider(["specialist answer"]))
    async with Worker(front, specialist, bus=bus, store=store,
                      directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("front", "hard question", timeout=300)
        # the engine sampled the handoff; the specialist answered the thread
        assert result.text == "specialist answer"


async def test_engine_sampled_handoff_routes_to_a_real_peer():
    await _handoff_round("cpu", False)


@pytest.mark.gpu
async def test_engine_sampled_handoff_routes_gpu():
    await _handoff_round("cuda", True)


async def test_tool_mode_required_forces_an_immediate_engine_call():
    """ToolMode(required=True) fires the grammar trigger at position 1 —
    the engine's FIRST decode step opens a tool call (no free-text
    preamble), and ToolMode(choice=...) restricts the sampled name."""
    from calfkit_amd import agent_tool
    from calfkit_amd.definitions.selections import ToolMode, Tools

    calls = []

    @agent_tool
    async def alpha(x: int) -> str:
        """Tool alpha."""
        calls.append(("alpha", x))
        return "a"

    @agent_tool
    async def beta(x: int) -> str:
        """Tool beta."""
        calls.append(("beta", x))
        return "b"

    bus, store, directory = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
    provider = LocalModelClient(TINY, device="cpu", max_batch=4,
                                max_model_len=256, use_graphs=False)
    agent = Agent(name="chooser", model_client=provider,
                  tools=Tools(names=("alpha", "beta"),
                              mode=ToolMode(choice="beta")),
                  model_settings=ModelSettings(max_tokens=24,
                                               temperature=0.7, seed=3))
    async with Worker(agent, alpha, beta, bus=bus, store=store,
                      directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("chooser", "pick a tool", timeout=300)
        # the named choice is the ONLY tool the grammar could sample
        assert calls and all(name == "beta" for name, _ in calls)
        assert result.text.strip()
