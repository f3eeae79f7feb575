"""End-to-end mesh tests on the in-memory adapters (the default lane).

Covers: client execute round-trip, tool fan-out via the wire, parallel
tool calls, tool error surfacing, handoff, peer messaging, structured
output, firehose streaming.
"""

import asyncio

import pytest

from calfkit_amd import Agent, Client, Worker, agent_tool, ToolError
from calfkit_amd.adapters.memory import InMemoryBus, InMemoryDirectory, InProcessThreadStore
from calfkit_amd.definitions.selections import Handoff, Messaging
from calfkit_amd.testing import EchoProvider, FakeProvider, FunctionModel, call_tool, reply


def mesh():
    return InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()


@agent_tool
def get_weather(city: str) -> str:
    """Get the current weather for a city."""
    return f"sunny in {city}"


@agent_tool
def fail_tool(x: str) -> str:
    """Always fails."""
    raise ToolError("flaky backend, try later")


async def test_execute_round_trip():
    bus, store, directory = mesh()
    agent = Agent(name="echoer", model_client=EchoProvider())
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("echoer", "hello mesh", timeout=5)
        assert result.text == "echo: hello mesh"


async def test_tool_call_over_the_wire():
    bus, store, directory = mesh()
    provider = FakeProvider([
        call_tool("get_weather", city="Lisbon"),
        "the weather is known",
    ])
    agent = Agent(name="weather_agent", tools=[get_weather], model_client=provider)
    async with Worker(agent, get_weather, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("weather_agent", "weather in Lisbon?", timeout=5)
        assert result.text == "the weather is known"
    # the tool's return reached the model on the next step
    last_request = provider.requests[-1]
    tool_msgs = [m for m in last_request.messages if m.role == "tool"]
    assert any("sunny in Lisbon" in str(p.content) for m in tool_msgs for p in m.parts)


async def test_parallel_tool_calls_one_frame():
    from calfkit_amd.messages import AssistantMessage, ToolCallPart

    bus, store, directory = mesh()
    batch = AssistantMessage((
        ToolCallPart(call_id="c1", tool_name="get_weather", arguments={"city": "Porto"}),
        ToolCallPart(call_id="c2", tool_name="get_weather", arguments={"city": "Faro"}),
    ))
    provider = FakeProvider([batch, "both known"])
    agent = Agent(name="multi", tools=[get_weather], model_client=provider)
    async with Worker(agent, get_weather, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("multi", "Porto and Faro?", timeout=5)
        assert result.text == "both known"
    tool_msgs = [m for m in provider.requests[-1].messages if m.role == "tool"]
    contents = [str(p.content) for m in tool_msgs for p in m.parts]
    assert any("Porto" in c for c in contents) and any("Faro" in c for c in contents)


async def test_tool_error_surfaces_no_hang():
    bus, store, directory = mesh()
    provider = FakeProvider([
        call_tool("fail_tool", x="boom"),
        "recovered gracefully",
    ])
    agent = Agent(name="recoverer", tools=[fail_tool], model_client=provider)
    async with Worker(agent, fail_tool, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("recoverer", "do it", timeout=5)
        assert result.text == "recovered gracefully"
    tool_msgs = [m for m in provider.requests[-1].messages if m.role == "tool"]
    assert any(getattr(p, "is_error", False) for m in tool_msgs for p in m.parts)


async def test_handoff_between_agents():
    bus, store, directory = mesh()
    front = Agent(name="front", model_client=FakeProvider([
        call_tool("handoff", to="specialist", reason="needs expertise"),
    ]), peers=[Handoff(names=("specialist",))])
    specialist = Agent(name="specialist", model_client=FakeProvider([
        "specialist answer",
    ]))
    async with Worker(front, specialist, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("front", "hard question", timeout=5)
        assert result.text == "specialist answer"


async def test_peer_messaging_with_reply():
    bus, store, directory = mesh()
    asker = Agent(name="asker", model_client=FakeProvider([
        call_tool("send_message", to="oracle", message="what is the answer?"),
        "the oracle says 42",
    ]), peers=[Messaging(names=("oracle",))])
    oracle = Agent(name="oracle", model_client=FakeProvider(["42"]))
    async with Worker(asker, oracle, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("asker", "ask the oracle", timeout=5)
        assert result.text == "the oracle says 42"


async def test_structured_output_with_correction():
    from dataclasses import dataclass

    @dataclass
    class Weather:
        city: str
        temp_c: int

    bus, store, directory = mesh()
    provider = FakeProvider([
        "not json at all",
        '{"city": "Lisbon", "temp_c": 29}',
    ])
    agent = Agent(name="typed", model_client=provider, output_type=Weather)
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("typed", "weather?", timeout=5,
                                      output_type=Weather)
        assert result.output == Weather(city="Lisbon", temp_c=29)
    assert len(provider.requests) == 2  # the correction re-step happened


async def test_firehose_streams_step_events():
    bus, store, directory = mesh()
    agent = Agent(name="streamy", model_client=EchoProvider())
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        thread_id = "t-stream"
        events = []

        async def collect():
            async for chunk in client.firehose(thread_id):
                events.append(chunk)
                if chunk.event == "step_finished":
                    break

        task = asyncio.create_task(collect())
        await asyncio.sleep(0)
        await client.execute("streamy", "hi", thread_id=thread_id, timeout=5)
        await asyncio.wait_for(task, 5)
        kinds = [c.event for c in events]
        assert "step_started" in kinds and "step_finished" in kinds
        assert any(c.event == "text_delta" for c in events)


async def test_mesh_view_from_client():
    bus, store, directory = mesh()
    agent = Agent(name="visible", description="a visible agent",
                  model_client=EchoProvider())
    async with Worker(agent, get_weather, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        view = await client.mesh()
        assert {a.node for a in view.agents()} == {"visible"}
        assert {t.node for t in view.tools()} == {"get_weather"}
        assert view.get("get_weather").capabilities["parameters"]["properties"]["city"]


async def test_multi_turn_same_thread_keeps_history():
    bus, store, directory = mesh()
    provider = FunctionModel(lambda req: f"seen {sum(1 for m in req.messages if m.role == 'user')} user messages")
    agent = Agent(name="memory", model_client=provider)
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        r1 = await client.execute("memory", "one", thread_id="t1", timeout=5)
        r2 = await client.execute("memory", "two", thread_id="t1", timeout=5)
        assert r1.text == "seen 1 user messages"
        assert r2.text == "seen 2 user messages"


async def test_runtime_model_settings_override():
    from calfkit_amd.schemas import ModelSettings

    bus, store, directory = mesh()
    provider = FakeProvider(["ok"])
    agent = Agent(name="tunable", model_client=provider,
                  model_settings=ModelSettings(max_tokens=100))
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        await client.execute("tunable", "hi", timeout=5,
                             model_settings=ModelSettings(max_tokens=7,
                                                          temperature=0.5))
    req = provider.requests[0]
    assert req.settings.max_tokens == 7
    assert req.settings.temperature == 0.5


async def test_invalid_tool_args_rejected_before_the_wire():
    bus, store, directory = mesh()
    from calfkit_amd.messages import AssistantMessage, ToolCallPart

    provider = FakeProvider([
        AssistantMessage((ToolCallPart("c1", "get_weather",
                                       {"town": "Lisbon"}),)),  # wrong key
        "fixed it",
    ])
    agent = Agent(name="validator", tools=[get_weather], model_client=provider)
    async with Worker(agent, get_weather, bus=bus, store=store,
                      directory=directory):
        client = await Client.connect(bus, directory)
        r = await client.execute("validator", "weather", timeout=5)
        assert r.text == "fixed it"
    tool_msgs = [m for m in provider.requests[-1].messages if m.role == "tool"]
    assert "invalid arguments" in str(tool_msgs[0].parts[0].content)


async def test_handoff_chain_three_agents():
    """A -> B -> C handoff chain: the thread follows to the terminal agent
    and the caller gets C's answer (reference: handoff follows across
    turns, CHANGELOG.md:47-area behavior)."""
    bus, store, directory = mesh()
    a = Agent(name="triage", model_client=FakeProvider([
        call_tool("handoff", to="router", reason="route it"),
    ]), peers=[Handoff(names=("router",))])
    b = Agent(name="router", model_client=FakeProvider([
        call_tool("handoff", to="expert", reason="needs the expert"),
    ]), peers=[Handoff(names=("expert",))])
    c = Agent(name="expert", model_client=FakeProvider(["expert verdict"]))
    async with Worker(a, b, c, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("triage", "deep question", timeout=5)
        assert result.text == "expert verdict"


async def test_handoff_followup_turn_reaches_new_owner():
    """After a handoff, the NEXT turn on the same thread goes to the new
    owner directly (the thread moved, not just the single reply)."""
    bus, store, directory = mesh()
    front = Agent(name="desk", model_client=FakeProvider([
        call_tool("handoff", to="owner", reason="yours now"),
    ]), peers=[Handoff(names=("owner",))])
    owner = Agent(name="owner", model_client=FakeProvider([
        "taking over", "still me on turn 2",
    ]))
    async with Worker(front, owner, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        r1 = await client.execute("desk", "start", timeout=5)
        assert r1.text == "taking over"
        target = r1.agent if getattr(r1, "agent", None) else "owner"
        r2 = await client.execute(target, "follow-up",
                                  thread_id=r1.thread_id, timeout=5)
        assert r2.text == "still me on turn 2"
