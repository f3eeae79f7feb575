"""Integration: the typed fault rail end-to-end + step-ledger records."""

import json

import pytest

from calfkit_amd import Agent, Client, Worker, agent_tool
from calfkit_amd.adapters.memory import (
    InMemoryBus, InMemoryDirectory, InProcessThreadStore)
from calfkit_amd.errors import NodeFaultError
from calfkit_amd.testing import FakeProvider, call_tool


def mesh():
    return InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()


@agent_tool
def crasher(x: str) -> str:
    """Crashes hard (not a ToolError)."""
    raise RuntimeError(f"hardware on fire: {x}")


async def test_tool_crash_rides_fault_rail_and_agent_recovers():
    bus, store, directory = mesh()
    provider = FakeProvider([call_tool("crasher", x="now"), "survived"])
    agent = Agent(name="survivor", tools=[crasher], model_client=provider)
    async with Worker(agent, crasher, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        r = await client.execute("survivor", "go", timeout=10)
        assert r.text == "survived"
    # the fault was harvested structurally and surfaced via on_tool_error
    tool_msgs = [m for m in provider.requests[-1].messages if m.role == "tool"]
    assert "hardware on fire" in str(tool_msgs[0].parts[0].content)


async def test_provider_crash_faults_to_client():
    bus, store, directory = mesh()

    class ExplodingProvider:
        async def complete(self, request, sink):
            raise RuntimeError("engine OOM")

    agent = Agent(name="doomed", model_client=ExplodingProvider())
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        with pytest.raises(NodeFaultError) as exc:
            await client.execute("doomed", "hello", timeout=10)
        assert "engine OOM" in str(exc.value)
        assert exc.value.report.kind == "agent"
        assert exc.value.report.detail.exc_type == "RuntimeError"


async def test_step_ledger_records_flow_with_engine_facts():
    from calfkit_amd.messages import AssistantMessage, ProviderFacts, TextPart, Usage

    bus, store, directory = mesh()
    provider = FakeProvider([AssistantMessage(
        (TextPart("ok"),),
        facts=ProviderFacts(model_name="m", usage=Usage(11, 7),
                            prefill_ms=3.25, decode_ms=17.5))])
    agent = Agent(name="ledgered", model_client=provider)
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        await client.execute("ledgered", "hi", timeout=10)
        await bus.quiesce()
        records = [json.loads(r.value) for r in bus.records("calf.steps")]
    agent_steps = [r for r in records if r["node"] == "ledgered"]
    assert agent_steps, records
    rec = agent_steps[-1]
    assert rec["kind"] == "agent"
    assert rec["outcome"] == "finished"
    assert rec["prefill_ms"] == 3.25 and rec["decode_ms"] == 17.5
    assert rec["prompt_tokens"] == 11 and rec["completion_tokens"] == 7
    assert rec["hop_ms"] > 0
