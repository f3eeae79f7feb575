"""Integration: MCP toolbox nodes on the mesh + the fire-and-forget event
plane (emits / listens_to)."""

import asyncio
import json

from calfkit_amd import Agent, Client, MCPToolbox, Worker
from calfkit_amd.adapters.memory import (
    InMemoryBus,
    InMemoryDirectory,
    InProcessThreadStore,
)
from calfkit_amd.definitions.selections import Toolboxes
from calfkit_amd.mcp import MCPToolInfo
from calfkit_amd.testing import FakeMCPSession, FakeProvider, call_tool


def mesh():
    return InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()


async def test_mcp_toolbox_capability_over_the_mesh():
    bus, store, directory = mesh()
    session = FakeMCPSession(
        tools=[MCPToolInfo(name="read_file", description="Read a file",
                           input_schema={"type": "object", "properties": {
                               "path": {"type": "string"}}})],
        results={"read_file": lambda args: f"contents of {args['path']}"})
    box = MCPToolbox(name="fsbox", session=session)
    provider = FakeProvider([
        call_tool("fsbox.read_file", path="/etc/motd"),  # namespaced name
        "file read ok",
    ])
    agent = Agent(name="reader", tools=Toolboxes(names=("fsbox",)),
                  model_client=provider)
    async with Worker(agent, box, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        # the toolbox advert carries the (un-namespaced) tool schemas
        view = await client.mesh()
        caps = view.get("fsbox").capabilities
        assert caps["tools"][0]["name"] == "read_file"
        result = await client.execute("reader", "read the motd", timeout=10)
        assert result.text == "file read ok"
    assert session.calls == [("read_file", {"path": "/etc/motd"})]
    # the model saw the namespaced schema
    offered = {t.name for t in provider.requests[0].tools}
    assert "fsbox.read_file" in offered


async def test_toolbox_error_surfaces_as_error_return():
    bus, store, directory = mesh()
    session = FakeMCPSession(tools=[MCPToolInfo(name="boom")])  # no script
    box = MCPToolbox(name="bad", session=session)
    provider = FakeProvider([call_tool("bad.boom"), "handled"])
    agent = Agent(name="handler", tools=Toolboxes(names=("bad",)),
                  model_client=provider)
    async with Worker(agent, box, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        result = await client.execute("handler", "go", timeout=10)
        assert result.text == "handled"
    tool_msgs = [m for m in provider.requests[-1].messages if m.role == "tool"]
    assert tool_msgs and tool_msgs[0].parts[0].is_error


async def test_event_plane_emit_and_listen():
    """Caller-side emit lands on listeners' inbox pipelines as a delivery."""
    bus, store, directory = mesh()
    seen = asyncio.Event()
    provider = FakeProvider(["noted"])
    agent = Agent(name="listener", model_client=provider,
                  listens_to=["metrics"])
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        await client.emit("metrics", {"cpu": 0.93}, thread_id="ev-1")
        await bus.quiesce()
        # the event delivery drove one agent step on the event's thread
        assert provider.requests, "listener never stepped on the event"
        text = " ".join(
            p.text for m in provider.requests[0].messages
            for p in m.parts if hasattr(p, "text"))
        assert "cpu" in text


async def test_agent_ctx_emit_reaches_event_topic():
    from calfkit_amd.context import TurnContext
    from calfkit_amd.hooks import AgentHook

    bus, store, directory = mesh()

    class Emitter(AgentHook):
        async def before_step(self, ctx: TurnContext):
            await ctx.emit("audit", {"who": ctx.agent_name})
            return None

    agent = Agent(name="emitter", model_client=FakeProvider(["done"]),
                  hooks=[Emitter()], emits=["audit"])
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        await client.execute("emitter", "hi", timeout=10)
        records = bus.records("calf.event.audit")
        assert records, "no event published"
        body = json.loads(records[0].value)
        assert "emitter" in json.dumps(body)
