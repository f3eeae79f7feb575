"""Unit: drive_step decisions D1-D7 with fakes at the ports."""

import pytest

from calfkit_amd._capabilities import Dispatch, Transfer
from calfkit_amd._drive import drive_step
from calfkit_amd._engine import Engine
from calfkit_amd._policy import TurnPolicy
from calfkit_amd._surface import build_step_surface
from calfkit_amd.context import TurnContext
from calfkit_amd.definitions.agent import Agent
from calfkit_amd.definitions.selections import Handoff, Messaging
from calfkit_amd.messages import TextPart, UserMessage
from calfkit_amd.ports import Advert, DirectoryView
from calfkit_amd.streaming import CollectingSink
from calfkit_amd.testing import FakeProvider, call_tool, reply


def view():
    return DirectoryView(adverts={
        "hammer": Advert(node="hammer", kind="tool", inbox="calf.inbox.hammer",
                         capabilities={"parameters": {"type": "object",
                                                       "properties": {}}}),
        "peer1": Advert(node="peer1", kind="agent", inbox="calf.inbox.peer1"),
    })


def setup(script, **agent_kw):
    provider = FakeProvider(script)
    agent = Agent(name="driver", model_client=provider, tools=["hammer"],
                  peers=[Messaging(names=("peer1",)), Handoff(names=("peer1",))],
                  **agent_kw)
    surface = build_step_surface(agent, view())
    engine = Engine(agent, provider)
    return engine, surface


def history():
    return (UserMessage((TextPart("go"),)),)


async def drive(engine, surface):
    return await drive_step(engine, surface, TurnPolicy(), history(),
                            CollectingSink(), TurnContext(agent_name="driver",
                                                          thread_id="t"))


async def test_plain_answer_finishes():
    engine, surface = setup(["done"])
    res = await drive(engine, surface)
    assert res.finished and res.final.text == "done"
    assert not res.effects


async def test_mesh_tool_call_pends_one_frame():
    engine, surface = setup([call_tool("hammer", call_id="c1")])
    res = await drive(engine, surface)
    assert not res.finished
    assert res.frame_id
    assert [p.call_id for p in res.pending_tool_calls] == ["c1"]
    assert res.pending_tool_calls[0].target_inbox == "calf.inbox.hammer"


async def test_capability_messaging_dispatch_pends():
    engine, surface = setup([
        call_tool("send_message", call_id="c2", to="peer1", message="hi")])
    res = await drive(engine, surface)
    assert isinstance(res.effects[0], Dispatch)
    assert res.pending_peer_calls[0].target == "peer1"
    assert not res.finished


async def test_fire_and_forget_dispatch_continues_turn():
    engine, surface = setup([
        call_tool("send_message", call_id="c3", to="peer1", message="fyi",
                  await_reply=False),
        "sent it"])
    res = await drive(engine, surface)
    assert res.finished and res.final.text == "sent it"
    assert isinstance(res.effects[0], Dispatch)
    assert not res.effects[0].reply_expected


async def test_handoff_wins_the_step():
    engine, surface = setup([
        call_tool("handoff", call_id="c4", to="peer1", reason="better suited")])
    res = await drive(engine, surface)
    assert res.finished  # nothing pending — the turn moved elsewhere
    transfer = res.effects[0]
    assert isinstance(transfer, Transfer)
    assert transfer.target == "peer1" and transfer.reason == "better suited"


async def test_unknown_tool_synthesizes_error_and_resteps():
    engine, surface = setup([
        call_tool("no_such_tool", call_id="c5"),
        "recovered"])
    res = await drive(engine, surface)
    assert res.finished and res.final.text == "recovered"
    tool_msgs = [m for m in res.new_messages if m.role == "tool"]
    assert tool_msgs and tool_msgs[0].parts[0].is_error


async def test_unknown_peer_rejected_surfaced():
    engine, surface = setup([
        call_tool("send_message", call_id="c6", to="ghost", message="?"),
        "ok then"])
    res = await drive(engine, surface)
    assert res.finished
    tool_msgs = [m for m in res.new_messages if m.role == "tool"]
    assert "ghost" in str(tool_msgs[0].parts[0].content)


def test_handoff_request_is_the_typed_argument_model():
    """HandoffRequest's field descriptions ARE the handoff tool schema
    (reference CHANGELOG.md:157)."""
    from calfkit_amd._capabilities import HandoffCapability, HandoffRequest, Transfer
    from calfkit_amd.messages import ToolCallPart
    from calfkit_amd.ports import Advert

    peers = {"expert": Advert(node="expert", kind="agent", inbox="in.expert",
                              description="knows things")}
    cap = HandoffCapability()
    schema = cap.contribute(peers)[0].schema
    props = schema.parameters["properties"]
    assert props["to"]["description"] == HandoffRequest.FIELD_DESCRIPTIONS["to"]
    assert props["reason"]["description"] == HandoffRequest.FIELD_DESCRIPTIONS["reason"]
    eff = cap.realize(ToolCallPart(call_id="c1", tool_name="handoff",
                                   arguments={"to": "expert", "reason": "r"}),
                      peers, "me")
    assert isinstance(eff, Transfer) and eff.target == "expert" and eff.reason == "r"


async def test_tool_mode_threads_to_the_request():
    """ToolMode(required/choice) rides StepSurface -> drive -> the
    EngineRequest the provider sees (reference selections v10.1)."""
    from calfkit_amd._drive import drive_step
    from calfkit_amd._engine import Engine
    from calfkit_amd._policy import TurnPolicy
    from calfkit_amd._surface import StepSurface
    from calfkit_amd.context import TurnContext
    from calfkit_amd.definitions.agent import Agent
    from calfkit_amd.definitions.selections import ToolMode
    from calfkit_amd.schemas import ToolBinding, ToolSchema
    from calfkit_amd.streaming import CollectingSink
    from calfkit_amd.testing import FakeProvider, call_tool

    schema = ToolSchema(name="add", description="Add.", parameters={
        "type": "object", "properties": {}})
    provider = FakeProvider([call_tool("add"), call_tool("add")])
    agent = Agent(name="a", model_client=provider)
    engine = Engine(agent, provider)
    ctx = TurnContext(agent_name="a", thread_id="t")

    surface = StepSurface(tools=[ToolBinding(schema=schema,
                                             target_inbox="in.add")],
                          mode=ToolMode(required=True))
    await drive_step(engine, surface, TurnPolicy(), (), CollectingSink(), ctx)
    assert provider.requests[-1].tool_choice == "required"

    surface.mode = ToolMode(choice="add")
    await drive_step(engine, surface, TurnPolicy(), (), CollectingSink(), ctx)
    assert provider.requests[-1].tool_choice == "add"
