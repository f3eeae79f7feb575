"""Unit: hook kernel, lifecycle brackets, stream chunk vocabulary."""

import pytest

from calfkit_amd.context import TurnContext
from calfkit_amd.errors import ErrorReport
from calfkit_amd.hooks import (
    AgentHook,
    BeforeStep,
    Drop,
    HookSet,
    OnToolError,
    parse_hooks,
)
from calfkit_amd.lifecycle import Lifecycle, resource
from calfkit_amd.messages import TextPart, UserMessage
from calfkit_amd.streaming import CollectingSink, StreamChunk


def ctx():
    return TurnContext(agent_name="a", thread_id="t")


async def test_drop_stops_delivery():
    class Dropper(AgentHook):
        async def on_delivery(self, ctx, delivery):
            return Drop()

    hooks = parse_hooks([Dropper()])
    assert await hooks.run_on_delivery(ctx(), None) is False
    assert await parse_hooks([]).run_on_delivery(ctx(), None) is True


async def test_before_step_merging():
    class A(AgentHook):
        async def before_step(self, ctx):
            return BeforeStep(inject=[UserMessage((TextPart("a"),))],
                              extra_instructions="one")

    class B(AgentHook):
        async def before_step(self, ctx):
            return BeforeStep(extra_instructions="two")

    merged = await parse_hooks([A(), B()]).run_before_step(ctx())
    assert len(merged.inject) == 1
    assert merged.extra_instructions == "one\ntwo"


async def test_function_hook_and_tool_error_default():
    calls = []

    def after_step(ctx, message):
        calls.append(message)

    hooks = parse_hooks([after_step])
    await hooks.run_after_step(ctx(), UserMessage((TextPart("m"),)))
    assert len(calls) == 1

    report = ErrorReport.from_exception("n", "tool", ValueError("kapow"))
    part = await hooks.run_on_tool_error(ctx(), report, "c1", "t")
    assert part.is_error and "kapow" in str(part.content)


async def test_parse_hooks_rejects_junk():
    with pytest.raises(TypeError):
        parse_hooks([42])


async def test_lifecycle_resource_bracket_lifo():
    events = []

    @resource
    async def db():
        events.append("open-db")
        yield "DB"
        events.append("close-db")

    @resource
    async def cache():
        events.append("open-cache")
        yield "CACHE"
        events.append("close-cache")

    lc = Lifecycle()
    lc.on_startup.append(lambda: events.append("on_startup"))
    lc.after_startup.append(lambda: events.append("after_startup"))
    lc.on_shutdown.append(lambda: events.append("on_shutdown"))
    lc.after_shutdown.append(lambda: events.append("after_shutdown"))
    lc.add_resource(db)
    lc.add_resource(cache)
    await lc.startup()
    assert lc.get("db") == "DB" and lc.get("cache") == "CACHE"
    await lc.shutdown()
    assert events == ["on_startup", "open-db", "open-cache", "after_startup",
                      "on_shutdown", "close-cache", "close-db",
                      "after_shutdown"]


def test_stream_chunk_wire_and_sink():
    c = StreamChunk(event="text_delta", seq=3, text="hi", node="n",
                    thread_id="t")
    assert StreamChunk.from_wire(c.to_wire()) == c
    sink = CollectingSink()
    sink.emit(c)
    sink.emit(StreamChunk(event="step_finished"))
    assert sink.text == "hi"


def test_accumulator_builds_final_message_from_deltas():
    """Accumulator (providers.py): deltas stream to the sink in real time
    and fold into ONE AssistantMessage — 'a real-time reflection of facts,
    never the facts' (reference streaming.py:1-2)."""
    from calfkit_amd.messages import ProviderFacts, TextPart, ToolCallPart
    from calfkit_amd.providers import Accumulator

    sink = CollectingSink()
    acc = Accumulator(sink, node="analyst", thread_id="t1")
    acc.text_delta("Hel")
    acc.text_delta("lo")
    acc.tool_call(ToolCallPart(call_id="c1", tool_name="search",
                               arguments={"q": "x"}))
    msg = acc.finish(ProviderFacts(model_name="m", finish_reason="tool_calls"),
                     author="analyst")
    # the reflection: ordered chunks with increasing seq
    events = [(c.event, c.seq) for c in sink.chunks]
    assert events == [("text_delta", 1), ("text_delta", 2), ("tool_call", 3)]
    assert all(c.node == "analyst" and c.thread_id == "t1"
               for c in sink.chunks)
    # the facts: one message, text folded, calls attached
    assert isinstance(msg.parts[0], TextPart) and msg.parts[0].text == "Hello"
    assert msg.tool_calls[0].tool_name == "search"
    assert msg.facts.finish_reason == "tool_calls"
    assert msg.author == "analyst"
