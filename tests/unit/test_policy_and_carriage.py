"""Unit: turn policy (end shapes, open-call scan) + carriage admission."""

import pytest

from calfkit_amd._carriage import Admission, ThreadCarriage
from calfkit_amd._policy import TurnPolicy, scan_open_calls
from calfkit_amd.adapters.memory import InProcessThreadStore
from calfkit_amd.messages import (
    AssistantMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    UserMessage,
)
from calfkit_amd.ports import AppendDisposition, ThreadStatus
from calfkit_amd.schemas import OutputSpec


def test_scan_open_calls():
    msgs = (
        AssistantMessage((ToolCallPart("a", "t1", {}), ToolCallPart("b", "t2", {}))),
        ToolMessage((ToolReturnPart("a", "t1", "ok"),)),
    )
    open_ = scan_open_calls(msgs)
    assert set(open_) == {"b"}


def test_end_shape_plain_text():
    p = TurnPolicy()
    assert p.check_end_shape(AssistantMessage((TextPart("done"),)), None).ok
    assert not p.check_end_shape(
        AssistantMessage((ToolCallPart("c", "t", {}),)), None).ok


def test_end_shape_structured_output():
    spec = OutputSpec(name="W", json_schema={
        "type": "object", "properties": {"city": {"type": "string"},
                                          "temp": {"type": "integer"}},
        "required": ["city", "temp"]})
    p = TurnPolicy()
    good = p.check_end_shape(
        AssistantMessage((TextPart('{"city": "x", "temp": 3}'),)), spec)
    assert good.ok and good.output == {"city": "x", "temp": 3}
    # final_result union prefix unwraps
    pref = p.check_end_shape(
        AssistantMessage((TextPart('{"final_result": {"city": "x", "temp": 1}}'),)),
        spec)
    assert pref.ok and pref.output["temp"] == 1
    # fenced json extracted
    fenced = p.check_end_shape(AssistantMessage(
        (TextPart('sure!\n```json\n{"city": "y", "temp": 9}\n```'),)), spec)
    assert fenced.ok
    bad = p.check_end_shape(AssistantMessage((TextPart('{"city": 5}'),)), spec)
    assert not bad.ok and bad.correction is not None
    assert "W" in bad.correction.parts[0].text


async def _carriage(store, pending=None, status=ThreadStatus.OPEN):
    c = ThreadCarriage(store, "node", "t1")
    await c.hydrate()
    c.record.status = status
    c.record.pending = pending or {}
    return c


async def test_admission_table():
    store = InProcessThreadStore()
    c = await _carriage(store)
    assert c.admit("call") == Admission.PROCEED
    assert c.admit("continuation", "x") == Admission.DISCARD  # nothing pending

    c = await _carriage(store, pending={"c1": {"done": False}},
                        status=ThreadStatus.PENDING)
    assert c.admit("continuation", "c1") == Admission.PROCEED
    assert c.admit("continuation", "zz") == Admission.ABSORBED  # unknown id
    assert c.admit("call") == Admission.ABSORBED  # early message mid-frame
    assert c.admit("fault", "c1") == Admission.PROCEED

    c = await _carriage(store, status=ThreadStatus.DEAD)
    assert c.admit("call") == Admission.DEAD


async def test_store_first_mirror_on_appended():
    store = InProcessThreadStore()
    c = ThreadCarriage(store, "n", "t")
    await c.hydrate()
    c.working_extend([UserMessage((TextPart("a"),))])
    assert len(c.messages) == 0  # not mirrored before flush
    assert len(c.working_view()) == 1
    assert await c.flush() == AppendDisposition.APPENDED
    assert len(c.messages) == 1 and c.record.revision == 1

    # optimistic concurrency: a stale twin conflicts and does NOT mirror
    twin = ThreadCarriage(store, "n", "t")
    await twin.hydrate()
    await c.working_extend([UserMessage((TextPart("b"),))]) if False else None
    c.working_extend([UserMessage((TextPart("b"),))])
    await c.flush()
    twin.working_extend([UserMessage((TextPart("X"),))])
    assert await twin.flush() == AppendDisposition.CONFLICT
    fresh = await store.load("n", "t")
    assert len(fresh.messages) == 2  # X never landed


async def test_metadata_persists():
    store = InProcessThreadStore()
    c = ThreadCarriage(store, "n", "meta")
    await c.hydrate()
    c.record.metadata["reply"] = {"inbox": "i", "key": "k", "node": "z"}
    c.working_extend([UserMessage((TextPart("m"),))])
    await c.flush()
    again = ThreadCarriage(store, "n", "meta")
    rec = await again.hydrate()
    assert rec.metadata["reply"]["inbox"] == "i"
