"""Unit: topology, envelope encode/decode, decode floor, fault funnel,
oversized-payload handling."""

import json

import pytest

from calfkit_amd._wire import topology
from calfkit_amd._wire.deliveries import (
    Call,
    Continuation,
    DeliveryDecoder,
    EchoMarker,
    Fault,
    ReplySlot,
    ToolInvoke,
    agent_call_payload,
    continuation_payload,
    encode_envelope,
    fault_payload,
    tool_call_payload,
)
from calfkit_amd._wire.publisher import TransportPublisher
from calfkit_amd.adapters.memory import InMemoryBus
from calfkit_amd.errors import DecodeError, ErrorReport
from calfkit_amd.messages import TextPart, UserMessage
from calfkit_amd.ports import RawDelivery


def test_topology_derivation():
    assert topology.inbox_topic("a") == "calf.inbox.a"
    assert topology.inbox_topic("a", "ns1") == "calf.ns1.inbox.a"
    assert topology.reply_topic("client9") == "calf.reply.client9"
    assert topology.derive_topics("a", ("tick",)) == [
        "calf.inbox.a", "calf.event.tick"]


def _raw(kind, payload, *, reply=None, marker=None, key="t1"):
    value, headers = encode_envelope(kind, to="node", thread_id="t1",
                                     payload=payload, reply=reply, marker=marker,
                                     emitter="me", hop=3)
    return RawDelivery(topic="calf.inbox.node", key=key, value=value,
                       headers=headers)


def test_agent_call_round_trip():
    d = DeliveryDecoder().decode(_raw(
        "agent_call", agent_call_payload(UserMessage((TextPart("hi"),)))))
    assert isinstance(d, Call)
    assert d.message.parts[0].text == "hi"
    assert d.thread_id == "t1"


def test_tool_call_requires_marker_and_reply():
    reply = ReplySlot(inbox="calf.inbox.caller", key="t1", node="caller")
    marker = EchoMarker(call_id="c1", frame_id="f1", caller="caller")
    d = DeliveryDecoder().decode(_raw(
        "tool_call", tool_call_payload("weather", {"city": "x"}),
        reply=reply, marker=marker))
    assert isinstance(d, ToolInvoke)
    assert d.marker.frame_id == "f1"
    assert d.reply.inbox == "calf.inbox.caller"
    with pytest.raises(DecodeError):
        DeliveryDecoder().decode(_raw("tool_call",
                                      tool_call_payload("weather", {})))


def test_continuation_and_fault():
    marker = EchoMarker(call_id="c1", caller="a")
    d = DeliveryDecoder().decode(_raw(
        "continuation", continuation_payload("result!", is_error=False),
        marker=marker))
    assert isinstance(d, Continuation) and d.content == "result!"
    report = ErrorReport.from_exception("n", "tool", ValueError("boom"))
    d = DeliveryDecoder().decode(_raw("fault", fault_payload(report),
                                      marker=marker))
    assert isinstance(d, Fault)
    assert d.report.detail.exc_type == "ValueError"
    assert d.report.detail.message == "boom"


def test_decode_floor():
    raw = RawDelivery(topic="x", key="k", value=b"\xff\xfe not json",
                      headers=(("x-calf-kind", b"agent_call"),))
    with pytest.raises(DecodeError):
        DeliveryDecoder().decode(raw)
    raw = RawDelivery(topic="x", key="k", value=b"{}", headers=())
    with pytest.raises(DecodeError):
        DeliveryDecoder().decode(raw)  # unknown kind


def test_header_fallback_to_payload_type():
    value, _ = encode_envelope("agent_call", to="n", thread_id="t",
                               payload=agent_call_payload(
                                   UserMessage((TextPart("x"),))))
    raw = RawDelivery(topic="x", key="t", value=value, headers=())  # stripped
    assert isinstance(DeliveryDecoder().decode(raw), Call)


async def test_max_message_bytes_and_fault_elision():
    bus = InMemoryBus()
    tp = TransportPublisher(bus, max_message_bytes=2048)
    pub = tp.bind("me")
    big = "z" * 10000
    with pytest.raises(ValueError):
        await pub.message_peer(inbox="calf.inbox.peer", thread_id="t",
                               message=UserMessage((TextPart(big),)),
                               reply=None)
    # oversized FAULTS are elided, not dropped
    report = ErrorReport.from_exception("n", "agent", RuntimeError("x"),
                                        huge_state=big)
    await pub.fault(report=report, thread_id="t",
                    reply=ReplySlot(inbox="calf.reply.c", key="t", node="c"))
    recs = bus.records("calf.reply.c")
    assert len(recs) == 1
    decoded = json.loads(recs[0].value)
    assert decoded["payload"]["report"]["elided"] is True
    assert decoded["payload"]["report"]["run_state"] == {}


async def test_emitter_headers_present():
    bus = InMemoryBus()
    tp = TransportPublisher(bus)
    pub = tp.bind("emitter-node")
    await pub.message_peer(inbox="calf.inbox.p", thread_id="t",
                           message=UserMessage((TextPart("x"),)), reply=None,
                           hop=7)
    rec = bus.records("calf.inbox.p")[0]
    assert rec.header("x-calf-emitter") == b"emitter-node"
    assert rec.header("x-calf-hop") == b"7"
    assert rec.header("x-calf-kind") == b"agent_call"


def test_envelope_golden_bytes():
    """Wire-format pin: the exact bytes and headers of a canonical agent
    call.  This is the byte-compatibility contract for Kafka (and the
    bundled mesh, which carries the same records verbatim) — any drift in
    field names, ordering, separators or header keys breaks consumers and
    MUST show up here as a deliberate change."""
    value, headers = encode_envelope(
        "agent_call", to="researcher", thread_id="t-1",
        payload={"message": {"role": "user",
                             "parts": [{"type": "text", "text": "hi"}]}},
        reply=ReplySlot(inbox="calf.reply.abc", key="t-1", node="client"),
        marker=EchoMarker(call_id="c-5", frame_id="f-9", caller="boss"),
        emitter="client-1", hop=2)
    assert value == (
        b'{"routing":{"to":"researcher","thread_id":"t-1",'
        b'"reply":{"inbox":"calf.reply.abc","key":"t-1","node":"client"}},'
        b'"state":{"marker":{"call_id":"c-5","frame_id":"f-9",'
        b'"caller":"boss"},"hop":2},'
        b'"payload":{"message":{"role":"user",'
        b'"parts":[{"type":"text","text":"hi"}]}}}'
    ), value
    assert [(k if isinstance(k, str) else k.decode()) for k, _ in headers] \
        == ["x-calf-kind", "x-calf-emitter", "x-calf-hop"]
    assert [v for _, v in headers] == [b"agent_call", b"client-1", b"2"]
