"""Byte-golden wire-format snapshots.

The Kafka/mesh byte-compatibility contract is the EXACT bytes of every
delivery kind — JSON envelope (routing / state / typed payload split per
/root/reference/CHANGELOG.md:506) plus the x-calf-* headers
(/root/reference/CHANGELOG.md:194,418-422).  These literals are frozen
snapshots: any serialization drift (key order, separators, header names,
slot shapes) fails here even though round-trip tests would still pass.
"""

from calfkit_amd._wire.deliveries import EchoMarker, ReplySlot, encode_envelope
from calfkit_amd.errors import ErrorDetail, ErrorReport

MARKER = EchoMarker(call_id="c-1", frame_id="f-1", caller="agent-a")
REPLY = ReplySlot(inbox="calf.reply.client-1", key="t-9", node="client-1")

GOLDEN = {
    "agent_call": (
        dict(payload={"message": {"role": "user",
                                  "parts": [{"kind": "text", "text": "hi"}]}},
             reply=REPLY),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":{"inbox":"calf.reply.client-1","key":"t-9","node":"client-1"}},"state":{"marker":null,"hop":2},"payload":{"message":{"role":"user","parts":[{"kind":"text","text":"hi"}]}}}',
    ),
    "tool_call": (
        dict(payload={"tool_name": "get_weather",
                      "arguments": {"city": "SF"}},
             reply=REPLY, marker=MARKER),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":{"inbox":"calf.reply.client-1","key":"t-9","node":"client-1"}},"state":{"marker":{"call_id":"c-1","frame_id":"f-1","caller":"agent-a"},"hop":2},"payload":{"tool_name":"get_weather","arguments":{"city":"SF"}}}',
    ),
    "continuation": (
        dict(payload={"content": "sunny"}, marker=MARKER),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":null},"state":{"marker":{"call_id":"c-1","frame_id":"f-1","caller":"agent-a"},"hop":2},"payload":{"content":"sunny"}}',
    ),
    "fault": (
        dict(payload={"report": ErrorReport(
            node="tool-x", kind="tool",
            detail=ErrorDetail("ValueError", "boom", ())).to_wire()}),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":null},"state":{"marker":null,"hop":2},"payload":{"report":{"node":"tool-x","kind":"tool","detail":{"exc_type":"ValueError","message":"boom","frames":[]},"causes":[],"run_state":{},"elided":false}}}',
    ),
    "event": (
        dict(payload={"name": "tick", "data": 1}),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":null},"state":{"marker":null,"hop":2},"payload":{"name":"tick","data":1}}',
    ),
    "chunk": (
        dict(payload={"seq": 1, "text": "he"}),
        b'{"routing":{"to":"agent-a","thread_id":"t-9","reply":null},"state":{"marker":null,"hop":2},"payload":{"seq":1,"text":"he"}}',
    ),
}


def test_every_delivery_kind_bytes_are_frozen():
    for kind, (kw, expected) in GOLDEN.items():
        value, headers = encode_envelope(kind, to="agent-a", thread_id="t-9",
                                         emitter="node-1", hop=2, **kw)
        assert value == expected, f"{kind} envelope bytes drifted:\n{value}"
        assert headers == (
            ("x-calf-kind", kind.encode()),
            ("x-calf-emitter", b"node-1"),
            ("x-calf-hop", b"2"),
        ), f"{kind} headers drifted: {headers}"


def test_decoder_round_trips_the_golden_bytes():
    """The frozen bytes must also DECODE — bytes and model stay in sync."""
    from calfkit_amd._wire.deliveries import DeliveryDecoder
    from calfkit_amd.ports import RawDelivery

    dec = DeliveryDecoder()
    for kind, (kw, value) in GOLDEN.items():
        if kind in ("event", "chunk"):
            continue  # client-plane kinds; node decoder doesn't take them
        raw = RawDelivery(topic="calf.inbox.agent-a", key="t-9", value=value,
                          headers=(("x-calf-kind", kind.encode()),),
                          offset=0, partition=0)
        d = dec.decode(raw)
        assert d is not None
        assert getattr(d, "thread_id", "t-9") == "t-9"
