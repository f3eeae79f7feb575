"""Frame protocol of the bundled mesh broker: length-prefixed JSON with
base64 payloads (the 8-bit-clean transport under the kafka-compatible
bus — adapters/mesh/protocol.py)."""

import asyncio

import pytest

from calfkit_amd.adapters.mesh import protocol as p


def test_encode_roundtrips_through_reader():
    frame = {"op": "publish", "topic": "t", "key": "k",
             "value_b64": p.b64e(b"\x00\xffbinary"), "headers": []}
    wire = p.encode_frame(frame)
    assert int.from_bytes(wire[:4], "big") == len(wire) - 4

    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data(wire)
        reader.feed_eof()
        return await p.read_frame(reader)

    got = asyncio.run(run())
    assert got == frame
    assert p.b64d(got["value_b64"]) == b"\x00\xffbinary"


def test_reader_survives_byte_at_a_time_delivery():
    """TCP gives no message boundaries; the reader must reassemble."""
    frames = [p.encode_frame({"op": "ping", "req_id": i}) for i in range(3)]
    stream = b"".join(frames)

    async def run():
        reader = asyncio.StreamReader()

        async def feeder():
            for i in range(0, len(stream), 1):
                reader.feed_data(stream[i:i + 1])
                if i % 7 == 0:
                    await asyncio.sleep(0)
            reader.feed_eof()

        task = asyncio.ensure_future(feeder())
        out = [await p.read_frame(reader) for _ in range(3)]
        await task
        assert await p.read_frame(reader) is None  # clean EOF -> None
        return out

    got = asyncio.run(run())
    assert [f["req_id"] for f in got] == [0, 1, 2]


def test_truncated_frame_yields_none_not_garbage():
    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data(p.encode_frame({"op": "ping", "req_id": 9})[:-2])
        reader.feed_eof()
        return await p.read_frame(reader)

    assert asyncio.run(run()) is None


def test_oversized_frame_is_rejected():
    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data((p.MAX_FRAME + 1).to_bytes(4, "big") + b"x")
        with pytest.raises(ValueError, match="too large"):
            await p.read_frame(reader)

    asyncio.run(run())


def test_headers_wire_roundtrip_preserves_binary():
    hs = (("x-calf-kind", b"call"), ("x-bin", bytes(range(256))))
    assert p.headers_from_wire(p.headers_to_wire(hs)) == hs
