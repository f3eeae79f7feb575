"""Frame protocol of the bundled mesh broker.

Length-prefixed JSON frames: 4-byte big-endian length + UTF-8 JSON object.
Record values/headers are base64 inside the JSON (values are themselves
JSON envelopes today, but the transport stays 8-bit clean on purpose).

Frame ops (client -> broker):
  publish   {op, topic, key, value_b64, headers: [[k, v_b64]...]}
  subscribe {op, topic, group}
  ensure    {op, topics: [...]}
  advertise {op, advert: {...}}
  withdraw  {op, node}
  view      {op, req_id}
  watch     {op}
  ping      {op, req_id}

broker -> client:
  record    {op, topic, key, value_b64, headers, offset}
  view      {op, req_id, adverts: [...]}
  dirupdate {op, adverts: [...]}
  pong      {op, req_id}
"""

from __future__ import annotations

import asyncio
import base64
import json
from typing import Any

MAX_FRAME = 64 * 1024 * 1024


def encode_frame(obj: dict[str, Any]) -> bytes:
    body = json.dumps(obj, separators=(",", ":")).encode()
    return len(body).to_bytes(4, "big") + body


async def read_frame(reader: asyncio.StreamReader) -> dict[str, Any] | None:
    try:
        head = await reader.readexactly(4)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    n = int.from_bytes(head, "big")
    if n > MAX_FRAME:
        raise ValueError(f"frame too large: {n}")
    try:
        body = await reader.readexactly(n)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    return json.loads(body)


def b64e(b: bytes) -> str:
    return base64.b64encode(b).decode()


def b64d(s: str) -> bytes:
    return base64.b64decode(s)


def headers_to_wire(headers: tuple[tuple[str, bytes], ...]) -> list[list[str]]:
    return [[k, b64e(v)] for k, v in headers]


def headers_from_wire(items: list[list[str]]) -> tuple[tuple[str, bytes], ...]:
    return tuple((k, b64d(v)) for k, v in items)
