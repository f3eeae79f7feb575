"""StepLedger: per-hop facts flushed as ONE step record (reference
_wire/steps.py:1; SURVEY §5.1 — the record carries the per-phase GPU
timings so observability sees where device time went)."""

import json

from calfkit_amd._wire import StepLedger, topology
from calfkit_amd.adapters.memory import InMemoryBus


def test_facts_accumulate_and_record_is_complete():
    led = StepLedger(node="analyst", thread_id="t1", hop_kind="agent")
    led.note(outcome="ok", prefill_ms=12.5, decode_ms=80.0)
    led.incr("tool_calls")
    led.incr("tool_calls", 3)
    rec = led.record()
    assert rec["node"] == "analyst" and rec["kind"] == "agent"
    assert rec["thread_id"] == "t1"
    assert rec["tool_calls"] == 4
    assert rec["prefill_ms"] == 12.5 and rec["decode_ms"] == 80.0
    assert rec["hop_ms"] >= 0 and rec["ts"] > 0


async def test_flush_publishes_one_wire_record():
    bus = InMemoryBus()
    seen = []

    async def sink(raw):
        seen.append(raw)

    await bus.subscribe(topology.ledger_topic(""), sink, group="obs")
    led = StepLedger(node="analyst", thread_id="t9", hop_kind="tool")
    led.note(outcome="ok")
    rec = await led.flush(bus)
    await bus.quiesce()
    assert rec["kind"] == "tool"
    assert len(seen) == 1
    wire = json.loads(seen[0].value)
    assert wire["node"] == "analyst" and wire["thread_id"] == "t9"
    assert dict(seen[0].headers)["x-calf-kind"] == b"step"


async def test_flush_never_takes_the_hop_down():
    class ExplodingBus:
        async def publish(self, *a, **k):
            raise RuntimeError("broker away")

    led = StepLedger(node="n", thread_id="t", hop_kind="agent")
    rec = await led.flush(ExplodingBus())  # must swallow, not raise
    assert rec["node"] == "n"
