"""StepLedger: per-hop facts flushed as one step record.

Capability parity: reference ``_wire/steps.py``
(/root/reference/src/calfkit/_wire/steps.py:1).

MI355X extension: the ledger carries per-phase GPU timings
(prefill_ms / decode_ms from ``ProviderFacts``) so every step record shows
where device time went.
"""

from __future__ import annotations

import json
import logging
import time
from dataclasses import dataclass, field
from typing import Any

from ..ports import MessageBus
from . import topology

__all__ = ["StepLedger"]

_log = logging.getLogger("calfkit.steps")


@dataclass
class StepLedger:
    """Accumulates one hop's facts; ``flush`` writes exactly one record."""

    node: str
    thread_id: str
    hop_kind: str = "agent"
    facts: dict[str, Any] = field(default_factory=dict)
    _t0: float = field(default_factory=time.perf_counter)

    def note(self, **facts: Any) -> None:
        self.facts.update(facts)

    def incr(self, key: str, by: int = 1) -> None:
        self.facts[key] = int(self.facts.get(key, 0)) + by

    def record(self) -> dict[str, Any]:
        return {
            "node": self.node,
            "thread_id": self.thread_id,
            "kind": self.hop_kind,
            "hop_ms": (time.perf_counter() - self._t0) * 1e3,
            "ts": time.time(),
            **self.facts,
        }

    async def flush(self, bus: MessageBus | None = None, namespace: str = "") -> dict[str, Any]:
        rec = self.record()
        _log.info("step %s", json.dumps(rec, default=str))
        from ..utils.metrics import metrics  # no-op shim unless enabled

        metrics.hops.labels(self.node, self.hop_kind,
                            str(rec.get("outcome", "ok"))).inc()
        metrics.hop_seconds.labels(self.node, self.hop_kind).observe(
            rec["hop_ms"] / 1e3)
        if bus is not None:
            try:
                await bus.publish(topology.ledger_topic(namespace), self.thread_id,
                                  json.dumps(rec, default=str).encode(),
                                  (("x-calf-kind", b"step"),))
            except Exception:  # the ledger never takes a hop down
                _log.exception("step-ledger publish failed")
        return rec
