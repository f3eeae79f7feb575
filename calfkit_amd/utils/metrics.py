"""Optional Prometheus metrics exporter (beyond-reference observability).

The reference ships no metrics exporter; this exposes worker
hop counts/latencies and engine throughput when prometheus_client is
installed — a no-op shim otherwise, so the core never depends on it.
"""

from __future__ import annotations

from typing import Any

__all__ = ["Metrics", "metrics"]


class _Noop:
    def labels(self, *a: Any, **kw: Any) -> "_Noop":
        return self

    def inc(self, *a: Any) -> None:
        pass

    def observe(self, *a: Any) -> None:
        pass

    def set(self, *a: Any) -> None:
        pass


class Metrics:
    def __init__(self) -> None:
        self.enabled = False
        self.hops = _Noop()
        self.hop_seconds = _Noop()
        self.engine_tokens = _Noop()
        self.engine_batch = _Noop()

    def enable(self, port: int = 9464) -> bool:
        try:
            from prometheus_client import Counter, Gauge, Histogram, start_http_server
        except ImportError:
            return False
        if not self.enabled:
            self.hops = Counter("calfkit_hops", "agent/tool hops",
                                ["node", "kind", "outcome"])
            self.hop_seconds = Histogram("calfkit_hop_seconds", "hop latency",
                                         ["node", "kind"])
            self.engine_tokens = Counter("calfkit_engine_tokens",
                                         "generated tokens", ["model", "phase"])
            self.engine_batch = Gauge("calfkit_engine_batch", "decode batch",
                                      ["model"])
            start_http_server(port)
            self.enabled = True
        return True


metrics = Metrics()
