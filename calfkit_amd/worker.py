"""Worker — the composition root. No kinds, no processors named here;
heavy adapter defaults imported lazily in-function.

Capability parity: reference ``worker.py``
(/root/reference/src/calfkit/worker.py:1-2); worker lifecycle per
/root/reference/CHANGELOG.md:322; horizontal scaling = replicas sharing a
consumer group per /root/reference/CHANGELOG.md:623.

MI355X addition: ``gpu`` pins this worker replica to one device
(``one-Worker-per-GPU`` — calfkit's independent-scaling semantics map to one
288 GB HBM rank each, BASELINE.json north_star).
"""

from __future__ import annotations

import asyncio
import logging
import os
from typing import Any, Sequence

from ._hosting import Processor, host
from ._wire.machinery import NodeMachinery
from ._wire.publisher import TransportPublisher
from ._wire import topology
from .lifecycle import Lifecycle, ResourceDef
from .ports import Advert, Directory, MessageBus, ThreadStore

_log = logging.getLogger("calfkit.worker")

__all__ = ["Worker"]


class Worker:
    def __init__(self, *nodes: Any, name: str = "", bus: MessageBus | None = None,
                 store: ThreadStore | None = None, directory: Directory | None = None,
                 namespace: str = "", gpu: int | None = None,
                 max_message_bytes: int = 1024 * 1024,
                 resources: Sequence[ResourceDef] = ()):
        # register the processor kinds (the registry import, not kind names)
        from . import _processors  # noqa: F401

        self.nodes = list(nodes)
        self.name = name or f"worker-{os.getpid()}"
        self.namespace = namespace
        self.gpu = gpu
        self.lifecycle = Lifecycle()
        for r in resources:
            self.lifecycle.add_resource(r)

        if bus is None or store is None or directory is None:
            from .adapters.memory import (  # lazy default: broker-free
                InMemoryBus,
                InMemoryDirectory,
                InProcessThreadStore,
            )
            bus = bus or InMemoryBus()
            store = store or InProcessThreadStore()
            directory = directory or InMemoryDirectory()
        self.bus = bus
        self.store = store
        self.directory = directory
        self.machinery = NodeMachinery(
            bus=bus, store=store, directory=directory,
            transport=TransportPublisher(bus, namespace=namespace,
                                         max_message_bytes=max_message_bytes),
            lifecycle=self.lifecycle, worker_name=self.name, namespace=namespace)
        self.processors: list[Processor] = []
        self._running = asyncio.Event()
        self._stop = asyncio.Event()

    # ------------------------------------------------------------ lifecycle
    async def start(self) -> None:
        if self.gpu is not None:
            self._pin_gpu(self.gpu)
        port = os.environ.get("CALF_METRICS_PORT")
        if port:  # opt-in Prometheus exporter (utils/metrics.py)
            from .utils.metrics import metrics

            if not metrics.enable(int(port)):
                _log.warning("CALF_METRICS_PORT set but prometheus_client "
                             "is not installed; metrics stay off")
        await self.lifecycle.startup()
        for node in self.nodes:
            processor = host(node, self.machinery)
            self.processors.append(processor)
        topics: list[str] = []
        for p in self.processors:
            topics.extend(p.topics())
        await self.bus.ensure_topics(topics)  # no reliance on auto-create
        for p in self.processors:
            await p.start()
            for topic in p.topics():
                await self.bus.subscribe(topic, p.handle, group=f"{p.node_name}")
            await self._advertise(p)
        self._running.set()
        _log.info("worker %s hosting %s", self.name,
                  [p.node_name for p in self.processors])

    async def _advertise(self, processor: Processor) -> None:
        node = next(n for n in self.nodes if getattr(n, "name", None) == processor.node_name)
        caps: dict[str, Any] = {}
        kind = getattr(node, "kind", "?")
        if kind == "tool":
            caps["parameters"] = node.schema.parameters
        elif kind == "toolbox" and hasattr(processor, "advert_capabilities"):
            caps = processor.advert_capabilities()
        elif kind == "agent":
            caps["handoff"] = True
        await self.directory.advertise(Advert(
            node=processor.node_name, kind=kind,
            description=getattr(node, "description", ""),
            capabilities=caps,
            inbox=topology.inbox_topic(processor.node_name, self.namespace),
            worker=self.name))

    async def stop(self) -> None:
        for p in self.processors:
            try:
                await self.directory.withdraw(p.node_name)
            except Exception:  # a dead transport must not wedge shutdown
                _log.warning("withdraw failed for %s", p.node_name,
                             exc_info=True)
            await p.stop()
        await self.lifecycle.shutdown()
        self._stop.set()
        self._running.clear()

    async def run(self) -> None:
        """Start and serve until cancelled (the CLI entry)."""
        await self.start()
        try:
            await self._stop.wait()
        finally:
            if self._running.is_set():
                await self.stop()

    async def __aenter__(self) -> "Worker":
        await self.start()
        return self

    async def __aexit__(self, *exc: Any) -> None:
        if self._running.is_set():
            await self.stop()

    # ------------------------------------------------------------------ GPU
    @staticmethod
    def _pin_gpu(index: int) -> None:
        """Pin this replica to one MI355X before any engine initializes."""
        os.environ.setdefault("HIP_VISIBLE_DEVICES", str(index))
