"""NodeMachinery: the worker's offering — kinds take what they need.

Capability parity: reference ``_wire/machinery.py``
(/root/reference/src/calfkit/_wire/machinery.py:1-2).
"""

from __future__ import annotations

from dataclasses import dataclass, field

from ..lifecycle import Lifecycle
from ..ports import Directory, MessageBus, ThreadStore
from .deliveries import DeliveryDecoder
from .publisher import TransportPublisher
from .steps import StepLedger

__all__ = ["NodeMachinery"]


@dataclass
class NodeMachinery:
    """Everything a processor may need, assembled once per worker."""

    bus: MessageBus
    store: ThreadStore
    directory: Directory
    transport: TransportPublisher
    lifecycle: Lifecycle
    worker_name: str = "worker"
    namespace: str = ""
    decoder: DeliveryDecoder = field(default_factory=DeliveryDecoder)

    def ledger(self, node: str, thread_id: str, hop_kind: str) -> StepLedger:
        return StepLedger(node=node, thread_id=thread_id, hop_kind=hop_kind)
