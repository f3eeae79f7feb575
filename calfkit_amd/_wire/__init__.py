"""The wire custodians: one gateway, two lifetimes.

Capability parity: reference ``_wire/``
(/root/reference/src/calfkit/_wire/__init__.py:1-3).
"""

from .deliveries import (
    Call,
    Continuation,
    Delivery,
    DeliveryDecoder,
    EchoMarker,
    Fault,
    ReplySlot,
    ToolInvoke,
)
from .machinery import NodeMachinery
from .publisher import BoundPublisher, FaultFunnel, TransportPublisher
from .steps import StepLedger

__all__ = [
    "Call",
    "Continuation",
    "Delivery",
    "DeliveryDecoder",
    "EchoMarker",
    "Fault",
    "ReplySlot",
    "ToolInvoke",
    "NodeMachinery",
    "BoundPublisher",
    "FaultFunnel",
    "TransportPublisher",
    "StepLedger",
]
