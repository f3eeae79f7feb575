"""Hosting seam: the Worker resolves through here and never names a kind.

Capability parity: reference ``_hosting.py``
(/root/reference/src/calfkit/_hosting.py:1-2) names ``HostableNode``, the
``@processes`` registry, ``host()``.
"""

from __future__ import annotations

from typing import Any, Callable, Protocol, runtime_checkable

from ._wire.machinery import NodeMachinery

__all__ = ["HostableNode", "Processor", "processes", "host", "PROCESSOR_REGISTRY"]


@runtime_checkable
class Processor(Protocol):
    """A hosted node's hop pipeline."""

    node_name: str

    def topics(self) -> list[str]: ...

    async def start(self) -> None: ...

    async def handle(self, raw: Any) -> None: ...

    async def stop(self) -> None: ...


@runtime_checkable
class HostableNode(Protocol):
    """Anything with a ``kind`` and a ``name`` can be hosted."""

    kind: str
    name: str


PROCESSOR_REGISTRY: dict[str, Callable[[Any, NodeMachinery], Processor]] = {}


def processes(kind: str):
    """Register a processor factory for a node kind."""

    def deco(factory: Callable[[Any, NodeMachinery], Processor]):
        PROCESSOR_REGISTRY[kind] = factory
        return factory

    return deco


def host(node: Any, machinery: NodeMachinery) -> Processor:
    """Resolve a node to its processor through the registry — no kind names."""
    kind = getattr(node, "kind", None)
    if kind is None:
        raise TypeError(f"{node!r} is not hostable (no .kind)")
    factory = PROCESSOR_REGISTRY.get(kind)
    if factory is None:
        raise TypeError(f"no processor registered for kind {kind!r} "
                        f"(known: {sorted(PROCESSOR_REGISTRY)})")
    return factory(node, machinery)
