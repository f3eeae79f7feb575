"""Worker lifecycle: startup/shutdown hooks + the @resource single-yield bracket.

Capability parity: reference ``lifecycle.py``
(/root/reference/src/calfkit/lifecycle.py:1-3) names
``on_startup/after_startup/on_shutdown/after_shutdown`` and the ``@resource``
single-yield bracket; node & worker lifecycle hooks per
/root/reference/CHANGELOG.md:322.
"""

from __future__ import annotations

import inspect
from dataclasses import dataclass, field
from typing import Any, AsyncGenerator, Callable

__all__ = ["Lifecycle", "resource", "ResourceDef"]


@dataclass
class ResourceDef:
    """A declared resource: a single-yield async generator factory.

    The part before ``yield`` runs at startup; the yielded value is the
    resource (exposed by name); the part after runs at shutdown — exactly
    once each, LIFO on shutdown.
    """

    name: str
    factory: Callable[..., AsyncGenerator[Any, None]]


def resource(fn: Callable[..., AsyncGenerator[Any, None]]) -> ResourceDef:
    """Decorator turning a single-yield async generator into a ResourceDef."""
    if not inspect.isasyncgenfunction(fn):
        raise TypeError("@resource requires an async generator function with one yield")
    return ResourceDef(name=fn.__name__, factory=fn)


@dataclass
class Lifecycle:
    """The worker's lifecycle registry and runner."""

    on_startup: list[Callable[..., Any]] = field(default_factory=list)
    after_startup: list[Callable[..., Any]] = field(default_factory=list)
    on_shutdown: list[Callable[..., Any]] = field(default_factory=list)
    after_shutdown: list[Callable[..., Any]] = field(default_factory=list)
    resources: list[ResourceDef] = field(default_factory=list)

    _live: dict[str, Any] = field(default_factory=dict)
    _gens: list[AsyncGenerator[Any, None]] = field(default_factory=list)

    def add_resource(self, rdef: ResourceDef) -> None:
        self.resources.append(rdef)

    def get(self, name: str) -> Any:
        return self._live[name]

    async def startup(self) -> None:
        for fn in self.on_startup:
            await _maybe_await(fn())
        for rdef in self.resources:
            gen = rdef.factory()
            value = await gen.__anext__()
            self._live[rdef.name] = value
            self._gens.append(gen)
        for fn in self.after_startup:
            await _maybe_await(fn())

    async def shutdown(self) -> None:
        for fn in self.on_shutdown:
            await _maybe_await(fn())
        for gen in reversed(self._gens):
            try:
                await gen.__anext__()
            except StopAsyncIteration:
                pass
            else:  # pragma: no cover - misuse guard
                raise RuntimeError("@resource generator must yield exactly once")
        self._gens.clear()
        self._live.clear()
        for fn in self.after_shutdown:
            await _maybe_await(fn())


async def _maybe_await(r: Any) -> Any:
    if inspect.isawaitable(r):
        return await r
    return r
