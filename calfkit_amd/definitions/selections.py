"""Selection algebra: leaves + wrappers + the pure resolver.

Capability parity: reference ``definitions/selections.py``
(/root/reference/src/calfkit/definitions/selections.py:1-4) names ``ToolMode``,
``Tools``, ``Toolboxes``, ``PeerAgents``, ``resolve_selection``, the
specificity ladder and ``terminal_names``; runtime discovery
(``Tools(discover=True)``) per /root/reference/CHANGELOG.md:156.

Semantics (original design):
- A selection tree is parsed from the Agent ctor at the boundary.
- ``resolve_selection(tree, view)`` is pure: tree + DirectoryView -> resolved
  bindings.  Discovery selections resolve against the live view at step time;
  static selections resolve against declared names and fail loudly when a
  named node is absent.
- The specificity ladder: an explicit name beats a discovered set; a later
  duplicate (same terminal name) is rejected at parse time, while static-vs-
  discovered collisions resolve to the static entry.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Union

from ..errors import SelectionError
from ..ports import DirectoryView
from ..schemas import ToolBinding, ToolSchema
from .refs import ToolRef, ToolboxRef, AgentRef

__all__ = [
    "ToolMode",
    "Tools",
    "Toolboxes",
    "PeerAgents",
    "Messaging",
    "Handoff",
    "Selection",
    "ResolvedSurface",
    "resolve_selection",
    "terminal_names",
    "parse_tool_selection",
    "parse_peer_selection",
]


@dataclass(frozen=True)
class ToolMode:
    """How tool choice is presented to the model."""

    required: bool = False
    choice: str | None = None  # force one named tool


@dataclass(frozen=True)
class Tools:
    """Select mesh tools by name, or every discovered tool."""

    names: tuple[str, ...] = ()
    discover: bool = False
    mode: ToolMode = field(default_factory=ToolMode)


@dataclass(frozen=True)
class Toolboxes:
    """Select MCP toolboxes. Dual-mode: by name or discovered.

    Namespaced tool names (``box.tool``) per /root/reference/CHANGELOG.md:160.
    """

    names: tuple[str, ...] = ()
    discover: bool = False


@dataclass(frozen=True)
class PeerAgents:
    """Select peer agents for messaging/handoff."""

    names: tuple[str, ...] = ()
    discover: bool = False


@dataclass(frozen=True)
class Messaging(PeerAgents):
    """Peer messaging capability selection."""


@dataclass(frozen=True)
class Handoff(PeerAgents):
    """Handoff capability selection."""


Selection = Union[Tools, Toolboxes, PeerAgents]


@dataclass
class ResolvedSurface:
    """Output of the pure resolver: bindings the step surface merges."""

    tools: list[ToolBinding] = field(default_factory=list)
    peers: dict[str, Any] = field(default_factory=dict)  # name -> Advert
    mode: ToolMode = field(default_factory=ToolMode)


def terminal_names(selections: list[Selection]) -> list[str]:
    """Every statically named terminal across the selection trees."""
    out: list[str] = []
    for s in selections:
        out.extend(s.names)
    return out


def parse_tool_selection(value: Any) -> list[Selection]:
    """Parse the Agent ctor ``tools=`` argument at the boundary.

    Accepts: a Tools/Toolboxes instance, a list mixing tool defs (anything
    with ``.name`` and ``.schema``), strings, refs, and selection objects.
    """
    from .tools import ToolDef  # cycle-free: tools imports refs only

    if value is None:
        return []
    if isinstance(value, (Tools, Toolboxes)):
        return [value]
    if not isinstance(value, (list, tuple)):
        value = [value]
    names: list[str] = []
    selections: list[Selection] = []
    for v in value:
        if isinstance(v, (Tools, Toolboxes)):
            selections.append(v)
        elif isinstance(v, ToolDef):
            names.append(v.name)
        elif isinstance(v, (ToolRef, ToolboxRef)):
            names.append(v.name)
        elif isinstance(v, str):
            names.append(v)
        else:
            raise SelectionError(f"cannot select tool from {v!r}")
    dupes = {n for n in names if names.count(n) > 1}
    if dupes:
        raise SelectionError(f"duplicate tool selections: {sorted(dupes)}")
    if names:
        selections.insert(0, Tools(names=tuple(names)))
    return selections


def parse_peer_selection(value: Any) -> list[PeerAgents]:
    if value is None:
        return []
    if isinstance(value, PeerAgents):
        return [value]
    if not isinstance(value, (list, tuple)):
        value = [value]
    out: list[PeerAgents] = []
    names: list[str] = []
    for v in value:
        if isinstance(v, PeerAgents):
            out.append(v)
        elif isinstance(v, AgentRef):
            names.append(v.name)
        elif isinstance(v, str):
            names.append(v)
        elif hasattr(v, "name"):  # an Agent definition object
            names.append(v.name)
        else:
            raise SelectionError(f"cannot select peer from {v!r}")
    if names:
        out.insert(0, PeerAgents(names=tuple(names)))
    return out


def resolve_selection(selections: list[Selection], view: DirectoryView,
                      *, strict: bool = True) -> ResolvedSurface:
    """Pure: selection trees + a directory view -> resolved bindings.

    The specificity ladder: static names resolve first and win collisions;
    discovered sets fill in around them.
    """
    surface = ResolvedSurface()
    seen: set[str] = set()

    def add_tool(name: str, advert: Any) -> None:
        if name in seen:
            return
        seen.add(name)
        caps = advert.capabilities or {}
        schema = ToolSchema(
            name=name,
            description=advert.description,
            parameters=dict(caps.get("parameters", {"type": "object", "properties": {}})),
        )
        surface.tools.append(ToolBinding(schema=schema, target_inbox=advert.inbox))

    # ladder rung 1: static names
    for sel in selections:
        if isinstance(sel, Tools):
            if sel.mode.required or sel.mode.choice:
                surface.mode = sel.mode
            for name in sel.names:
                advert = view.get(name)
                if advert is None or advert.kind != "tool":
                    if strict:
                        raise SelectionError(f"tool {name!r} not present in the mesh")
                    continue
                add_tool(name, advert)
        elif isinstance(sel, Toolboxes):
            for name in sel.names:
                advert = view.get(name)
                if advert is None or advert.kind != "toolbox":
                    if strict:
                        raise SelectionError(f"toolbox {name!r} not present in the mesh")
                    continue
                _add_toolbox(surface, seen, name, advert)
        elif isinstance(sel, PeerAgents):
            for name in sel.names:
                advert = view.get(name)
                if advert is None or advert.kind != "agent":
                    if strict:
                        raise SelectionError(f"peer agent {name!r} not present in the mesh")
                    continue
                surface.peers[name] = advert

    # ladder rung 2: discovery fills around static picks
    for sel in selections:
        if not getattr(sel, "discover", False):
            continue
        if isinstance(sel, Tools):
            for advert in view.tools():
                add_tool(advert.node, advert)
        elif isinstance(sel, Toolboxes):
            for advert in view.toolboxes():
                _add_toolbox(surface, seen, advert.node, advert)
        elif isinstance(sel, PeerAgents):
            for advert in view.agents():
                surface.peers.setdefault(advert.node, advert)

    return surface


def _add_toolbox(surface: ResolvedSurface, seen: set[str], box: str, advert: Any) -> None:
    """Expand a toolbox advert into namespaced tool bindings (``box.tool``)."""
    caps = advert.capabilities or {}
    for tool in caps.get("tools", ()):  # [{name, description, parameters}]
        qual = f"{box}.{tool['name']}"
        if qual in seen:
            continue
        seen.add(qual)
        schema = ToolSchema(name=qual, description=tool.get("description", ""),
                            parameters=dict(tool.get("parameters", {})))
        surface.tools.append(ToolBinding(schema=schema, target_inbox=advert.inbox))
