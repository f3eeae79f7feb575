"""StepSurface: merge resolved selections with capability contributions.

Capability parity: reference ``_surface.py``
(/root/reference/src/calfkit/_surface.py:1-3) names ``StepSurface`` /
``build_step_surface`` — "merges resolved selections (the ruled ladder) with
capability contributions".
"""

from __future__ import annotations

from dataclasses import dataclass, field

from ._capabilities import HandoffCapability, MessagingCapability, SideloadedCapability
from .definitions.agent import Agent
from .definitions.selections import ToolMode, resolve_selection
from .errors import SelectionError
from .ports import Advert, DirectoryView
from .schemas import ToolBinding, ToolSchema

__all__ = ["StepSurface", "build_step_surface"]


@dataclass
class StepSurface:
    """Everything callable from one step, fully resolved. Per-hop data."""

    tools: list[ToolBinding] = field(default_factory=list)
    messaging_peers: dict[str, Advert] = field(default_factory=dict)
    handoff_peers: dict[str, Advert] = field(default_factory=dict)
    capabilities: list[SideloadedCapability] = field(default_factory=list)
    mode: ToolMode = field(default_factory=ToolMode)

    def tool_schemas(self) -> tuple[ToolSchema, ...]:
        return tuple(b.schema for b in self.tools)

    def binding(self, tool_name: str) -> ToolBinding | None:
        for b in self.tools:
            if b.schema.name == tool_name:
                return b
        return None


def build_step_surface(agent: Agent, view: DirectoryView, *,
                       strict: bool = True) -> StepSurface:
    """Resolve the agent's selections against the live view, then side-load.

    Mesh tools first (the ruled ladder), then capability pseudo-tools —
    a mesh tool shadowing a capability name is a configuration error.
    """
    surface = StepSurface()

    resolved = resolve_selection(agent.tool_selections, view, strict=strict)
    surface.tools.extend(resolved.tools)
    surface.mode = resolved.mode

    msg_peers = resolve_selection(list(agent.messaging_selections), view,
                                  strict=strict).peers
    hand_peers = resolve_selection(list(agent.handoff_selections), view,
                                   strict=strict).peers
    # an agent never messages or hands off to itself
    msg_peers.pop(agent.name, None)
    hand_peers.pop(agent.name, None)
    surface.messaging_peers = msg_peers
    surface.handoff_peers = hand_peers

    mesh_names = {b.schema.name for b in surface.tools}
    if msg_peers:
        cap = MessagingCapability()
        surface.capabilities.append(cap)
        for binding in cap.contribute(msg_peers):
            if binding.schema.name in mesh_names:
                raise SelectionError(
                    f"mesh tool {binding.schema.name!r} shadows the messaging capability")
            surface.tools.append(binding)
    if hand_peers:
        cap = HandoffCapability()
        surface.capabilities.append(cap)
        for binding in cap.contribute(hand_peers):
            if binding.schema.name in mesh_names:
                raise SelectionError(
                    f"mesh tool {binding.schema.name!r} shadows the handoff capability")
            surface.tools.append(binding)

    return surface
