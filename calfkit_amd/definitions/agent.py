"""The Agent definition: ctor surface parsed at the boundary.

Capability parity: reference ``definitions/agent.py``
(/root/reference/src/calfkit/definitions/agent.py:1-3) — ctor surface
``tools=/messaging=/handoffs=/emits=/listens_to=`` parsed into selection
trees, peer declarations and EventRefs; public ctor shape per
/root/reference/README.md:43-54 (``Agent(name, description, system_prompt,
model_client, peers=[Messaging(discover=True), Handoff(discover=True)])``);
agent-name injection into instructions per /root/reference/CHANGELOG.md:83;
``@agent.instructions`` per /root/reference/CHANGELOG.md:462.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Sequence

from ..context import TurnContext
from ..hooks import HookSet, parse_hooks
from ..model import Model
from ..schemas import ModelSettings, OutputSpec
from .refs import EventRef, validate_node_name
from .selections import (
    Handoff,
    Messaging,
    PeerAgents,
    Selection,
    parse_peer_selection,
    parse_tool_selection,
)
from .tools import schema_for_type

__all__ = ["Agent"]


@dataclass
class Agent:
    """A pure deployable: everything an AgentProcessor needs, as data."""

    name: str
    description: str = ""
    system_prompt: str = ""
    model_client: Any = None  # Provider instance, factory, or Model
    model: Model | None = None
    tools: Any = None
    peers: Any = None
    messaging: Any = None
    handoffs: Any = None
    emits: Sequence[str | EventRef] = ()
    listens_to: Sequence[str | EventRef] = ()
    hooks: Sequence[Any] = ()
    output_type: Any = None
    model_settings: ModelSettings | None = None
    max_steps: int = 16
    kind: str = "agent"

    # parsed-at-boundary internals
    tool_selections: list[Selection] = field(default_factory=list, repr=False)
    messaging_selections: list[PeerAgents] = field(default_factory=list, repr=False)
    handoff_selections: list[PeerAgents] = field(default_factory=list, repr=False)
    emit_refs: tuple[EventRef, ...] = field(default=(), repr=False)
    listen_refs: tuple[EventRef, ...] = field(default=(), repr=False)
    hook_set: HookSet = field(default_factory=HookSet, repr=False)
    output_spec: OutputSpec | None = field(default=None, repr=False)
    _instructions_fn: Callable[[TurnContext], str] | None = field(default=None, repr=False)

    def __post_init__(self) -> None:
        validate_node_name(self.name)
        if self.model is None:
            if isinstance(self.model_client, Model):
                self.model = self.model_client
            else:
                self.model = Model(name=f"{self.name}-model", client=self.model_client)
        self.tool_selections = parse_tool_selection(self.tools)
        # ``peers=`` mixes Messaging/Handoff wrappers; split them at the boundary
        for sel in parse_peer_selection(self.peers):
            if isinstance(sel, Handoff):
                self.handoff_selections.append(sel)
            elif isinstance(sel, Messaging):
                self.messaging_selections.append(sel)
            else:  # bare names => messaging peers
                self.messaging_selections.append(Messaging(names=sel.names,
                                                           discover=sel.discover))
        self.messaging_selections.extend(parse_peer_selection(self.messaging))
        self.handoff_selections.extend(
            Handoff(names=s.names, discover=s.discover)
            for s in parse_peer_selection(self.handoffs))
        self.emit_refs = tuple(
            e if isinstance(e, EventRef) else EventRef(e) for e in self.emits)
        self.listen_refs = tuple(
            e if isinstance(e, EventRef) else EventRef(e) for e in self.listens_to)
        self.hook_set = parse_hooks(self.hooks)
        if self.output_type is not None:
            self.output_spec = OutputSpec(
                name=getattr(self.output_type, "__name__", "output"),
                json_schema=schema_for_type(self.output_type),
            )

    # ------------------------------------------------------------ seam
    def instructions(self, fn: Callable[[TurnContext], str]) -> Callable[[TurnContext], str]:
        """``@agent.instructions`` — dynamic per-turn system-prompt suffix."""
        self._instructions_fn = fn
        return fn

    def render_system_prompt(self, ctx: TurnContext) -> str:
        """Compose the effective system prompt, with agent-name injection."""
        parts = [f"You are the agent named {self.name!r}."]
        if self.description:
            parts.append(self.description)
        if self.system_prompt:
            parts.append(self.system_prompt)
        if self._instructions_fn is not None:
            parts.append(self._instructions_fn(ctx))
        if self.model is not None:
            extra = self.model.render_instructions(ctx)
            if extra:
                parts.append(extra)
        return "\n\n".join(p for p in parts if p)
