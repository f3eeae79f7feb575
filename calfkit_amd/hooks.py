"""Agent hook kernel: five hook points, Drop, prebuilt hooks.

Capability parity: reference ``hooks.py``
(/root/reference/src/calfkit/hooks.py:1-3) names ``AgentHook``, ``Drop``,
``BeforeStep``, prebuilt ``HandoffNotice``/``OnToolError``, ``BUILTIN_HOOKS``,
``parse_hooks``; the ``on_tool_error`` surfacing guarantee is
/root/reference/CHANGELOG.md:386,56.

Hook points (the five): on_delivery, before_step, after_step, on_tool_error,
on_handoff.  A hook may return ``Drop`` from ``on_delivery`` to discard the
delivery, or a ``BeforeStep`` amendment from ``before_step`` to inject
messages / override settings for this one step.
"""

from __future__ import annotations

import inspect
from dataclasses import dataclass, field
from typing import Any, Sequence

from .context import TurnContext
from .errors import ErrorReport
from .messages import Message, TextPart, ToolReturnPart, UserMessage
from .schemas import ModelSettings

__all__ = [
    "Drop",
    "BeforeStep",
    "AgentHook",
    "HandoffNotice",
    "OnToolError",
    "BUILTIN_HOOKS",
    "parse_hooks",
    "HookSet",
]

HOOK_POINTS = ("on_delivery", "before_step", "after_step", "on_tool_error", "on_handoff")


class Drop:
    """Sentinel: returned from ``on_delivery`` to discard the delivery."""

    def __repr__(self) -> str:  # pragma: no cover
        return "Drop"


DROP = Drop()


@dataclass
class BeforeStep:
    """Amendment a ``before_step`` hook may return for this one step."""

    inject: list[Message] = field(default_factory=list)
    settings: ModelSettings | None = None
    extra_instructions: str = ""


class AgentHook:
    """Base class. Subclass and override any of the five points.

    Every point is optional; sync or async implementations both work.
    """

    async def on_delivery(self, ctx: TurnContext, delivery: Any) -> Any:
        return None

    async def before_step(self, ctx: TurnContext) -> BeforeStep | None:
        return None

    async def after_step(self, ctx: TurnContext, message: Message) -> None:
        return None

    async def on_tool_error(self, ctx: TurnContext, report: ErrorReport,
                            call_id: str, tool_name: str) -> ToolReturnPart | None:
        return None

    async def on_handoff(self, ctx: TurnContext, target: str, reason: str) -> None:
        return None


class HandoffNotice(AgentHook):
    """Prebuilt: when a thread arrives via handoff, surface a notice message."""

    async def on_handoff(self, ctx: TurnContext, target: str, reason: str) -> None:
        ctx.metadata["handoff_notice"] = f"(handed off to {target}: {reason})"

    async def before_step(self, ctx: TurnContext) -> BeforeStep | None:
        notice = ctx.metadata.pop("handoff_incoming", None)
        if notice:
            return BeforeStep(inject=[UserMessage(
                (TextPart(f"[handoff] {notice}"),), author="system")])
        return None


class OnToolError(AgentHook):
    """Prebuilt: turn a tool fault into a tool-return the model can see.

    Guarantees the no-hang property: a failed tool call always produces a
    continuation, never a stuck pending batch.
    """

    async def on_tool_error(self, ctx: TurnContext, report: ErrorReport,
                            call_id: str, tool_name: str) -> ToolReturnPart | None:
        return ToolReturnPart(call_id=call_id, tool_name=tool_name,
                              content=f"tool error: {report.detail.message}",
                              is_error=True)


BUILTIN_HOOKS: tuple[AgentHook, ...] = (HandoffNotice(), OnToolError())


def parse_hooks(hooks: Sequence[Any] | None) -> "HookSet":
    """Parse the user-facing ``hooks=`` ctor argument at the boundary.

    Accepts AgentHook instances, AgentHook subclasses (instantiated), or
    bare callables named after a hook point.
    """
    parsed: list[AgentHook] = list(BUILTIN_HOOKS)
    for h in hooks or ():
        if isinstance(h, AgentHook):
            parsed.append(h)
        elif isinstance(h, type) and issubclass(h, AgentHook):
            parsed.append(h())
        elif callable(h) and getattr(h, "__name__", "") in HOOK_POINTS:
            parsed.append(_FunctionHook(h.__name__, h))
        else:
            raise TypeError(f"not a hook: {h!r}")
    return HookSet(tuple(parsed))


class _FunctionHook(AgentHook):
    def __init__(self, point: str, fn: Any):
        self._point = point
        self._fn = fn

    def __getattribute__(self, name: str) -> Any:
        if name in HOOK_POINTS and name == object.__getattribute__(self, "_point"):
            fn = object.__getattribute__(self, "_fn")

            async def call(*args: Any, **kw: Any) -> Any:
                r = fn(*args, **kw)
                if inspect.isawaitable(r):
                    r = await r
                return r

            return call
        return object.__getattribute__(self, name)


@dataclass(frozen=True)
class HookSet:
    """The parsed, ordered hook chain for one agent."""

    hooks: tuple[AgentHook, ...] = ()

    async def run_on_delivery(self, ctx: TurnContext, delivery: Any) -> bool:
        """Returns False when a hook dropped the delivery."""
        for h in self.hooks:
            if isinstance(await h.on_delivery(ctx, delivery), Drop) :
                return False
        return True

    async def run_before_step(self, ctx: TurnContext) -> BeforeStep:
        merged = BeforeStep()
        for h in self.hooks:
            amendment = await h.before_step(ctx)
            if amendment is None:
                continue
            merged.inject.extend(amendment.inject)
            if amendment.settings is not None:
                merged.settings = amendment.settings
            if amendment.extra_instructions:
                merged.extra_instructions = (
                    merged.extra_instructions + "\n" + amendment.extra_instructions
                ).strip()
        return merged

    async def run_after_step(self, ctx: TurnContext, message: Message) -> None:
        for h in self.hooks:
            await h.after_step(ctx, message)

    async def run_on_tool_error(self, ctx: TurnContext, report: ErrorReport,
                                call_id: str, tool_name: str) -> ToolReturnPart:
        result: ToolReturnPart | None = None
        for h in self.hooks:
            r = await h.on_tool_error(ctx, report, call_id, tool_name)
            if r is not None:
                result = r
        if result is None:  # unreachable with builtins installed, but total anyway
            result = ToolReturnPart(call_id=call_id, tool_name=tool_name,
                                    content=f"tool error: {report.detail.message}",
                                    is_error=True)
        return result

    async def run_on_handoff(self, ctx: TurnContext, target: str, reason: str) -> None:
        for h in self.hooks:
            await h.on_handoff(ctx, target, reason)
