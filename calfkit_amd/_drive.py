"""drive_step — the orchestrator: one pure decision function per step,
wire-free on both edges.

Capability parity: reference ``_drive.py``
(/root/reference/src/calfkit/_drive.py:1-3) names ``drive_step``, the
decision inventory D1–D7, ``StepResolution``, ``PendingToolCall``,
``PendingPeerCall``; parallel tool calls keyed by ``frame_id`` per
/root/reference/CHANGELOG.md:469,372.

The decision inventory as implemented here:

  D1  step the engine (projection + surface are inputs; hooks already ran)
  D2  classify the output's calls: capability calls realize into effects
  D3  a Transfer effect ends the turn immediately (handoff wins the step)
  D4  mesh tool calls become one pending frame (parallel fan-out, frame_id)
  D5  dispatches with reply_expected become pending peer calls (same frame)
  D6  no calls: end-shape check against the output spec; bounded correction
      loop re-steps the engine on failure
  D7  resolve: final message | pending frame, plus the effect list

A hop that ends in a pending frame is re-entered later by a Continuation
delivery; drive_step is called again only when the frame completes
(the carriage absorbs partial arrivals).
"""

from __future__ import annotations

import uuid
from dataclasses import dataclass, field
from typing import Any

from ._capabilities import Dispatch, Effect, Reject, Transfer
from ._engine import Engine
from ._policy import TurnPolicy
from ._surface import StepSurface
from .context import TurnContext
from .messages import (
    AssistantMessage,
    Message,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
)
from .schemas import ModelSettings
from .streaming import ChunkSink

__all__ = ["PendingToolCall", "PendingPeerCall", "StepResolution", "drive_step"]


@dataclass(frozen=True)
class PendingToolCall:
    call_id: str
    tool_name: str
    arguments: dict[str, Any]
    target_inbox: str


@dataclass(frozen=True)
class PendingPeerCall:
    call_id: str
    target: str
    target_inbox: str


@dataclass
class StepResolution:
    """What one drive produced. The realize phase acts on this and only this."""

    #: messages to persist, in order (engine outputs + synthesized returns)
    new_messages: list[Message] = field(default_factory=list)
    #: wire effects to publish after persisting
    effects: list[Effect] = field(default_factory=list)
    pending_tool_calls: list[PendingToolCall] = field(default_factory=list)
    pending_peer_calls: list[PendingPeerCall] = field(default_factory=list)
    #: the batch identity when anything is pending
    frame_id: str | None = None
    #: the turn's final message when it finished
    final: AssistantMessage | None = None
    #: parsed structured output when an OutputSpec was in force
    output: Any = None

    @property
    def finished(self) -> bool:
        return not self.pending_tool_calls and not self.pending_peer_calls


async def drive_step(engine: Engine, surface: StepSurface, policy: TurnPolicy,
                     history: tuple[Message, ...], sink: ChunkSink,
                     ctx: TurnContext, *,
                     settings_override: ModelSettings | None = None) -> StepResolution:
    resolution = StepResolution()
    corrections = 0
    work = history

    while True:
        # D1 — step the engine
        # ToolMode rides the request: required forces some call, choice
        # forces ONE named tool (reference selections v10.1 ToolMode)
        mode = surface.mode
        tool_choice = mode.choice or ("required" if mode.required else "")
        message = await engine.step(work, surface.tool_schemas(), sink, ctx,
                                    settings_override=settings_override,
                                    tool_choice=tool_choice)
        resolution.new_messages.append(message)
        work = work + (message,)

        calls = message.tool_calls
        if calls:
            # D2 — capability calls realize into effects; mesh calls pend
            effects: list[Effect] = []
            rejects: list[ToolReturnPart] = []
            pending_tools: list[PendingToolCall] = []
            pending_peers: list[PendingPeerCall] = []
            unknown: list[ToolReturnPart] = []
            for call in calls:
                binding = surface.binding(call.tool_name)
                if binding is not None and binding.mode == "capability":
                    effect = _realize_capability(surface, call, engine.agent.name)
                    if isinstance(effect, Reject):
                        rejects.append(ToolReturnPart(
                            call_id=call.call_id, tool_name=call.tool_name,
                            content=effect.reason, is_error=True))
                    elif isinstance(effect, Transfer):
                        # D3 — handoff ends the turn now; other calls are moot
                        resolution.effects.append(effect)
                        resolution.final = message
                        return resolution
                    elif isinstance(effect, Dispatch):
                        effects.append(effect)
                        if effect.reply_expected:
                            pending_peers.append(PendingPeerCall(
                                call_id=call.call_id, target=effect.target,
                                target_inbox=effect.target_inbox))
                        else:
                            rejects.append(ToolReturnPart(
                                call_id=call.call_id, tool_name=call.tool_name,
                                content=f"message sent to {effect.target}"))
                elif binding is not None:
                    # D4 — mesh tool fan-out; args are validated against the
                    # ADVERTISED schema before anything hits the wire, so a
                    # malformed call costs one correction step, not a mesh
                    # round trip (caller-side validation,
                    # /root/reference/CHANGELOG.md:43)
                    err = _validate_args(binding.schema.parameters,
                                         call.arguments)
                    if err is not None:
                        rejects.append(ToolReturnPart(
                            call_id=call.call_id, tool_name=call.tool_name,
                            content=f"invalid arguments: {err}",
                            is_error=True))
                    else:
                        pending_tools.append(PendingToolCall(
                            call_id=call.call_id, tool_name=call.tool_name,
                            arguments=dict(call.arguments),
                            target_inbox=binding.target_inbox))
                else:
                    unknown.append(ToolReturnPart(
                        call_id=call.call_id, tool_name=call.tool_name,
                        content=f"unknown tool {call.tool_name!r}", is_error=True))

            # effects publish regardless of how the rest of the step resolves
            resolution.effects.extend(effects)

            synthesized = rejects + unknown
            if synthesized:
                resolution.new_messages.append(ToolMessage(tuple(synthesized)))
                work = work + (ToolMessage(tuple(synthesized)),)

            if pending_tools or pending_peers:
                # D5/D7 — a pending frame ends the hop
                resolution.pending_tool_calls = pending_tools
                resolution.pending_peer_calls = pending_peers
                resolution.frame_id = uuid.uuid4().hex
                return resolution

            # synthesized returns and/or fire-and-forget dispatches:
            # the turn continues with another engine step (bounded)
            corrections += 1
            if corrections > policy.max_corrections:
                resolution.final = message
                return resolution
            continue

        # D6 — no calls: end-shape
        shape = policy.check_end_shape(message, engine.agent.output_spec)
        if shape.ok:
            resolution.final = message
            resolution.output = shape.output
            return resolution
        corrections += 1
        if corrections > policy.max_corrections or shape.correction is None:
            resolution.final = message  # give up; ship what we have
            return resolution
        resolution.new_messages.append(shape.correction)
        work = work + (shape.correction,)


def _validate_args(schema: dict[str, Any], args: dict[str, Any]) -> str | None:
    """Structural check of call args against a JSON-schema parameters dict."""
    from ._policy import _check_schema

    if not isinstance(args, dict):
        return "arguments must be an object"
    props = schema.get("properties", {})
    unknown = set(args) - set(props) if props else set()
    if props and unknown:
        return f"unknown arguments: {sorted(unknown)}"
    for req in schema.get("required", ()):  # required fields
        if req not in args:
            return f"missing required argument {req!r}"
    for k, v in args.items():
        if k in props:
            err = _check_schema(v, props[k], f"$.{k}")
            if err:
                return err
    return None


def _realize_capability(surface: StepSurface, call: ToolCallPart,
                        agent_name: str) -> Effect:
    for cap in surface.capabilities:
        peers = (surface.messaging_peers if cap.capability_name == "messaging"
                 else surface.handoff_peers)
        effect = cap.realize(call, peers, agent_name)
        if effect is not None:
            return effect
    return Reject(call_id=call.call_id, tool_name=call.tool_name,
                  reason=f"no capability realizes {call.tool_name!r}")
