"""Pure projections: agent-POV history and transfer seeds.

Capability parity: reference ``_projection.py``
(/root/reference/src/calfkit/_projection.py:1-2) names ``project_for_model``
and ``project_for_transfer``; agent-POV projection for multi-agent channels
per /root/reference/CHANGELOG.md:308 (and its bug-fix history at
/root/reference/CHANGELOG.md:167 — the correctness-sensitive part is
*attribution*: another agent's assistant output must reach this agent's model
as attributed input, never as its own assistant turn).
"""

from __future__ import annotations

from .messages import (
    AssistantMessage,
    Message,
    SystemMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    UserMessage,
    parts_to_text,
)

__all__ = ["project_for_model", "project_for_transfer"]


def project_for_model(messages: tuple[Message, ...], *, agent_name: str,
                      system_prompt: str = "") -> tuple[Message, ...]:
    """Project the shared thread history into *this* agent's point of view.

    Rules:
    - exactly one SystemMessage, first, composed from ``system_prompt``
      (stored system messages are dropped — the prompt is per-agent);
    - an AssistantMessage authored by this agent stays assistant, with its
      tool-call parts kept only when their returns are also present for this
      agent (open calls survive; foreign tool chatter does not leak);
    - an AssistantMessage authored by a *different* agent becomes an
      attributed UserMessage ("[name] text") — peers are inputs, not self;
    - ToolMessages pass through only when they answer a call this agent made
      (identified by call_id); orphaned tool returns are attributed as text;
    - UserMessages pass through.
    """
    own_call_ids = {
        p.call_id
        for m in messages if isinstance(m, AssistantMessage) and m.author == agent_name
        for p in m.parts if isinstance(p, ToolCallPart)
    }

    out: list[Message] = []
    if system_prompt:
        out.append(SystemMessage((TextPart(system_prompt),)))
    for m in messages:
        if isinstance(m, SystemMessage):
            continue
        if isinstance(m, AssistantMessage):
            if m.author == agent_name:
                out.append(m)
            else:
                text = parts_to_text(m.parts)
                if text:
                    out.append(UserMessage((TextPart(f"[{m.author}] {text}"),),
                                           author=m.author))
        elif isinstance(m, ToolMessage):
            keep = tuple(p for p in m.parts
                         if isinstance(p, ToolReturnPart) and p.call_id in own_call_ids)
            if keep:
                out.append(ToolMessage(keep))
            else:
                for p in m.parts:
                    if isinstance(p, ToolReturnPart):
                        out.append(UserMessage(
                            (TextPart(f"[tool:{p.tool_name}] {p.content}"),),
                            author=p.tool_name))
        else:
            out.append(m)
    return tuple(out)


def project_for_transfer(messages: tuple[Message, ...], *, source: str,
                         target: str, reason: str = "") -> tuple[Message, ...]:
    """Build the seed history a handoff target receives.

    The target gets the conversation as attributed inputs — the source
    agent's assistant turns become "[source] ..." user messages, open tool
    plumbing is stripped — plus a final notice naming the transfer.
    """
    seed: list[Message] = []
    for m in messages:
        if isinstance(m, SystemMessage):
            continue
        if isinstance(m, AssistantMessage):
            text = parts_to_text(m.parts)
            if text:
                seed.append(UserMessage((TextPart(f"[{m.author}] {text}"),),
                                        author=m.author))
        elif isinstance(m, ToolMessage):
            for p in m.parts:
                if isinstance(p, ToolReturnPart) and not p.is_error:
                    seed.append(UserMessage(
                        (TextPart(f"[tool:{p.tool_name}] {p.content}"),),
                        author=p.tool_name))
        else:
            seed.append(m)
    notice = f"[handoff] {source} handed this conversation to {target}."
    if reason:
        notice += f" Reason: {reason}"
    seed.append(UserMessage((TextPart(notice),), author="system"))
    return tuple(seed)
