"""Shipped test doubles — fakes come from the product, never test-local.

Capability parity: reference ``testing.py``
(/root/reference/src/calfkit/testing.py:1-2): ``FakeProvider``,
``FakeMCPSession``; scripted ``FunctionModel`` pattern per
/root/reference/CONTRIBUTING.md:83-87.
"""

from __future__ import annotations

import uuid
from typing import Any, Callable, Sequence

from .mcp import MCPResult, MCPToolInfo
from .messages import (
    AssistantMessage,
    ProviderFacts,
    TextPart,
    ToolCallPart,
    Usage,
)
from .schemas import EngineRequest
from .streaming import ChunkSink, StreamChunk

__all__ = ["FakeProvider", "FunctionModel", "EchoProvider", "FakeMCPSession",
           "ToolCallDirector", "reply", "call_tool"]


def reply(text: str) -> AssistantMessage:
    return AssistantMessage((TextPart(text),))


def call_tool(tool_name: str, call_id: str = "", **arguments: Any) -> AssistantMessage:
    return AssistantMessage((ToolCallPart(
        call_id=call_id or uuid.uuid4().hex, tool_name=tool_name,
        arguments=arguments),))


class FakeProvider:
    """Scripted provider: returns the queued messages in order.

    When the script runs dry it raises, surfacing test bugs loudly.
    Every request is recorded on ``.requests`` for assertion.
    """

    def __init__(self, script: Sequence[AssistantMessage | str] = ()):
        self.script: list[AssistantMessage] = [
            reply(s) if isinstance(s, str) else s for s in script]
        self.requests: list[EngineRequest] = []

    def queue(self, *messages: AssistantMessage | str) -> "FakeProvider":
        self.script.extend(reply(m) if isinstance(m, str) else m for m in messages)
        return self

    async def complete(self, request: EngineRequest, sink: ChunkSink) -> AssistantMessage:
        self.requests.append(request)
        if not self.script:
            raise AssertionError("FakeProvider script exhausted")
        message = self.script.pop(0)
        for p in message.parts:
            if isinstance(p, TextPart):
                sink.emit(StreamChunk(event="text_delta", text=p.text))
        facts = message.facts or ProviderFacts(model_name="fake",
                                               usage=Usage(1, 1))
        return AssistantMessage(message.parts, author=message.author, facts=facts)


class FunctionModel:
    """A provider whose behavior is a function of the request."""

    def __init__(self, fn: Callable[[EngineRequest], AssistantMessage | str]):
        self.fn = fn
        self.requests: list[EngineRequest] = []

    async def complete(self, request: EngineRequest, sink: ChunkSink) -> AssistantMessage:
        self.requests.append(request)
        out = self.fn(request)
        if isinstance(out, str):
            out = reply(out)
        for p in out.parts:
            if isinstance(p, TextPart):
                sink.emit(StreamChunk(event="text_delta", text=p.text))
        if out.facts is None:
            out = AssistantMessage(out.parts, author=out.author,
                                   facts=ProviderFacts(model_name="function"))
        return out


class EchoProvider:
    """The stub echo provider of BASELINE config 1: echoes the last user text,
    calling each offered tool once first when asked to."""

    def __init__(self, *, use_tools: bool = False):
        self.use_tools = use_tools
        self._called: set[str] = set()

    async def complete(self, request: EngineRequest, sink: ChunkSink) -> AssistantMessage:
        if self.use_tools:
            for t in request.tools:
                if t.name not in self._called and not t.name.startswith(("send_", "handoff")):
                    self._called.add(t.name)
                    args = {}
                    props = t.parameters.get("properties", {})
                    for name, spec in props.items():
                        if spec.get("type") == "string":
                            args[name] = "test"
                        elif spec.get("type") in ("integer", "number"):
                            args[name] = 1
                    return call_tool(t.name, **args)
        last = ""
        for m in reversed(request.messages):
            if m.role in ("user", "tool"):
                for p in m.parts:
                    if isinstance(p, TextPart):
                        last = p.text
                        break
                else:
                    continue
                break
        text = f"echo: {last}"
        sink.emit(StreamChunk(event="text_delta", text=text))
        return AssistantMessage((TextPart(text),),
                                facts=ProviderFacts(model_name="echo"))


class FakeMCPSession:
    """Scripted MCP session: tools + canned results (or functions)."""

    def __init__(self, tools: Sequence[MCPToolInfo] = (),
                 results: dict[str, Any] | None = None):
        self.tools = list(tools)
        self.results = dict(results or {})
        self.calls: list[tuple[str, dict[str, Any]]] = []
        self.closed = False

    async def list_tools(self) -> list[MCPToolInfo]:
        return list(self.tools)

    async def call_tool(self, name: str, arguments: dict[str, Any]) -> MCPResult:
        self.calls.append((name, dict(arguments)))
        r = self.results.get(name)
        if callable(r):
            r = r(arguments)
        if isinstance(r, MCPResult):
            return r
        if r is None:
            return MCPResult(content=f"no script for {name}", is_error=True)
        return MCPResult(content=r)

    async def close(self) -> None:
        self.closed = True


class ToolCallDirector:
    """Wraps a real provider, directing its tool-call behavior.

    The inner provider (e.g. the GPU LocalModelClient) still does ALL its
    generation work; the director then overrides the *decision* — on the
    first step of a thread it swaps the text for the scripted tool calls,
    and after the returns arrive it lets the provider's text stand.  This
    keeps benches honest (full engine work per step) while random-init
    weights cannot emit real calls (BASELINE config 3's shape).
    """

    def __init__(self, inner: Any, calls: Sequence[tuple[str, dict]] = (),
                 rounds: int = 1):
        self.inner = inner
        self.calls = list(calls)
        self.rounds = rounds

    async def complete(self, request: Any, sink: Any) -> AssistantMessage:
        message = await self.inner.complete(request, sink)
        # count this thread's prior assistant steps to find our position
        steps = sum(1 for m in request.messages if m.role == "assistant")
        if steps < self.rounds and self.calls:
            parts = tuple(
                ToolCallPart(call_id=uuid.uuid4().hex, tool_name=name,
                             arguments=dict(args))
                for name, args in self.calls)
            return AssistantMessage(parts, facts=message.facts)
        return message
