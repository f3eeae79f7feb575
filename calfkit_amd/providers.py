"""Provider SPI: the base classes a model client implements.

Capability parity: reference ``providers.py``
(/root/reference/src/calfkit/providers.py:1-3) names ``HttpProvider`` and
``Accumulator``; OpenAI/Anthropic clients per /root/reference/README.md:49 and
/root/reference/CHANGELOG.md:448,455.

MI355X divergence (the north star): the first-class provider here is the
in-process ``LocalModelClient`` (``calfkit_amd.engine.local``) running
hand-written CDNA4 HIP kernels — re-exported lazily below so importing this
module never loads torch.  Remote HTTP providers remain as a thin SPI for
parity (``HttpProvider``), the only core importer of an HTTP client, imported
in-function.
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Any

from .messages import (
    AssistantMessage,
    Part,
    ProviderFacts,
    TextPart,
    ToolCallPart,
)
from .schemas import EngineRequest
from .streaming import ChunkSink, StreamChunk

if TYPE_CHECKING:  # pragma: no cover
    from .engine.local import LocalModelClient  # noqa: F401

__all__ = ["Accumulator", "HttpProvider", "LocalModelClient"]


class Accumulator:
    """Builds the final AssistantMessage from streamed increments.

    Providers push deltas here; the accumulator forwards them to the sink
    (the real-time reflection) and keeps the facts for the final message.
    """

    def __init__(self, sink: ChunkSink, *, node: str = "", thread_id: str = ""):
        self._sink = sink
        self._node = node
        self._thread_id = thread_id
        self._seq = 0
        self._text: list[str] = []
        self._tool_calls: list[ToolCallPart] = []

    def text_delta(self, delta: str) -> None:
        self._text.append(delta)
        self._sink.emit(StreamChunk(event="text_delta", seq=self._next(), text=delta,
                                    node=self._node, thread_id=self._thread_id))

    def tool_call(self, part: ToolCallPart) -> None:
        self._tool_calls.append(part)
        self._sink.emit(StreamChunk(
            event="tool_call", seq=self._next(),
            data={"call_id": part.call_id, "tool_name": part.tool_name,
                  "arguments": part.arguments},
            node=self._node, thread_id=self._thread_id))

    def finish(self, facts: ProviderFacts, *, author: str = "assistant") -> AssistantMessage:
        parts: list[Part] = []
        text = "".join(self._text)
        if text:
            parts.append(TextPart(text))
        parts.extend(self._tool_calls)
        return AssistantMessage(tuple(parts), author=author, facts=facts)

    def _next(self) -> int:
        self._seq += 1
        return self._seq


class HttpProvider:
    """Tier-2 base for remote HTTP model APIs.

    Subclasses implement ``build_payload``/``parse_response``.  The HTTP
    client import happens in-function so the core stays import-light.
    """

    base_url: str = ""
    timeout_s: float = 120.0

    def __init__(self, base_url: str = "", headers: dict[str, str] | None = None):
        if base_url:
            self.base_url = base_url
        self.headers = dict(headers or {})

    def build_payload(self, request: EngineRequest) -> dict[str, Any]:  # pragma: no cover
        raise NotImplementedError

    def parse_response(self, data: dict[str, Any], sink: ChunkSink) -> AssistantMessage:  # pragma: no cover
        raise NotImplementedError

    async def complete(self, request: EngineRequest, sink: ChunkSink) -> AssistantMessage:
        import httpx  # sole core importer, in-function by design

        async with httpx.AsyncClient(timeout=self.timeout_s) as client:
            resp = await client.post(self.base_url, json=self.build_payload(request),
                                     headers=self.headers)
            resp.raise_for_status()
            return self.parse_response(resp.json(), sink)


def __getattr__(name: str) -> Any:
    if name == "LocalModelClient":
        from .engine.local import LocalModelClient as L
        return L
    raise AttributeError(name)
