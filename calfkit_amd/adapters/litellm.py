"""LiteLLMProvider — the sole litellm importer in the codebase.

Capability parity: reference ``adapters/litellm.py``
(/root/reference/src/calfkit/adapters/litellm.py:1) pins the litellm SDK to
exactly this module (pyproject sole-importer contract,
/root/reference/pyproject.toml:130-155).  litellm is not installed in this
image, so the import happens lazily on first ``complete()`` and raises a
clear error when absent; the request/response mapping is the OpenAI
chat-completions shape litellm normalizes every backend to.
"""

from __future__ import annotations

import json
from typing import Any

from ..messages import (
    AssistantMessage,
    Part,
    ProviderFacts,
    TextPart,
    ToolCallPart,
    Usage,
)
from ..schemas import EngineRequest
from ..streaming import ChunkSink

__all__ = ["LiteLLMProvider"]


class LiteLLMProvider:
    """Provider over ``litellm.acompletion`` (any backend litellm routes to).

    Implements the same ``Provider`` port as ``LocalModelClient`` /
    ``OpenAIChatProvider``; construction is SDK-free so importing the
    adapters package never loads litellm (the import-weight guarantee).
    """

    def __init__(self, model: str, **completion_kwargs: Any):
        self.model = model
        self._kwargs = completion_kwargs
        self._litellm: Any = None

    def _sdk(self) -> Any:
        if self._litellm is None:
            try:
                import litellm
            except ImportError as e:
                raise RuntimeError(
                    "LiteLLMProvider needs the litellm SDK (not installed in "
                    "this image); use LocalModelClient or OpenAIChatProvider"
                ) from e
            self._litellm = litellm
        return self._litellm

    # ------------------------------------------------------------- mapping
    def build_payload(self, request: EngineRequest) -> dict[str, Any]:
        msgs: list[dict[str, Any]] = []
        for m in request.messages:
            text = " ".join(p.text for p in m.parts if isinstance(p, TextPart))
            entry: dict[str, Any] = {"role": m.role, "content": text}
            calls = [p for p in m.parts if isinstance(p, ToolCallPart)]
            if calls:
                entry["tool_calls"] = [
                    {"id": c.call_id, "type": "function",
                     "function": {"name": c.tool_name,
                                  "arguments": json.dumps(c.arguments)}}
                    for c in calls]
            if m.role == "tool":
                entry["tool_call_id"] = getattr(m, "call_id", "")
            msgs.append(entry)
        s = request.settings
        payload: dict[str, Any] = {
            "model": request.model_name or self.model,
            "messages": msgs,
            "max_tokens": s.max_tokens,
            "temperature": s.temperature,
            "top_p": s.top_p,
            **self._kwargs,
        }
        if s.seed is not None:
            payload["seed"] = s.seed
        if s.stop:
            payload["stop"] = list(s.stop)
        if request.tools:
            payload["tools"] = [{"type": "function", "function": t.to_wire()}
                                for t in request.tools]
            if request.tool_choice == "required":
                payload["tool_choice"] = "required"
            elif request.tool_choice:
                payload["tool_choice"] = {
                    "type": "function",
                    "function": {"name": request.tool_choice}}
        return payload

    @staticmethod
    def parse_response(resp: Any) -> AssistantMessage:
        choice = resp.choices[0]
        msg = choice.message
        parts: list[Part] = []
        content = getattr(msg, "content", None)
        if content:
            parts.append(TextPart(str(content)))
        for tc in getattr(msg, "tool_calls", None) or ():
            fn = tc.function
            try:
                args = json.loads(fn.arguments) if fn.arguments else {}
            except json.JSONDecodeError:
                args = {"_raw": fn.arguments}
            parts.append(ToolCallPart(call_id=str(tc.id),
                                      tool_name=str(fn.name), arguments=args))
        u = getattr(resp, "usage", None)
        usage = Usage(prompt_tokens=getattr(u, "prompt_tokens", 0) or 0,
                      completion_tokens=getattr(u, "completion_tokens", 0) or 0)
        facts = ProviderFacts(model_name=str(getattr(resp, "model", "")),
                              finish_reason=str(getattr(choice, "finish_reason",
                                                        "") or "stop"),
                              usage=usage)
        return AssistantMessage(tuple(parts), author="assistant", facts=facts)

    # ---------------------------------------------------------------- port
    async def complete(self, request: EngineRequest,
                       sink: ChunkSink) -> AssistantMessage:
        litellm = self._sdk()
        resp = await litellm.acompletion(**self.build_payload(request))
        return self.parse_response(resp)
