"""AnthropicMessagesProvider — the Anthropic Messages API client.

Capability parity: the reference ships ``AnthropicModelClient``
(/root/reference/CHANGELOG.md:455).  `POST {base_url}/v1/messages` with
`x-api-key` + `anthropic-version` headers; tool calls arrive as
``tool_use`` content blocks and tool results go back as ``tool_result``
blocks inside a *user* message — a different wire shape from both
chat-completions and the Responses API, mapped here on both edges.
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
    ToolReturnPart,
    Usage,
)
from ..providers import HttpProvider
from ..schemas import EngineRequest
from ..streaming import ChunkSink

__all__ = ["AnthropicMessagesProvider"]

API_VERSION = "2023-06-01"


class AnthropicMessagesProvider(HttpProvider):
    """Non-streaming Messages API client behind the Provider port."""

    def __init__(self, base_url: str = "https://api.anthropic.com",
                 api_key: str = "", model: str = "claude-sonnet-4-20250514",
                 transport: Any = None):
        headers = {"anthropic-version": API_VERSION}
        if api_key:
            headers["x-api-key"] = api_key
        super().__init__(base_url=base_url, headers=headers)
        self.model = model
        self._transport = transport

    # ------------------------------------------------------------- payload
    def build_payload(self, request: EngineRequest) -> dict[str, Any]:
        system: list[str] = []
        msgs: list[dict[str, Any]] = []

        def push(role: str, blocks: list[dict[str, Any]]) -> None:
            # the API requires strict user/assistant alternation-friendly
            # form; adjacent same-role messages merge into one
            if msgs and msgs[-1]["role"] == role:
                msgs[-1]["content"].extend(blocks)
            else:
                msgs.append({"role": role, "content": blocks})

        for m in request.messages:
            text = " ".join(p.text for p in m.parts if isinstance(p, TextPart))
            if m.role == "system":
                if text:
                    system.append(text)
                continue
            if m.role == "tool":
                blocks = [{"type": "tool_result", "tool_use_id": p.call_id,
                           "content": str(p.content),
                           **({"is_error": True} if p.is_error else {})}
                          for p in m.parts if isinstance(p, ToolReturnPart)]
                if not blocks and text:
                    blocks = [{"type": "text", "text": text}]
                push("user", blocks)
                continue
            blocks = []
            if text:
                blocks.append({"type": "text", "text": text})
            for p in m.parts:
                if isinstance(p, ToolCallPart):
                    blocks.append({"type": "tool_use", "id": p.call_id,
                                   "name": p.tool_name, "input": p.arguments})
            if blocks:
                push("assistant" if m.role == "assistant" else "user", blocks)
        s = request.settings
        payload: dict[str, Any] = {
            "model": request.model_name or self.model,
            "max_tokens": s.max_tokens,
            "temperature": s.temperature,
            "top_p": s.top_p,
            "messages": msgs,
        }
        if system:
            payload["system"] = "\n\n".join(system)
        if s.stop:
            payload["stop_sequences"] = list(s.stop)
        if request.tools:
            payload["tools"] = [{"name": t.name, "description": t.description,
                                 "input_schema": t.parameters}
                                for t in request.tools]
            if request.tool_choice == "required":
                payload["tool_choice"] = {"type": "any"}
            elif request.tool_choice:
                payload["tool_choice"] = {"type": "tool",
                                          "name": request.tool_choice}
        return payload

    # ------------------------------------------------------------ response
    def parse_response(self, data: dict[str, Any],
                       sink: ChunkSink) -> AssistantMessage:
        parts: list[Part] = []
        has_calls = False
        for block in data.get("content", ()):
            if block.get("type") == "text" and block.get("text"):
                parts.append(TextPart(block["text"]))
            elif block.get("type") == "tool_use":
                has_calls = True
                parts.append(ToolCallPart(call_id=block.get("id", ""),
                                          tool_name=block.get("name", ""),
                                          arguments=dict(block.get("input")
                                                         or {})))
        u = data.get("usage") or {}
        stop = data.get("stop_reason")
        finish = ("tool_calls" if has_calls or stop == "tool_use" else
                  "length" if stop == "max_tokens" else "stop")
        facts = ProviderFacts(
            model_name=data.get("model", self.model),
            usage=Usage(prompt_tokens=u.get("input_tokens", 0),
                        completion_tokens=u.get("output_tokens", 0)),
            finish_reason=finish)
        return AssistantMessage(tuple(parts) or (TextPart(""),), facts=facts)

    # ---------------------------------------------------------------- port
    async def complete(self, request: EngineRequest,
                       sink: ChunkSink) -> AssistantMessage:
        import httpx  # in-function: the core stays import-light

        url = f"{self.base_url.rstrip('/')}/v1/messages"
        kw: dict[str, Any] = {"timeout": self.timeout_s}
        if self._transport is not None:
            kw["transport"] = self._transport
        async with httpx.AsyncClient(**kw) as client:
            resp = await client.post(url, json=self.build_payload(request),
                                     headers=self.headers)
            resp.raise_for_status()
            return self.parse_response(resp.json(), sink)
