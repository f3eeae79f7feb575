"""OpenAI-compatible chat-completions provider — the concrete HttpProvider.

Capability parity: the reference ships OpenAI/Anthropic remote clients
(/root/reference/README.md:49, /root/reference/CHANGELOG.md:448,455).  The
MI355X build's first-class provider is the in-process LocalModelClient;
this adapter exists so the same Agent can point at ANY OpenAI-compatible
endpoint (vLLM, llama.cpp server, OpenAI itself) — kept strictly off the
hot path.

Implements SSE streaming with Accumulator-backed delta emission (text and
tool-call argument fragments), plus a non-streaming fallback.  httpx is
imported in-function (the providers layer is the sole HTTP importer); a
test can inject an ASGI transport to run against a scripted app.
"""

from __future__ import annotations

import json
from typing import Any

from ..messages import (
    AssistantMessage,
    ProviderFacts,
    TextPart,
    ToolCallPart,
    Usage,
)
from ..providers import Accumulator, HttpProvider
from ..schemas import EngineRequest
from ..streaming import ChunkSink

__all__ = ["OpenAIChatProvider", "OpenAIResponsesProvider"]


class OpenAIChatProvider(HttpProvider):
    """`POST {base_url}/chat/completions` with `stream=true`."""

    def __init__(self, base_url: str = "https://api.openai.com/v1",
                 api_key: str = "", model: str = "gpt-4o-mini",
                 stream: bool = True, transport: Any = None):
        headers = {"Authorization": f"Bearer {api_key}"} if api_key else {}
        super().__init__(base_url=base_url, headers=headers)
        self.model = model
        self.stream = stream
        self._transport = transport  # tests: httpx.ASGITransport

    # ------------------------------------------------------------- payload
    def build_payload(self, request: EngineRequest) -> dict[str, Any]:
        msgs = []
        for m in request.messages:
            role = m.role
            text = " ".join(p.text for p in m.parts
                            if isinstance(p, TextPart))
            entry: dict[str, Any] = {"role": role, "content": text}
            calls = [p for p in m.parts if isinstance(p, ToolCallPart)]
            if calls:
                entry["tool_calls"] = [
                    {"id": c.call_id, "type": "function",
                     "function": {"name": c.tool_name,
                                  "arguments": json.dumps(c.arguments)}}
                    for c in calls]
            msgs.append(entry)
        s = request.settings
        payload: dict[str, Any] = {
            "model": request.model_name or self.model,
            "messages": msgs,
            "max_tokens": s.max_tokens,
            "temperature": s.temperature,
            "top_p": s.top_p,
            "stream": self.stream,
        }
        if s.seed is not None:
            payload["seed"] = s.seed
        if s.stop:
            payload["stop"] = list(s.stop)
        if request.tools:
            payload["tools"] = [
                {"type": "function", "function": t.to_wire()}
                for t in request.tools]
            payload["parallel_tool_calls"] = s.parallel_tool_calls
            if request.tool_choice == "required":
                payload["tool_choice"] = "required"
            elif request.tool_choice:
                payload["tool_choice"] = {
                    "type": "function",
                    "function": {"name": request.tool_choice}}
        return payload

    # ------------------------------------------------------------ non-stream
    def parse_response(self, data: dict[str, Any],
                       sink: ChunkSink) -> AssistantMessage:
        choice = data["choices"][0]
        msg = choice["message"]
        parts: list[Any] = []
        if msg.get("content"):
            parts.append(TextPart(msg["content"]))
        for tc in msg.get("tool_calls") or ():
            fn = tc.get("function", {})
            parts.append(ToolCallPart(
                call_id=tc.get("id", ""), tool_name=fn.get("name", ""),
                arguments=json.loads(fn.get("arguments") or "{}")))
        u = data.get("usage") or {}
        facts = ProviderFacts(
            model_name=data.get("model", self.model),
            usage=Usage(prompt_tokens=u.get("prompt_tokens", 0),
                        completion_tokens=u.get("completion_tokens", 0)),
            finish_reason=_finish(choice.get("finish_reason"),
                                  bool(msg.get("tool_calls"))))
        return AssistantMessage(tuple(parts) or (TextPart(""),), facts=facts)

    # --------------------------------------------------------------- stream
    async def complete(self, request: EngineRequest,
                       sink: ChunkSink) -> AssistantMessage:
        import httpx  # in-function: the core stays import-light

        url = f"{self.base_url.rstrip('/')}/chat/completions"
        kw: dict[str, Any] = {"timeout": self.timeout_s}
        if self._transport is not None:
            kw["transport"] = self._transport
        async with httpx.AsyncClient(**kw) as client:
            if not self.stream:
                resp = await client.post(url, json=self.build_payload(request),
                                         headers=self.headers)
                resp.raise_for_status()
                return self.parse_response(resp.json(), sink)

            acc = Accumulator(sink)
            pending: dict[int, dict[str, Any]] = {}
            finish_reason = "stop"
            model_name = self.model
            usage = Usage()
            async with client.stream(
                    "POST", url, json=self.build_payload(request),
                    headers=self.headers) as resp:
                resp.raise_for_status()
                async for line in resp.aiter_lines():
                    if not line.startswith("data:"):
                        continue
                    body = line[5:].strip()
                    if body == "[DONE]":
                        break
                    evt = json.loads(body)
                    model_name = evt.get("model", model_name)
                    if evt.get("usage"):
                        u = evt["usage"]
                        usage = Usage(u.get("prompt_tokens", 0),
                                      u.get("completion_tokens", 0))
                    for choice in evt.get("choices", ()):
                        if choice.get("finish_reason"):
                            finish_reason = choice["finish_reason"]
                        delta = choice.get("delta") or {}
                        if delta.get("content"):
                            acc.text_delta(delta["content"])
                        for tc in delta.get("tool_calls") or ():
                            slot = pending.setdefault(
                                tc.get("index", 0),
                                {"id": "", "name": "", "args": []})
                            if tc.get("id"):
                                slot["id"] = tc["id"]
                            fn = tc.get("function") or {}
                            if fn.get("name"):
                                slot["name"] = fn["name"]
                            if fn.get("arguments"):
                                slot["args"].append(fn["arguments"])
            for _, slot in sorted(pending.items()):
                acc.tool_call(ToolCallPart(
                    call_id=slot["id"], tool_name=slot["name"],
                    arguments=json.loads("".join(slot["args"]) or "{}")))
            facts = ProviderFacts(
                model_name=model_name, usage=usage,
                finish_reason=_finish(finish_reason, bool(pending)))
            return acc.finish(facts)


def _finish(reason: str | None, has_calls: bool) -> str:
    if has_calls or reason == "tool_calls":
        return "tool_calls"
    return {"length": "length", "stop": "stop"}.get(reason or "stop", "stop")


class OpenAIResponsesProvider(HttpProvider):
    """OpenAI Responses API client — `POST {base_url}/responses`.

    Capability parity: the reference names ``OpenAIResponsesModelClient``
    (/root/reference/README.md:49; Responses API support
    /root/reference/CHANGELOG.md:448).  Non-streaming: the Responses API
    returns typed output ITEMS (message / function_call) rather than
    chat-completions choices, so the mapping differs from
    OpenAIChatProvider on both edges.
    """

    def __init__(self, base_url: str = "https://api.openai.com/v1",
                 api_key: str = "", model: str = "gpt-4o-mini",
                 transport: Any = None):
        headers = {"Authorization": f"Bearer {api_key}"} if api_key else {}
        super().__init__(base_url=base_url, headers=headers)
        self.model = model
        self._transport = transport

    def build_payload(self, request: EngineRequest) -> dict[str, Any]:
        items: list[dict[str, Any]] = []
        for m in request.messages:
            from ..messages import ToolReturnPart

            returns = [p for p in m.parts if isinstance(p, ToolReturnPart)]
            calls = [p for p in m.parts if isinstance(p, ToolCallPart)]
            text = " ".join(p.text for p in m.parts if isinstance(p, TextPart))
            if m.role == "tool" or returns:
                for r in returns:
                    items.append({"type": "function_call_output",
                                  "call_id": r.call_id,
                                  "output": str(r.content)})
                if m.role == "tool" and not returns and text:
                    items.append({"type": "function_call_output",
                                  "call_id": "", "output": text})
                continue
            for c in calls:
                items.append({"type": "function_call", "call_id": c.call_id,
                              "name": c.tool_name,
                              "arguments": json.dumps(c.arguments)})
            if text or not calls:
                ctype = ("output_text" if m.role == "assistant"
                         else "input_text")
                items.append({"role": m.role,
                              "content": [{"type": ctype, "text": text}]})
        s = request.settings
        payload: dict[str, Any] = {
            "model": request.model_name or self.model,
            "input": items,
            "max_output_tokens": s.max_tokens,
            "temperature": s.temperature,
            "top_p": s.top_p,
        }
        if request.tools:
            # Responses API tools are FLAT (no nested "function" object)
            payload["tools"] = [{"type": "function", **t.to_wire()}
                                for t in request.tools]
            if request.tool_choice == "required":
                payload["tool_choice"] = "required"
            elif request.tool_choice:
                payload["tool_choice"] = {"type": "function",
                                          "name": request.tool_choice}
        return payload

    def parse_response(self, data: dict[str, Any],
                       sink: ChunkSink) -> AssistantMessage:
        parts: list[Any] = []
        has_calls = False
        for item in data.get("output", ()):
            if item.get("type") == "message":
                for c in item.get("content", ()):
                    if c.get("type") == "output_text" and c.get("text"):
                        parts.append(TextPart(c["text"]))
            elif item.get("type") == "function_call":
                has_calls = True
                try:
                    args = json.loads(item.get("arguments") or "{}")
                except json.JSONDecodeError:
                    args = {"_raw": item.get("arguments")}
                parts.append(ToolCallPart(call_id=item.get("call_id", ""),
                                          tool_name=item.get("name", ""),
                                          arguments=args))
        u = data.get("usage") or {}
        facts = ProviderFacts(
            model_name=data.get("model", self.model),
            usage=Usage(prompt_tokens=u.get("input_tokens", 0),
                        completion_tokens=u.get("output_tokens", 0)),
            finish_reason="tool_calls" if has_calls else (
                "length" if data.get("incomplete_details") else "stop"))
        return AssistantMessage(tuple(parts) or (TextPart(""),), facts=facts)

    async def complete(self, request: EngineRequest,
                       sink: ChunkSink) -> AssistantMessage:
        import httpx

        url = f"{self.base_url.rstrip('/')}/responses"
        kw: dict[str, Any] = {"timeout": self.timeout_s}
        if self._transport is not None:
            kw["transport"] = self._transport
        async with httpx.AsyncClient(**kw) as client:
            resp = await client.post(url, json=self.build_payload(request),
                                     headers=self.headers)
            resp.raise_for_status()
            return self.parse_response(resp.json(), sink)
