"""OpenAIChatProvider against a scripted OpenAI-compatible server
(VERDICT r01 #9): SSE streaming accumulation, tool-call fragment
reassembly, non-stream fallback, and a full Agent turn over the mesh with
the remote provider plugged in behind the same Provider port."""

import json

import httpx
import pytest

from calfkit_amd.adapters.openai import OpenAIChatProvider
from calfkit_amd.messages import ToolCallPart, UserMessage, TextPart
from calfkit_amd.schemas import EngineRequest, ModelSettings, ToolSchema
from calfkit_amd.streaming import ChunkSink


class _Sink(ChunkSink):
    def __init__(self):
        self.chunks = []

    def emit(self, chunk):
        self.chunks.append(chunk)


def _sse(events):
    out = b""
    for e in events:
        out += b"data: " + json.dumps(e).encode() + b"\n\n"
    out += b"data: [DONE]\n\n"
    return out


def _scripted_app(events, capture):
    async def app(scope, receive, send):
        assert scope["type"] == "http"
        assert scope["path"].endswith("/chat/completions")
        body = b""
        while True:
            msg = await receive()
            body += msg.get("body", b"")
            if not msg.get("more_body"):
                break
        capture.append(json.loads(body))
        await send({"type": "http.response.start", "status": 200,
                    "headers": [(b"content-type", b"text/event-stream")]})
        await send({"type": "http.response.body", "body": _sse(events)})
    return app


def _provider(events, capture, **kw):
    return OpenAIChatProvider(
        base_url="http://testserver/v1", api_key="sk-test",
        transport=httpx.ASGITransport(app=_scripted_app(events, capture)),
        **kw)


async def test_streaming_text_accumulates_and_reports_facts():
    capture = []
    events = [
        {"model": "scripted-1", "choices": [{"delta": {"content": "Hel"}}]},
        {"choices": [{"delta": {"content": "lo!"}}]},
        {"choices": [{"delta": {}, "finish_reason": "stop"}],
         "usage": {"prompt_tokens": 7, "completion_tokens": 2}},
    ]
    p = _provider(events, capture)
    sink = _Sink()
    msg = await p.complete(EngineRequest(
        messages=(UserMessage((TextPart("hi"),)),),
        settings=ModelSettings(max_tokens=16, temperature=0.5, seed=3)), sink)
    assert msg.text == "Hello!"
    assert msg.facts.model_name == "scripted-1"
    assert msg.facts.usage.prompt_tokens == 7
    assert [c.text for c in sink.chunks if c.event == "text_delta"] == \
        ["Hel", "lo!"]
    # the request carried the OpenAI wire shape + auth header semantics
    sent = capture[0]
    assert sent["messages"] == [{"role": "user", "content": "hi"}]
    assert sent["stream"] is True and sent["seed"] == 3


async def test_streaming_tool_call_fragments_reassemble():
    capture = []
    events = [
        {"choices": [{"delta": {"tool_calls": [
            {"index": 0, "id": "call_9",
             "function": {"name": "get_weather", "arguments": "{\"ci"}}]}}]},
        {"choices": [{"delta": {"tool_calls": [
            {"index": 0, "function": {"arguments": "ty\": \"SF\"}"}}]}}]},
        {"choices": [{"delta": {}, "finish_reason": "tool_calls"}]},
    ]
    p = _provider(events, capture)
    sink = _Sink()
    msg = await p.complete(EngineRequest(
        messages=(UserMessage((TextPart("weather?"),)),),
        tools=(ToolSchema(name="get_weather",
                          parameters={"type": "object", "properties": {
                              "city": {"type": "string"}}}),)), sink)
    assert msg.facts.finish_reason == "tool_calls"
    (call,) = msg.tool_calls
    assert call == ToolCallPart(call_id="call_9", tool_name="get_weather",
                                arguments={"city": "SF"})
    assert capture[0]["tools"][0]["function"]["name"] == "get_weather"


async def test_non_streaming_fallback():
    capture = []

    async def app(scope, receive, send):
        while True:
            msg = await receive()
            if not msg.get("more_body"):
                break
        await send({"type": "http.response.start", "status": 200,
                    "headers": [(b"content-type", b"application/json")]})
        await send({"type": "http.response.body", "body": json.dumps({
            "model": "scripted-1",
            "choices": [{"message": {"role": "assistant",
                                     "content": "plain"},
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 3, "completion_tokens": 1},
        }).encode()})

    p = OpenAIChatProvider(base_url="http://t/v1", stream=False,
                           transport=httpx.ASGITransport(app=app))
    msg = await p.complete(EngineRequest(
        messages=(UserMessage((TextPart("x"),)),)), _Sink())
    assert msg.text == "plain" and msg.facts.usage.completion_tokens == 1


async def test_agent_turn_with_remote_provider_on_the_mesh():
    """The remote provider slots behind the same Provider port: a full
    Client.execute -> Worker(Agent) turn with the scripted endpoint."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import (
        InMemoryBus,
        InMemoryDirectory,
        InProcessThreadStore,
    )

    capture = []
    events = [
        {"choices": [{"delta": {"content": "remote pong"},
                      "finish_reason": "stop"}]},
    ]
    agent = Agent(name="remoter", model_client=_provider(events, capture))
    bus, store, d = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    async with Worker(agent, bus=bus, store=store, directory=d):
        client = await Client.connect(bus, d)
        result = await client.execute("remoter", "ping", timeout=30)
        assert result.text == "remote pong"
    # the agent's system/user projection reached the remote payload
    assert capture[0]["messages"][-1]["content"] == "ping"
