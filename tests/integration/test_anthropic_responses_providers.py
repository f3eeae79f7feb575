"""AnthropicMessagesProvider + OpenAIResponsesProvider against scripted
ASGI endpoints (the reference names both clients: README.md:49,
CHANGELOG.md:448,455).  Each test drives the real HTTP client through an
in-process transport and checks both wire edges."""

import json

import pytest

httpx = pytest.importorskip("httpx")

from calfkit_amd.adapters.anthropic import AnthropicMessagesProvider
from calfkit_amd.adapters.openai import OpenAIResponsesProvider
from calfkit_amd.messages import (
    AssistantMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    SystemMessage,
    UserMessage,
)
from calfkit_amd.schemas import EngineRequest, ModelSettings, ToolSchema
from calfkit_amd.streaming import NullSink

TOOL = ToolSchema(name="add", description="Add.", parameters={
    "type": "object", "properties": {"a": {"type": "integer"}},
    "required": ["a"]})


def _request(messages) -> EngineRequest:
    return EngineRequest(messages=tuple(messages), tools=(TOOL,),
                        settings=ModelSettings(max_tokens=32,
                                               temperature=0.1, seed=1))


def _asgi_app(seen: dict, reply: dict):
    async def app(scope, receive, send):
        assert scope["type"] == "http"
        body = b""
        while True:
            ev = await receive()
            body += ev.get("body", b"")
            if not ev.get("more_body"):
                break
        seen["path"] = scope["path"]
        seen["headers"] = dict(
            (k.decode(), v.decode()) for k, v in scope["headers"])
        seen["payload"] = json.loads(body)
        data = json.dumps(reply).encode()
        await send({"type": "http.response.start", "status": 200,
                    "headers": [(b"content-type", b"application/json")]})
        await send({"type": "http.response.body", "body": data})
    return app


async def test_anthropic_messages_round_trip():
    seen: dict = {}
    reply = {
        "model": "claude-test", "stop_reason": "tool_use",
        "usage": {"input_tokens": 11, "output_tokens": 7},
        "content": [
            {"type": "text", "text": "let me add"},
            {"type": "tool_use", "id": "tu_1", "name": "add",
             "input": {"a": 4}},
        ]}
    app = _asgi_app(seen, reply)
    p = AnthropicMessagesProvider(base_url="http://test", api_key="k",
                                  model="claude-test",
                                  transport=httpx.ASGITransport(app=app))
    msgs = [
        SystemMessage((TextPart("be brief"),)),
        UserMessage((TextPart("add 4"),)),
        AssistantMessage((ToolCallPart(call_id="tu_0", tool_name="add",
                                       arguments={"a": 1}),)),
        ToolMessage((ToolReturnPart(call_id="tu_0", tool_name="add",
                                    content="1"),)),
    ]
    out = await p.complete(_request(msgs), NullSink())
    # request edge: headers + blocks shaped per the Messages API
    assert seen["path"] == "/v1/messages"
    assert seen["headers"]["x-api-key"] == "k"
    assert seen["headers"]["anthropic-version"]
    payload = seen["payload"]
    assert payload["system"] == "be brief"
    roles = [m["role"] for m in payload["messages"]]
    assert roles == ["user", "assistant", "user"]  # tool_result rides user
    assert payload["messages"][1]["content"][0]["type"] == "tool_use"
    assert payload["messages"][2]["content"][0]["type"] == "tool_result"
    assert payload["tools"][0]["input_schema"]["required"] == ["a"]
    # response edge
    assert out.parts[0].text == "let me add"
    assert out.tool_calls[0].call_id == "tu_1"
    assert out.tool_calls[0].arguments == {"a": 4}
    assert out.facts.finish_reason == "tool_calls"
    assert out.facts.usage.total_tokens == 18


async def test_openai_responses_round_trip():
    seen: dict = {}
    reply = {
        "model": "gpt-test",
        "usage": {"input_tokens": 9, "output_tokens": 5},
        "output": [
            {"type": "message", "content": [
                {"type": "output_text", "text": "done"}]},
            {"type": "function_call", "call_id": "fc_1", "name": "add",
             "arguments": "{\"a\": 2}"},
        ]}
    app = _asgi_app(seen, reply)
    p = OpenAIResponsesProvider(base_url="http://test", api_key="k",
                                model="gpt-test",
                                transport=httpx.ASGITransport(app=app))
    msgs = [
        UserMessage((TextPart("add 2"),)),
        AssistantMessage((ToolCallPart(call_id="fc_0", tool_name="add",
                                       arguments={"a": 1}),)),
        ToolMessage((ToolReturnPart(call_id="fc_0", tool_name="add",
                                    content="1"),)),
    ]
    out = await p.complete(_request(msgs), NullSink())
    assert seen["path"] == "/responses"
    payload = seen["payload"]
    kinds = [i.get("type") or i["role"] for i in payload["input"]]
    assert kinds == ["user", "function_call", "function_call_output"]
    assert payload["tools"][0] == {"type": "function", **TOOL.to_wire()}
    assert payload["max_output_tokens"] == 32
    assert out.parts[0].text == "done"
    assert out.tool_calls[0].tool_name == "add"
    assert out.facts.finish_reason == "tool_calls"


def test_tool_choice_maps_per_provider_dialect():
    """tool_choice='required' and a named tool map to each API's shape."""
    from calfkit_amd.adapters.litellm import LiteLLMProvider
    from calfkit_amd.adapters.openai import OpenAIChatProvider

    msgs = (UserMessage((TextPart("q"),)),)

    def req(choice):
        return EngineRequest(messages=msgs, tools=(TOOL,),
                             tool_choice=choice)

    chat = OpenAIChatProvider()
    assert chat.build_payload(req("required"))["tool_choice"] == "required"
    assert chat.build_payload(req("add"))["tool_choice"] == {
        "type": "function", "function": {"name": "add"}}
    assert "tool_choice" not in chat.build_payload(req(""))

    responses = OpenAIResponsesProvider()
    assert responses.build_payload(req("required"))["tool_choice"] == "required"
    assert responses.build_payload(req("add"))["tool_choice"] == {
        "type": "function", "name": "add"}

    anthropic = AnthropicMessagesProvider()
    assert anthropic.build_payload(req("required"))["tool_choice"] == {
        "type": "any"}
    assert anthropic.build_payload(req("add"))["tool_choice"] == {
        "type": "tool", "name": "add"}

    lite = LiteLLMProvider("m")
    assert lite.build_payload(req("required"))["tool_choice"] == "required"
