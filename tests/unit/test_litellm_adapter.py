"""LiteLLMProvider: sole-litellm-importer adapter (reference
adapters/litellm.py:1).  The SDK is absent in this image, so the tests
cover the clear-error path, the chat-completions mapping, and a full
complete() round trip against a stubbed litellm module."""

import json
import sys
import types
from dataclasses import dataclass, field
from typing import Any

import pytest

from calfkit_amd.adapters.litellm import LiteLLMProvider
from calfkit_amd.messages import (
    AssistantMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    UserMessage,
)
from calfkit_amd.schemas import EngineRequest, ModelSettings, ToolSchema
from calfkit_amd.streaming import NullSink


def _request(**kw) -> EngineRequest:
    defaults = dict(
        model_name="gpt-test",
        messages=(UserMessage((TextPart("hi"),)),),
        settings=ModelSettings(max_tokens=16, temperature=0.5, seed=11),
    )
    defaults.update(kw)
    return EngineRequest(**defaults)


async def test_missing_sdk_raises_clearly():
    p = LiteLLMProvider("gpt-test")
    with pytest.raises(RuntimeError, match="litellm"):
        await p.complete(_request(), NullSink())


def test_payload_maps_messages_settings_and_tools():
    p = LiteLLMProvider("fallback-model", api_base="http://box:4000")
    tool = ToolSchema(name="add", description="Add.", parameters={
        "type": "object", "properties": {"a": {"type": "integer"}},
        "required": ["a"]})
    call = ToolCallPart(call_id="c1", tool_name="add", arguments={"a": 1})
    req = _request(
        messages=(UserMessage((TextPart("q"),)),
                  AssistantMessage((call,)),
                  ToolMessage((TextPart("2"),))),
        tools=(tool,),
        settings=ModelSettings(max_tokens=8, temperature=0.0, seed=3,
                               stop=("END",)),
    )
    payload = p.build_payload(req)
    assert payload["model"] == "gpt-test"          # request wins over default
    assert payload["api_base"] == "http://box:4000"  # ctor kwargs fan out
    assert payload["seed"] == 3 and payload["stop"] == ["END"]
    roles = [m["role"] for m in payload["messages"]]
    assert roles == ["user", "assistant", "tool"]
    tc = payload["messages"][1]["tool_calls"][0]
    assert tc["function"]["name"] == "add"
    assert json.loads(tc["function"]["arguments"]) == {"a": 1}
    assert payload["tools"][0]["function"]["name"] == "add"


def _fake_response():
    @dataclass
    class Fn:
        name: str
        arguments: str

    @dataclass
    class TC:
        id: str
        function: Fn

    @dataclass
    class Msg:
        content: str
        tool_calls: list = field(default_factory=list)

    @dataclass
    class Choice:
        message: Msg
        finish_reason: str = "tool_calls"

    @dataclass
    class Use:
        prompt_tokens: int = 7
        completion_tokens: int = 5

    @dataclass
    class Resp:
        choices: list
        usage: Use
        model: str = "gpt-test"

    msg = Msg(content="thinking", tool_calls=[
        TC(id="c9", function=Fn(name="add", arguments='{"a": 4}'))])
    return Resp(choices=[Choice(message=msg)], usage=Use())


def test_parse_response_builds_assistant_message():
    m = LiteLLMProvider.parse_response(_fake_response())
    assert isinstance(m, AssistantMessage)
    assert m.parts[0].text == "thinking"
    assert m.tool_calls[0].tool_name == "add"
    assert m.tool_calls[0].arguments == {"a": 4}
    assert m.facts.finish_reason == "tool_calls"
    assert m.facts.usage.total_tokens == 12


async def test_complete_round_trip_with_stubbed_sdk(monkeypatch):
    seen: dict[str, Any] = {}

    async def acompletion(**kw):
        seen.update(kw)
        return _fake_response()

    stub = types.ModuleType("litellm")
    stub.acompletion = acompletion
    monkeypatch.setitem(sys.modules, "litellm", stub)
    p = LiteLLMProvider("gpt-test")
    m = await p.complete(_request(), NullSink())
    assert seen["model"] == "gpt-test" and seen["max_tokens"] == 16
    assert m.tool_calls[0].call_id == "c9"
