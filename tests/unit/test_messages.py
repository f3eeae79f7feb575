"""Unit: messages vocabulary + wire round-trip."""

from calfkit_amd.messages import (
    AssistantMessage,
    ProviderFacts,
    SystemMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    Usage,
    UserMessage,
    message_from_wire,
    message_to_wire,
    parts_to_text,
    thread_from_wire,
    thread_to_wire,
    value_to_parts,
)


def test_value_to_parts():
    assert value_to_parts("hi") == (TextPart("hi"),)
    assert value_to_parts(None) == ()
    assert value_to_parts(["a", "b"]) == (TextPart("a"), TextPart("b"))
    (p,) = value_to_parts({"k": 1})
    assert '"k"' in p.text


def test_assistant_helpers():
    m = AssistantMessage((TextPart("x"), ToolCallPart("c1", "t", {"a": 1})))
    assert m.text == "x"
    assert m.tool_calls[0].tool_name == "t"


def test_wire_round_trip_all_roles():
    msgs = [
        SystemMessage((TextPart("sys"),)),
        UserMessage((TextPart("hi"),), author="alice"),
        AssistantMessage(
            (TextPart("ok"), ToolCallPart("c1", "weather", {"city": "x"})),
            author="bot",
            facts=ProviderFacts(model_name="m", usage=Usage(10, 5),
                                prefill_ms=1.5, decode_ms=8.25)),
        ToolMessage((ToolReturnPart("c1", "weather", "sunny"),)),
    ]
    wired = thread_to_wire(msgs)
    back = thread_from_wire(wired)
    assert back[1].author == "alice"
    assert back[2].facts.usage.total_tokens == 15
    assert back[2].facts.decode_ms == 8.25
    assert back[2].tool_calls[0].arguments == {"city": "x"}
    assert back[3].parts[0].content == "sunny"
    # byte-stability: encoding the decoded form is identical
    assert thread_to_wire(list(back)) == wired


def test_unknown_role_rejected():
    import pytest

    with pytest.raises(ValueError):
        message_from_wire({"role": "alien", "parts": []})


def test_parts_to_text_skips_non_text():
    m = message_to_wire(AssistantMessage((ToolCallPart("c", "t", {}),)))
    assert parts_to_text(message_from_wire(m).parts) == ""
