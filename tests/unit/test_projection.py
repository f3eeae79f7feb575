"""Unit: agent-POV projection + transfer seeds (the bug-prone part of the
reference — attribution must hold in multi-agent channels)."""

from calfkit_amd._projection import project_for_model, project_for_transfer
from calfkit_amd.messages import (
    AssistantMessage,
    SystemMessage,
    TextPart,
    ToolCallPart,
    ToolMessage,
    ToolReturnPart,
    UserMessage,
)


def _thread():
    return (
        SystemMessage((TextPart("stored system"),)),
        UserMessage((TextPart("question"),)),
        AssistantMessage((TextPart("let me check"),
                          ToolCallPart("c1", "lookup", {})), author="alpha"),
        ToolMessage((ToolReturnPart("c1", "lookup", "data42"),)),
        AssistantMessage((TextPart("alpha says hi"),), author="alpha"),
        AssistantMessage((TextPart("beta interjects"),), author="beta"),
    )


def test_own_pov_keeps_own_turns():
    out = project_for_model(_thread(), agent_name="alpha", system_prompt="SP")
    assert out[0].role == "system" and out[0].parts[0].text == "SP"
    roles = [m.role for m in out]
    # own assistant turns survive; beta's turn became attributed user input
    assert roles.count("assistant") == 2
    last = out[-1]
    assert last.role == "user" and "[beta]" in last.parts[0].text


def test_foreign_pov_sees_alpha_as_input():
    out = project_for_model(_thread(), agent_name="beta", system_prompt="SP")
    assistants = [m for m in out if m.role == "assistant"]
    assert len(assistants) == 1  # only beta's own
    # alpha's tool interaction is attributed, not raw tool plumbing
    texts = " ".join(p.text for m in out if m.role == "user" for p in m.parts)
    assert "[alpha]" in texts
    assert "data42" in texts  # orphaned tool return surfaced as text
    assert not any(m.role == "tool" for m in out)


def test_open_calls_survive_own_pov():
    thread = (
        UserMessage((TextPart("q"),)),
        AssistantMessage((ToolCallPart("c9", "slow_tool", {}),), author="a"),
    )
    out = project_for_model(thread, agent_name="a", system_prompt="s")
    assert out[-1].tool_calls[0].call_id == "c9"


def test_transfer_seed_attributes_and_notices():
    seed = project_for_transfer(_thread(), source="alpha", target="gamma",
                                reason="expertise")
    assert all(m.role in ("user",) for m in seed)
    final = seed[-1].parts[0].text
    assert "alpha handed this conversation to gamma" in final
    assert "expertise" in final
    texts = " ".join(p.text for m in seed for p in m.parts)
    assert "[alpha]" in texts and "data42" in texts
    assert "stored system" not in texts  # system prompts never transfer
