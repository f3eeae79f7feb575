"""Unit: refs, selection algebra, tool schema derivation, Agent parsing."""

from dataclasses import dataclass
from typing import Literal, Optional

import pytest

from calfkit_amd import Agent, ToolError, agent_tool, tool
from calfkit_amd.context import ToolContext
from calfkit_amd.definitions.refs import AgentRef, validate_node_name
from calfkit_amd.definitions.selections import (
    Handoff,
    Messaging,
    Tools,
    parse_tool_selection,
    resolve_selection,
)
from calfkit_amd.errors import SelectionError
from calfkit_amd.ports import Advert, DirectoryView
from calfkit_amd.testing import FakeProvider


def test_node_name_validation():
    validate_node_name("good-name_2")
    with pytest.raises(ValueError):
        validate_node_name("bad name")
    with pytest.raises(ValueError):
        AgentRef("0starts-with-digit")


# ------------------------------------------------------------------- tools

@tool
def weather(city: str, units: Literal["c", "f"] = "c",
            days: Optional[int] = None) -> str:
    """Get the weather.

    Extra doc lines are not part of the description."""
    return f"{city}:{units}:{days}"


def test_schema_derivation():
    s = weather.schema
    assert s.name == "weather"
    assert s.description == "Get the weather."
    props = s.parameters["properties"]
    assert props["city"] == {"type": "string"}
    assert props["units"] == {"enum": ["c", "f"]}
    assert props["days"]["anyOf"][1] == {"type": "null"}
    assert s.parameters["required"] == ["city"]


async def test_invoke_coerces_and_validates():
    assert await weather.invoke({"city": "x"}) == "x:c:None"
    assert await weather.invoke({"city": "x", "days": 3}) == "x:c:3"
    with pytest.raises(ToolError):
        await weather.invoke({})  # missing required
    with pytest.raises(ToolError):
        await weather.invoke({"city": "x", "bogus": 1})  # unknown
    with pytest.raises(ToolError):
        await weather.invoke({"city": 42})  # wrong type


@dataclass
class Point:
    x: int
    y: int


@agent_tool
async def shift(p: Point, by: int, ctx: ToolContext) -> dict:
    """Shift a point."""
    return {"x": p.x + by, "caller": ctx.caller}


async def test_dataclass_args_and_context_injection():
    out = await shift.invoke({"p": {"x": 1, "y": 2}, "by": 5},
                             ToolContext(tool_name="shift", call_id="c",
                                         caller="alice"))
    assert out == {"x": 6, "caller": "alice"}
    assert "p" in shift.schema.parameters["properties"]
    assert "ctx" not in shift.schema.parameters["properties"]


# --------------------------------------------------------------- selections

def _view():
    return DirectoryView(adverts={
        "weather": Advert(node="weather", kind="tool", inbox="calf.inbox.weather",
                          capabilities={"parameters": {"type": "object",
                                                        "properties": {}}}),
        "box": Advert(node="box", kind="toolbox", inbox="calf.inbox.box",
                      capabilities={"tools": [
                          {"name": "read", "description": "", "parameters": {}}]}),
        "alpha": Advert(node="alpha", kind="agent", inbox="calf.inbox.alpha"),
        "beta": Advert(node="beta", kind="agent", inbox="calf.inbox.beta"),
    })


def test_parse_duplicates_rejected():
    with pytest.raises(SelectionError):
        parse_tool_selection(["weather", "weather"])


def test_resolve_static_and_discovery():
    surface = resolve_selection([Tools(names=("weather",))], _view())
    assert [b.schema.name for b in surface.tools] == ["weather"]
    surface = resolve_selection([Tools(discover=True)], _view())
    assert {b.schema.name for b in surface.tools} == {"weather"}
    # toolbox tools are namespaced
    from calfkit_amd.definitions.selections import Toolboxes

    surface = resolve_selection([Toolboxes(names=("box",))], _view())
    assert [b.schema.name for b in surface.tools] == ["box.read"]


def test_resolve_missing_strict_raises():
    with pytest.raises(SelectionError):
        resolve_selection([Tools(names=("nope",))], _view())
    assert resolve_selection([Tools(names=("nope",))], _view(),
                             strict=False).tools == []


def test_agent_ctor_parses_everything():
    a = Agent(
        name="planner", description="plans",
        system_prompt="be useful", model_client=FakeProvider(["x"]),
        tools=[weather], peers=[Messaging(discover=True),
                                 Handoff(names=("alpha",))],
        emits=["metric"], listens_to=["tick"])
    assert a.tool_selections[0].names == ("weather",)
    assert a.messaging_selections[0].discover
    assert a.handoff_selections[0].names == ("alpha",)
    assert a.emit_refs[0].name == "metric"
    assert a.listen_refs[0].name == "tick"
    from calfkit_amd.context import TurnContext

    sp = a.render_system_prompt(TurnContext(agent_name="planner", thread_id="t"))
    assert "planner" in sp and "be useful" in sp


def test_agent_instructions_decorator():
    a = Agent(name="deco", model_client=FakeProvider(["x"]))

    @a.instructions
    def extra(ctx):
        return f"thread={ctx.thread_id}"

    from calfkit_amd.context import TurnContext

    assert "thread=T9" in a.render_system_prompt(
        TurnContext(agent_name="deco", thread_id="T9"))
