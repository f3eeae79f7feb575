"""Typed structured output + engine-sampled tool calls, end to end.

The agent advertises two tools and a typed output: on GPU the engine's
grammar-constrained sampler decides the calls (no scripting anywhere),
and the closing turn materializes the `Report` schema from the sampled
stream as a `final_result` union the policy validates.

    ck dev run examples/typed_research.py:analyst --detach
    ck dev chat
"""

from dataclasses import dataclass

from calfkit_amd import Agent, agent_tool
from calfkit_amd.providers import LocalModelClient
from calfkit_amd.schemas import ModelSettings


@agent_tool
def search(query: str) -> str:
    """Search the corpus for a query."""
    return f"three hits for {query}"


@agent_tool
def summarize(text: str, max_words: int) -> str:
    """Summarize text to at most max_words words."""
    return text[: max_words * 8]


@dataclass
class Report:
    topic: str
    confidence: float
    sources: int


analyst = Agent(
    name="analyst",
    system_prompt="Research the question with the tools, then report.",
    tools=[search, summarize],
    output_type=Report,
    model_client=LocalModelClient("llama-3-8b"),
    model_settings=ModelSettings(max_tokens=48, temperature=0.7, seed=7),
)
