"""A three-agent team with tools, messaging and handoff — run it with:

    ck dev run examples/research_team.py --detach
    ck dev chat --agent frontdesk
"""

from calfkit_amd import Agent, agent_tool
from calfkit_amd.definitions.selections import Handoff, Messaging
from calfkit_amd.testing import EchoProvider


@agent_tool
def search(query: str) -> str:
    """Search the corpus for a query."""
    return f"results for {query!r}: [doc1, doc2]"


@agent_tool
def summarize(text: str, max_words: int = 50) -> str:
    """Summarize text to at most max_words words."""
    return " ".join(text.split()[:max_words])


# swap EchoProvider for LocalModelClient("llama-3-8b") on a GPU box
frontdesk = Agent(
    name="frontdesk",
    description="Routes research questions to the right specialist.",
    system_prompt="Hand research questions to the researcher.",
    peers=[Handoff(names=("researcher",)), Messaging(discover=True)],
    model_client=EchoProvider(),
)

researcher = Agent(
    name="researcher",
    description="Digs through the corpus and asks the critic to check.",
    tools=[search, summarize],
    peers=[Messaging(names=("critic",))],
    model_client=EchoProvider(use_tools=True),
)

critic = Agent(
    name="critic",
    description="Reviews research summaries for errors.",
    model_client=EchoProvider(),
)
