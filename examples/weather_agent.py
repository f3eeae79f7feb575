"""The quickstart: a weather agent + one tool (BASELINE config 1).

Run it on a local mesh:

    ck dev run examples/weather_agent.py &    # starts a broker if needed
    ck dev chat --agent weather_agent

Or swap the echo provider for the real GPU engine:

    weather_agent.model_client = LocalModelClient("llama-3-8b")
"""

from calfkit_amd import Agent, agent_tool
from calfkit_amd.testing import EchoProvider


@agent_tool
def get_weather(city: str) -> str:
    """Get the current weather for a city."""
    return f"It is sunny in {city} today, 27 degrees."


weather_agent = Agent(
    name="weather_agent",
    description="Answers weather questions using the get_weather tool.",
    system_prompt="You answer weather questions. Use the tool.",
    tools=[get_weather],
    model_client=EchoProvider(use_tools=True),
)
