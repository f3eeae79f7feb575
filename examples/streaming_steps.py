"""Intermediate-step streaming: watch an agent think in real time.

The caller subscribes to the thread's firehose BEFORE sending the turn
and prints every StreamChunk (text deltas, tool-call events) as the
agent produces them — "a real-time reflection of facts, never the
facts" (reference CHANGELOG.md:124, streaming example CHANGELOG.md:117).

    python examples/streaming_steps.py
"""

import asyncio

from calfkit_amd import Agent, Client, Worker, agent_tool
from calfkit_amd.adapters.memory import (
    InMemoryBus,
    InMemoryDirectory,
    InProcessThreadStore,
)
from calfkit_amd.testing import FakeProvider, call_tool, reply


@agent_tool
def lookup(city: str) -> str:
    """Look up a city's weather."""
    return f"sunny in {city}"


narrator = Agent(
    name="narrator",
    system_prompt="Answer with the tool, then summarize.",
    tools=[lookup],
    model_client=FakeProvider([
        call_tool("lookup", city="Osaka"),
        reply("It is sunny in Osaka today."),
    ]),
)


async def main() -> None:
    bus, store, d = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    async with Worker(narrator, bus=bus, store=store, directory=d):
        client = await Client.connect(bus, d)
        thread_id = "demo-thread"

        async def watch() -> None:
            async for chunk in client.firehose(thread_id):
                print(f"  [{chunk.event}] {chunk.text or chunk.data or ''}")

        watcher = asyncio.create_task(watch())
        result = await client.execute("narrator", "weather in Osaka?",
                                      thread_id=thread_id, timeout=30)
        print("final:", result.text)
        watcher.cancel()


if __name__ == "__main__":
    asyncio.run(main())
