#!/usr/bin/env python3
"""BASELINE config 1: quickstart weather_agent + get_weather tool, stub echo
provider, local mesh broker — pure plumbing (no GPU).

Measures agent turns/sec of the wire + step machine alone.
"""

from __future__ import annotations

import asyncio
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def run(turns: int | None = None, concurrency: int | None = None) -> None:
    turns = turns or int(os.environ.get("C1_TURNS", "200"))
    concurrency = concurrency or int(os.environ.get("C1_CONCURRENCY", "16"))
    from calfkit_amd import Agent, Client, Worker, agent_tool
    from calfkit_amd.adapters.memory import InProcessThreadStore
    from calfkit_amd.adapters.mesh import MeshBroker, connect_mesh
    from calfkit_amd.testing import EchoProvider

    @agent_tool
    def get_weather(city: str) -> str:
        """Get the current weather for a city."""
        return f"sunny in {city}"

    broker = await MeshBroker().start()
    bus_w, dir_w = await connect_mesh(broker.address)
    agent = Agent(name="weather_agent", tools=[get_weather],
                  model_client=EchoProvider(use_tools=True))
    async with Worker(agent, get_weather, bus=bus_w,
                      store=InProcessThreadStore(), directory=dir_w):
        bus_c, dir_c = await connect_mesh(broker.address)
        client = await Client.connect(bus_c, dir_c)
        lat: list[float] = []

        async def turn(i: int) -> None:
            t0 = time.perf_counter()
            r = await client.execute("weather_agent", f"weather in city{i}?",
                                     timeout=60)
            assert r.text
            lat.append((time.perf_counter() - t0) * 1e3)

        # warmup
        await asyncio.gather(*(turn(i) for i in range(concurrency)))
        lat.clear()
        t0 = time.perf_counter()
        for base in range(0, turns, concurrency):
            await asyncio.gather(*(turn(base + i) for i in range(concurrency)))
        wall = time.perf_counter() - t0
        print(json.dumps({
            "metric": "agent turns/sec (plumbing, echo provider, local mesh)",
            "value": round(turns / wall, 1),
            "unit": "turns/s",
            "p50_turn_ms": round(statistics.median(lat), 2),
            "config": {"model": "echo-stub", "broker": "bundled mesh",
                        "concurrency": concurrency},
        }))
    await broker.stop()


if __name__ == "__main__":
    asyncio.run(run())
