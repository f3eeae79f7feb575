#!/usr/bin/env python3
"""BASELINE config 3: concurrent tool-call bench — 1 Agent + 4 @agent_tool
nodes per worker replica, Llama-3-8B provider, replicas pinned 1/GPU.

Each turn: prefill + decode steered by the ENGINE's tool-call grammar into
4 parallel tool calls (zero ToolCallDirector — the sampled tokens ARE the
calls), 4 tool hops over the mesh, then a second full model
step for the final answer — the whole thing through the wire.  Run under
torchrun for N replicas:

  torchrun --standalone --nproc-per-node N benchmarks/config3_toolcalls.py
"""

from __future__ import annotations

import asyncio
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

TURNS_PER_STEP = int(os.environ.get("C3_TURNS", "48"))
STEPS = int(os.environ.get("C3_STEPS", "3"))
WARMUP = int(os.environ.get("C3_WARMUP", "1"))


async def run() -> None:
    from calfkit_amd import Agent, Client, Worker, agent_tool
    from calfkit_amd.adapters.memory import (
        InMemoryBus, InMemoryDirectory, InProcessThreadStore)
    from calfkit_amd.engine.local import LocalModelClient
    from calfkit_amd.models.config import get_config
    from calfkit_amd.parallel import barrier, init_distributed
    from calfkit_amd.schemas import ModelSettings

    # 4-way tool fan-out per turn, sampled by the engine's grammar
    os.environ.setdefault("CALF_TOOLCALL_FANOUT", "4")
    dist = init_distributed()
    on_gpu = torch.cuda.is_available()
    model = "llama-3-8b" if on_gpu else "tiny"

    tool_hits = [0]
    tools = []
    for i in range(4):
        def make(i=i):
            def fn(query: str) -> str:
                tool_hits[0] += 1
                return f"tool{i} result for {query}"
            fn.__name__ = f"search_{i}"
            fn.__doc__ = f"Search backend {i}."
            return fn
        tools.append(agent_tool(make()))

    provider = LocalModelClient(get_config(model),
                                max_batch=TURNS_PER_STEP * 2,
                                max_model_len=1024, use_graphs=on_gpu)
    agent = Agent(name="toolcaller", tools=[t.name for t in tools],
                  model_client=provider,
                  model_settings=ModelSettings(max_tokens=32,
                                               temperature=0.7, seed=17))
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    lat: list[float] = []

    async with Worker(agent, *tools, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)

        async def turn(i: int, record: bool) -> None:
            t0 = time.perf_counter()
            r = await client.execute("toolcaller",
                                     f"turn {i}: research the benchmark topic",
                                     timeout=600)
            assert r.text.strip()
            if record:
                lat.append((time.perf_counter() - t0) * 1e3)

        async def step(s: int, record: bool) -> None:
            await asyncio.gather(*(turn(s * TURNS_PER_STEP + i, record)
                                   for i in range(TURNS_PER_STEP)))

        for w in range(WARMUP):
            await step(-1 - w, False)
        barrier()
        if on_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for s in range(STEPS):
            await step(s, True)
        if on_gpu:
            torch.cuda.synchronize()
        barrier()
        wall = time.perf_counter() - t0

    world = dist.world_size
    if world > 1:
        import torch.distributed as td
        t = torch.tensor([wall], dtype=torch.float64,
                         device=dist.device if dist.backend == "nccl" else "cpu")
        td.all_reduce(t, op=td.ReduceOp.MAX)
        wall = float(t.item())
    if dist.rank == 0:
        turns = STEPS * TURNS_PER_STEP * world
        print(json.dumps({
            "metric": "agent turns/sec with 4-way parallel tool calls",
            "value": round(turns / wall, 2),
            "unit": "turns/s",
            "n_gpus": world,
            "p50_turn_ms": round(statistics.median(lat), 1),
            "tool_calls_per_turn": round(
                tool_hits[0] / max(1, (STEPS + WARMUP) * TURNS_PER_STEP), 2),
            "config": {"model": model, "tools": 4, "director": "none",
                        "replicas": f"{world} worker(s), 1/GPU",
                        "turns_per_step": TURNS_PER_STEP},
        }), flush=True)


if __name__ == "__main__":
    asyncio.run(run())
