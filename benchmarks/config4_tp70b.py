#!/usr/bin/env python3
"""BASELINE config 4: Llama-3-70B TP=8 over RCCL/xGMI, agent output
streaming to a mesh topic.

Run on an 8-GPU MI355X node:

  torchrun --standalone --nproc-per-node 8 --local-addr 127.0.0.1 \\
      benchmarks/config4_tp70b.py

Rank 0 hosts the agent mesh (one Agent, Llama-3-70B TP=8 provider) and a
firehose subscriber counting streamed chunks; ranks 1-7 run the TP follower
loop (lockstep replay of rank 0's engine schedule; every rank executes the
same forwards and RCCL all-reduces in the same order).
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

TURNS_PER_STEP = int(os.environ.get("C4_TURNS", "16"))
STEPS = int(os.environ.get("C4_STEPS", "2"))
WARMUP = int(os.environ.get("C4_WARMUP", "1"))
DECODE_TOKENS = int(os.environ.get("C4_DECODE", "64"))


async def run_rank0(dist, cfg) -> None:
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import (
        InMemoryBus, InMemoryDirectory, InProcessThreadStore)
    from calfkit_amd.engine.local import LocalModelClient
    from calfkit_amd.schemas import ModelSettings

    provider = LocalModelClient(cfg, max_batch=TURNS_PER_STEP * 2,
                                max_model_len=1024, use_graphs=False, tp=dist.world_size)
    agent = Agent(name="big_agent", model_client=provider,
                  model_settings=ModelSettings(max_tokens=DECODE_TOKENS,
                                               temperature=0.0))
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    prompt = " ".join(f"w{i}" for i in range(120))
    lat: list[float] = []
    chunks = 0

    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)

        async def count_chunks(tid: str) -> None:
            nonlocal chunks
            async for c in client.firehose(tid):
                chunks += 1
                if c.event == "step_finished":
                    return

        async def turn(i: int, record: bool) -> None:
            t0 = time.perf_counter()
            tid = f"t-{i}"
            watcher = asyncio.create_task(count_chunks(tid))
            r = await client.execute("big_agent", f"turn {i}: {prompt}",
                                     thread_id=tid, timeout=1800)
            assert r.text.strip()
            await asyncio.wait_for(watcher, 30)
            if record:
                lat.append((time.perf_counter() - t0) * 1e3)

        async def step(s: int, record: bool) -> None:
            await asyncio.gather(*(turn(s * TURNS_PER_STEP + i, record)
                                   for i in range(TURNS_PER_STEP)))

        for w in range(WARMUP):
            await step(-1 - w, False)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for s in range(STEPS):
            await step(s, True)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        wall = time.perf_counter() - t0
        provider.engine.stop()

    turns = STEPS * TURNS_PER_STEP
    print(json.dumps({
        "metric": "agent turns/sec, Llama-3-70B TP=8 over RCCL/xGMI",
        "value": round(turns / wall, 2),
        "unit": "turns/s",
        "n_gpus": dist.world_size,
        "p50_turn_ms": round(statistics.median(lat), 1),
        "streamed_chunks": chunks,
        "config": {"model": cfg.name, "parallelism": f"tp{dist.world_size}",
                    "decode_tokens": DECODE_TOKENS},
    }), flush=True)


def main() -> None:
    from calfkit_amd.engine.local import shared_engine
    from calfkit_amd.models.config import get_config
    from calfkit_amd.parallel import init_distributed

    dist = init_distributed()
    if torch.cuda.is_available():
        cfg = get_config("llama-3-70b")
    else:
        # sanity mode: tiny widened so heads/kv/ffn shard over the world
        from dataclasses import replace

        cfg = replace(get_config("tiny"), heads=2 * dist.world_size,
                      kv_heads=dist.world_size, ffn=256 * dist.world_size,
                      name="tiny")
    if dist.rank == 0:
        asyncio.run(run_rank0(dist, cfg))
    else:
        engine = shared_engine(cfg, tp=dist.world_size, use_graphs=False,
                               autostart=False,  # followers replay, no scheduler
                               max_batch=TURNS_PER_STEP * 2, max_model_len=1024)
        engine.follow()


if __name__ == "__main__":
    main()
