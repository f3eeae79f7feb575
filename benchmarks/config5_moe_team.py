#!/usr/bin/env python3
"""BASELINE config 5: Mixtral-8x7B MoE provider driving a 3-agent team.

A coordinator messages two specialist peers (awaiting replies), then
answers.  All three agents share one MoE engine per GPU rank.
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

TURNS_PER_STEP = 8
STEPS = int(os.environ.get("C5_STEPS", "2"))
WARMUP = int(os.environ.get("C5_WARMUP", "1"))


async def run() -> None:
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import (
        InMemoryBus, InMemoryDirectory, InProcessThreadStore)
    from calfkit_amd.definitions.selections import Messaging
    from calfkit_amd.engine.local import LocalModelClient
    from calfkit_amd.models.config import get_config
    from calfkit_amd.parallel import init_distributed
    from calfkit_amd.schemas import ModelSettings
    from calfkit_amd.testing import ToolCallDirector

    dist = init_distributed()
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        cfg = get_config("mixtral-8x7b")
    else:
        from tests.unit.test_mixtral import TINY_MOE as cfg  # type: ignore

    # Multi-rank launch (torchrun) shards the experts: EP = world size,
    # rank 0 drives the mesh + engine schedule, followers replay in
    # lockstep (the per-layer all-to-alls line up).  Single rank = the
    # BASELINE single-GPU measurement.
    ep = dist.world_size if dist.world_size > 1 else 1
    if dist.rank != 0:
        from calfkit_amd.engine.local import shared_engine

        engine = shared_engine(cfg, ep=ep, max_batch=TURNS_PER_STEP * 4,
                               max_model_len=1024, autostart=False)
        engine.follow()
        return

    def provider(directed: bool):
        # use_graphs defaults on (EP=1): MoE decode is capturable through
        # the grouped kernels (device-side bucketing, no host sync)
        inner = LocalModelClient(cfg, max_batch=TURNS_PER_STEP * 4,
                                 max_model_len=1024, ep=ep)
        if not directed:
            return inner
        return ToolCallDirector(inner, calls=[
            ("send_message", {"to": "analyst", "message": "analyze this"}),
            ("send_message", {"to": "critic", "message": "critique this"}),
        ])

    settings = ModelSettings(max_tokens=24, temperature=0.0)
    coordinator = Agent(name="coordinator", model_client=provider(True),
                        peers=[Messaging(names=("analyst", "critic"))],
                        model_settings=settings)
    analyst = Agent(name="analyst", model_client=provider(False),
                    model_settings=settings)
    critic = Agent(name="critic", model_client=provider(False),
                   model_settings=settings)
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    lat: list[float] = []

    async with Worker(coordinator, analyst, critic, bus=bus, store=store,
                      directory=directory):
        client = await Client.connect(bus, directory)

        async def turn(i: int, record: bool) -> None:
            t0 = time.perf_counter()
            r = await client.execute("coordinator", f"team task {i}",
                                     timeout=900)
            assert r.text.strip()
            if record:
                lat.append((time.perf_counter() - t0) * 1e3)

        async def step(s: int, record: bool) -> None:
            await asyncio.gather(*(turn(s * TURNS_PER_STEP + i, record)
                                   for i in range(TURNS_PER_STEP)))

        for w in range(WARMUP):
            await step(-1 - w, False)
        if on_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for s in range(STEPS):
            await step(s, True)
        if on_gpu:
            torch.cuda.synchronize()
        wall = time.perf_counter() - t0

    if dist.rank == 0:
        turns = STEPS * TURNS_PER_STEP
        print(json.dumps({
            "metric": "3-agent team turns/sec (MoE provider)",
            "value": round(turns / wall, 2),
            "unit": "team-turns/s",
            "p50_turn_ms": round(statistics.median(lat), 1),
            "config": {"model": cfg.name, "agents": 3,
                        "pattern": "coordinator + 2 awaited peers"},
        }), flush=True)


if __name__ == "__main__":
    asyncio.run(run())
