"""Multi-process expert parallelism on CPU (gloo, world_size 2).

EP=2 (experts split across ranks, tokens exchanged over all-to-all) must
reproduce the EP=1 output.  Per-expert weight seeds make every layout
generate identical experts, so no state copying is needed — the layouts
agree by construction (models/mixtral.py).
"""

import os
from dataclasses import replace

import pytest
import torch
import torch.multiprocessing as mp

from calfkit_amd.models.config import ModelConfig

pytestmark = pytest.mark.dist

CFG = ModelConfig(
    name="tiny-moe-ep", vocab_size=128, hidden=128, layers=2, heads=2,
    kv_heads=1, head_dim=128, ffn=256, rope_theta=10000.0, max_seq=256,
    num_experts=4, experts_per_token=2)


def _mlp_inputs(cfg):
    torch.manual_seed(11)
    return torch.randn(6, cfg.hidden, dtype=torch.bfloat16)


def _worker(rank, world, port, result_q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")
    from calfkit_amd.models.mixtral import MixtralShard

    model = MixtralShard(CFG, ep_rank=rank, ep_size=world, device="cpu")
    x = _mlp_inputs(CFG)
    out = model._mlp(model.layers[0], x)
    if rank == 0:
        result_q.put(out.float())
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_ep2_matches_ep1():
    from calfkit_amd.models.mixtral import MixtralShard

    single = MixtralShard(CFG, device="cpu")
    x = _mlp_inputs(CFG)
    ref = single._mlp(single.layers[0], x).float()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29519, q))
             for r in range(2)]
    for p in procs:
        p.start()
    ep_out = q.get(timeout=120)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert torch.allclose(ep_out, ref, atol=5e-2, rtol=5e-2), \
        (ep_out - ref).abs().max()


def test_ep_rejects_uneven():
    from calfkit_amd.models.mixtral import MixtralShard

    with pytest.raises(ValueError):
        MixtralShard(replace(CFG, num_experts=3), ep_size=2, device="cpu")
    # EP x TP composes since r02 — construction succeeds
    MixtralShard(replace(CFG, heads=4, kv_heads=2, ffn=512),
                 ep_size=2, tp_size=2, device="cpu")


def _serve_worker(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")
    from calfkit_amd.engine.engine import GenRequest
    from calfkit_amd.engine.local import shared_engine

    engine = shared_engine(CFG, "cpu", ep=2, max_batch=4, use_graphs=False,
                           max_model_len=128, autostart=False)
    assert engine.tp_lockstep
    if rank == 0:
        engine.start()
        fut = engine.submit(GenRequest(prompt_ids=[1, 7, 9], max_new_tokens=4))
        out = fut.result(120).token_ids
        engine.stop()
        q.put(out)
    else:
        engine.follow()
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_ep_lockstep_serving_completes():
    """EP=2 engine serves in lockstep (rank 0 schedules, rank 1 follows,
    MoE all-to-alls line up) and completes a generation."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_serve_worker, args=(r, 2, 29523, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = q.get(timeout=180)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert len(out) == 4


def _worker_loop_rank(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import asyncio

    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")

    if rank != 0:
        from calfkit_amd.engine.local import shared_engine

        engine = shared_engine(CFG, "cpu", ep=2, max_batch=4,
                               use_graphs=False, max_model_len=128,
                               autostart=False)
        engine.follow()
    else:
        async def main():
            from calfkit_amd import Agent, Client, Worker
            from calfkit_amd.adapters.memory import (
                InMemoryBus,
                InMemoryDirectory,
                InProcessThreadStore,
            )
            from calfkit_amd.engine.local import LocalModelClient
            from calfkit_amd.schemas import ModelSettings

            provider = LocalModelClient(CFG, device="cpu", ep=2, max_batch=4,
                                        use_graphs=False, max_model_len=128)
            agent = Agent(name="moe-server", model_client=provider,
                          model_settings=ModelSettings(max_tokens=4,
                                                       temperature=0.0))
            bus, store, d = (InMemoryBus(), InProcessThreadStore(),
                             InMemoryDirectory())
            async with Worker(agent, bus=bus, store=store, directory=d):
                client = await Client.connect(bus, d)
                r = await client.execute("moe-server", "hello experts",
                                         timeout=240)
                assert r.text.strip()
                q.put(r.text)
            provider.engine.stop()

        asyncio.run(main())
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_ep_under_the_serving_worker():
    """the EP engine serves under the REAL Worker loop —
    agent plane on rank 0, EP follower on rank 1, full mesh round trip."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_loop_rank, args=(r, 2, 29531, q))
             for r in range(2)]
    for p in procs:
        p.start()
    text = q.get(timeout=240)
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    assert text.strip()
