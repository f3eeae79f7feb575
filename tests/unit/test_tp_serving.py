"""Multi-process TP SERVING lockstep on CPU (gloo, world 2).

Rank 0 drives the engine scheduler; rank 1 replays broadcast ops.  The test
proves no deadlock, identical batch evolution, and that rank 0 completes
generations while rank 1 follows through prefill + decode + stop.
"""

import os

import pytest
import torch.multiprocessing as mp

from calfkit_amd.models.config import TINY

pytestmark = pytest.mark.dist


def _worker(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from dataclasses import replace

    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")
    cfg = replace(TINY, heads=4, kv_heads=2, ffn=512)

    from calfkit_amd.engine.engine import GenRequest, InferenceEngine
    from calfkit_amd.models.llama import LlamaShard

    model = LlamaShard(cfg, tp_rank=rank, tp_size=world, device="cpu")
    engine = InferenceEngine(model, max_batch=4, use_graphs=False,
                             max_model_len=256)
    if rank == 0:
        futs = [engine.submit(GenRequest(prompt_ids=[1, 5 + i, 9],
                                         max_new_tokens=5))
                for i in range(3)]
        outs = [f.result(120).token_ids for f in futs]
        engine.stop()
        q.put(outs)
    else:
        engine.follow()
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_tp_lockstep_serving_completes():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29521, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = q.get(timeout=180)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert len(outs) == 3
    assert all(len(o) == 5 for o in outs)


def _worker4(rank, world, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import time
    from dataclasses import replace

    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")
    cfg = replace(TINY, heads=4, kv_heads=4, ffn=512)

    from calfkit_amd.engine.engine import GenRequest, InferenceEngine
    from calfkit_amd.models.llama import LlamaShard

    model = LlamaShard(cfg, tp_rank=rank, tp_size=world, device="cpu")
    engine = InferenceEngine(model, max_batch=8, use_graphs=False,
                             max_model_len=256)
    if rank == 0:
        # wave 1: plain decode requests
        futs = [engine.submit(GenRequest(prompt_ids=[1, 5 + i, 9],
                                         max_new_tokens=8))
                for i in range(2)]
        time.sleep(0.5)  # wave 2 admits MID-RUN, while wave 1 decodes
        futs.append(engine.submit(GenRequest(
            prompt_ids=[1, 30, 31, 32], max_new_tokens=10)))
        # a tool-grammar request exercises the tensor-coded tool plan
        # across the lockstep broadcast (the engine-level "handoff" shape)
        futs.append(engine.submit(GenRequest(
            prompt_ids=[1, 40, 41], max_new_tokens=12, temperature=0.7,
            seed=5, ignore_eos=False, tool_names=[100, 200],
            tool_trigger=1, tool_ncalls=1)))
        outs = [f.result(240).token_ids for f in futs]
        engine.stop()
        q.put(outs)
    else:
        engine.follow()
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_tp4_lockstep_mid_run_admission_and_tool_plan():
    """4-rank gloo lockstep: wave-2 prefills admit while
    wave 1 is mid-decode, and a grammar tool-call request rides the
    tensor-coded broadcast — all ranks stay in step to completion."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker4, args=(r, 4, 29529, q))
             for r in range(4)]
    for p in procs:
        p.start()
    outs = q.get(timeout=300)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert len(outs) == 4
    assert len(outs[0]) == 8 and len(outs[2]) == 10
    # the tool request emitted a grammar call span
    from calfkit_amd import ops
    toks = outs[3]
    assert ops.G_TCALL in toks
    name = toks[toks.index(ops.G_TCALL) + 1]
    assert name in (100, 200)
