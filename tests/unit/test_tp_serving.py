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
