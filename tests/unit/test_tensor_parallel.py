"""Multi-process TP correctness on CPU (gloo, world_size 2).

TP=2 sharded forward must reproduce the TP=1 logits: the same global
weights are what both decompositions compute (weights here are derived from
a shared full-model tensor, sliced per rank).
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from calfkit_amd.models.config import TINY

pytestmark = pytest.mark.dist


def _shard_from_full(full_sd, cfg, rank, world):
    """Slice a full model's weights into rank's TP shard layout."""
    from calfkit_amd.models.llama import LlamaShard

    shard = LlamaShard(cfg, tp_rank=rank, tp_size=world, device="cpu")
    dh = cfg.head_dim
    hq_s, hkv_s, ffn_s = shard.hq, shard.hkv, shard.ffn_s
    shard.embed.copy_(full_sd["embed"])
    shard.lm_head.copy_(full_sd["lm_head"])
    shard.norm_final.copy_(full_sd["norm_final"])
    for i, lw in enumerate(shard.layers):
        # weights are [out, in]: column-parallel = ROW slices of wqkv/wgateup,
        # row-parallel = COLUMN slices of wo/wdown
        wqkv = full_sd[f"l{i}.wqkv"]  # [(hq + 2hkv)*dh, d] full
        hq, hkv = cfg.heads, cfg.kv_heads
        qf = wqkv[: hq * dh]
        kf = wqkv[hq * dh:(hq + hkv) * dh]
        vf = wqkv[(hq + hkv) * dh:]
        lw.wqkv.copy_(torch.cat([
            qf[rank * hq_s * dh:(rank + 1) * hq_s * dh],
            kf[rank * hkv_s * dh:(rank + 1) * hkv_s * dh],
            vf[rank * hkv_s * dh:(rank + 1) * hkv_s * dh]], dim=0))
        lw.wo.copy_(
            full_sd[f"l{i}.wo"][:, rank * hq_s * dh:(rank + 1) * hq_s * dh])
        wg = full_sd[f"l{i}.wgateup"]  # [2*ffn, d] full: [gate rows | up rows]
        gate = wg[: cfg.ffn]
        up = wg[cfg.ffn:]
        lw.wgateup.copy_(torch.cat([
            gate[rank * ffn_s:(rank + 1) * ffn_s],
            up[rank * ffn_s:(rank + 1) * ffn_s]], dim=0))
        lw.wdown.copy_(
            full_sd[f"l{i}.wdown"][:, rank * ffn_s:(rank + 1) * ffn_s])
        lw.norm_attn.copy_(full_sd[f"l{i}.norm_attn"])
        lw.norm_mlp.copy_(full_sd[f"l{i}.norm_mlp"])
    return shard


def _full_state(cfg, seed=7):
    from calfkit_amd.models.llama import LlamaShard

    full = LlamaShard(cfg, tp_rank=0, tp_size=1, device="cpu", seed=seed)
    sd = {"embed": full.embed, "lm_head": full.lm_head,
          "norm_final": full.norm_final}
    for i, lw in enumerate(full.layers):
        sd[f"l{i}.wqkv"] = lw.wqkv
        sd[f"l{i}.wo"] = lw.wo
        sd[f"l{i}.wgateup"] = lw.wgateup
        sd[f"l{i}.wdown"] = lw.wdown
        sd[f"l{i}.norm_attn"] = lw.norm_attn
        sd[f"l{i}.norm_mlp"] = lw.norm_mlp
    return full, sd


def _run_forward(model, cfg):
    from calfkit_amd.engine.kvcache import KVCache

    kv = KVCache(layers=cfg.layers, kv_heads=model.hkv, head_dim=model.dh,
                 num_pages=4, device="cpu")
    toks = [1, 9, 33, 77]
    slots = torch.as_tensor(kv.slots_for(1, len(toks)), dtype=torch.int32)
    scratch = [(torch.zeros(64, model.hkv, model.dh, dtype=model.dtype),
                torch.zeros(64, model.hkv, model.dh, dtype=model.dtype))
               for _ in range(cfg.layers)]
    return model.forward(
        torch.as_tensor(toks), torch.arange(len(toks), dtype=torch.int32),
        kv.kcache, kv.vcache, slots, prefill_scratch=scratch,
        logits_rows=slice(len(toks) - 1, len(toks)))


def _worker(rank, world, port, cfg, result_q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import calfkit_amd.parallel.pg as pg

    pg._DIST = None
    pg.init_distributed(backend="gloo", device_override="cpu")
    _, sd = _full_state(cfg)
    shard = _shard_from_full(sd, cfg, rank, world)
    logits = _run_forward(shard, cfg)
    if rank == 0:
        result_q.put(logits.float())
    import torch.distributed as td

    td.barrier()
    td.destroy_process_group()


def test_tp2_matches_tp1():
    cfg = TINY  # heads=2, kv=1... kv_heads=1 can't shard by 2 — use a wider cfg
    from dataclasses import replace

    cfg = replace(TINY, heads=4, kv_heads=2, ffn=512)
    full, sd = _full_state(cfg)
    ref = _run_forward(full, cfg).float()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_worker, args=(r, 2, port, cfg, q))
             for r in range(2)]
    for p in procs:
        p.start()
    tp_logits = q.get(timeout=120)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert torch.allclose(tp_logits, ref, atol=0.15, rtol=0.05), \
        (tp_logits - ref).abs().max()
    assert int(tp_logits.argmax()) == int(ref.argmax())
