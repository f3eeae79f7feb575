"""Unit: Mixtral MoE layer + engine integration (tiny MoE config, CPU)."""

import pytest
import torch

from calfkit_amd.engine.engine import GenRequest, InferenceEngine
from calfkit_amd.models.config import ModelConfig
from calfkit_amd.models.mixtral import MixtralShard

TINY_MOE = ModelConfig(
    name="tiny-moe", vocab_size=256, hidden=128, layers=2, heads=2, kv_heads=1,
    head_dim=128, ffn=256, rope_theta=10000.0, max_seq=512,
    num_experts=4, experts_per_token=2)


def test_moe_mlp_matches_dense_reference():
    torch.manual_seed(0)
    model = MixtralShard(TINY_MOE, device="cpu")
    lw = model.layers[0]
    x = torch.randn(5, TINY_MOE.hidden, dtype=torch.bfloat16)
    out = model._mlp(lw, x)
    assert out.shape == x.shape

    # reference: dense per-token loop over its top-2 experts
    logits = torch.nn.functional.linear(x, lw.router).float()
    w, e = torch.topk(torch.softmax(logits, -1), 2, dim=-1)
    w = w / w.sum(-1, keepdim=True)
    ref = torch.zeros_like(x, dtype=torch.float32)
    for t in range(x.shape[0]):
        for j in range(2):
            ei = int(e[t, j])
            gu = torch.nn.functional.linear(x[t:t + 1], lw.wgateup[ei])
            f = gu.shape[-1] // 2
            act = torch.nn.functional.silu(gu[:, :f].float()) * gu[:, f:].float()
            ys = torch.nn.functional.linear(act.to(x.dtype), lw.wdown[ei])
            ref[t] += float(w[t, j]) * ys[0].float()
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2), \
        (out.float() - ref).abs().max()


def test_moe_engine_generates():
    model = MixtralShard(TINY_MOE, device="cpu")
    eng = InferenceEngine(model, max_batch=2, use_graphs=False,
                          max_model_len=256)
    try:
        r = eng.generate(GenRequest(prompt_ids=[1, 5, 9], max_new_tokens=4),
                         timeout=120)
        assert len(r.token_ids) == 4
    finally:
        eng.stop()


def test_moe_buckets_device_free():
    """moe_buckets: bucket rows sorted by expert, offsets = cumcounts."""
    from calfkit_amd import ops

    torch.manual_seed(3)
    T, E, k = 11, 4, 2
    logits = torch.randn(T, E)
    weights, experts = torch.topk(torch.softmax(logits, -1), k, dim=-1)
    tok, off, scale = ops.moe_buckets(experts, weights, E)
    assert tok.dtype == torch.int32 and off.dtype == torch.int32
    assert int(off[0]) == 0 and int(off[-1]) == T * k
    # each bucket row must be a (token, expert) pair the router picked,
    # with the matching weight
    offs = off.tolist()
    for e in range(E):
        for r in range(offs[e], offs[e + 1]):
            t = int(tok[r])
            j = (experts[t] == e).nonzero()
            assert len(j) == 1, (t, e)
            assert abs(float(scale[r]) - float(weights[t, j[0, 0]])) < 1e-6


def test_moe_grouped_fallback_matches_host_loop():
    """ops.moe_gateup_silu/moe_down CPU references == the host-loop MLP."""
    from calfkit_amd import ops

    torch.manual_seed(4)
    model = MixtralShard(TINY_MOE, device="cpu")
    lw = model.layers[0]
    T, k, E = 7, 2, TINY_MOE.num_experts
    d, F = TINY_MOE.hidden, model.ffn_s
    x = torch.randn(T, d, dtype=torch.bfloat16)
    host = model._mlp(lw, x)  # host-grouped path

    logits = torch.nn.functional.linear(x, lw.router).float()
    weights, experts = torch.topk(torch.softmax(logits, -1), k, dim=-1)
    weights = weights / weights.sum(-1, keepdim=True)
    tok, off, scale = ops.moe_buckets(experts, weights, E)
    R = T * k
    act = torch.empty(R, F, dtype=torch.bfloat16)
    ops.moe_gateup_silu(act, x, lw.wgateup, tok, off, torch.empty(0))
    y = torch.empty(R, d, dtype=torch.bfloat16)
    ops.moe_down(y, act, lw.wdown, off, scale)
    out = torch.zeros(T, d, dtype=torch.float32)
    out.index_add_(0, tok.long(), y.float())
    assert torch.allclose(out, host.float(), atol=5e-2, rtol=5e-2), \
        (out - host.float()).abs().max()


@pytest.mark.gpu
def test_moe_engine_decodes_under_graphs_gpu():
    """On GPU the MoE engine must run decode under hipGraphs (the grouped
    kernels make it capturable) — a silent eager fallback is a regression."""
    model = MixtralShard(TINY_MOE, device="cuda")
    eng = InferenceEngine(model, max_batch=2, max_model_len=256)
    try:
        assert eng.use_graphs
        r = eng.generate(GenRequest(prompt_ids=[1, 5, 9], max_new_tokens=8),
                         timeout=120)
        assert len(r.token_ids) == 8
        assert eng._stats["graph_replays"] > 0
    finally:
        eng.stop()
