"""Unit: Mixtral MoE layer + engine integration (tiny MoE config, CPU)."""

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
