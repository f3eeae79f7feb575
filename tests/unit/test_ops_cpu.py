"""CPU sanity for the torch reference implementations of the op surface.

The GPU lane (test_ops_gpu.py) compares the HIP kernels against these same
formulas; here we pin the reference semantics themselves (shapes, masking,
in-place contracts) so both lanes share one truth.
"""

import math

import torch

import calfkit_amd.ops as ops


def test_rmsnorm_reference():
    x = torch.randn(5, 64, dtype=torch.bfloat16)
    w = torch.randn(64, dtype=torch.bfloat16)
    out = torch.empty_like(x)
    ops.rmsnorm(out, x, w, 1e-5)
    xf = x.float()
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * w.float()
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2)


def test_fused_add_rmsnorm_inplace_contract():
    h = torch.randn(3, 64, dtype=torch.bfloat16)
    r = torch.randn(3, 64, dtype=torch.bfloat16)
    w = torch.ones(64, dtype=torch.bfloat16)
    h0, r0 = h.clone(), r.clone()
    ops.fused_add_rmsnorm(h, r, w, 1e-5)
    assert torch.allclose(r.float(), (h0.float() + r0.float()), atol=2e-2)
    assert not torch.allclose(h.float(), h0.float())  # h now holds the norm


def test_rope_zero_position_is_identity():
    q = torch.randn(4, 2, 128, dtype=torch.bfloat16)
    k = torch.randn(4, 1, 128, dtype=torch.bfloat16)
    q0 = q.clone()
    table = ops.build_rope_table(16, 128)
    ops.rope_qk(q, k, table, torch.zeros(4, dtype=torch.int32))
    assert torch.allclose(q.float(), q0.float(), atol=1e-2)


def test_attn_prefill_reference_is_causal():
    S, hq, hkv, dh = 8, 2, 1, 128
    q = torch.randn(S, hq, dh, dtype=torch.bfloat16)
    k = torch.randn(S, hkv, dh, dtype=torch.bfloat16)
    v = torch.randn(S, hkv, dh, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    ops.attn_prefill(out, q, k, v, 1.0 / math.sqrt(dh))
    # row 0 attends only key 0 -> output equals v[0]
    assert torch.allclose(out[0, 0].float(), v[0, 0].float(), atol=2e-2)


def test_attn_decode_matches_prefill_last_row():
    torch.manual_seed(0)
    S, hq, hkv, dh, page = 70, 4, 2, 128, 64
    k = torch.randn(S, hkv, dh, dtype=torch.bfloat16)
    v = torch.randn(S, hkv, dh, dtype=torch.bfloat16)
    q = torch.randn(1, hq, dh, dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(dh)

    # paged layout
    npages = (S + page - 1) // page
    kc = torch.zeros(npages, hkv, page, dh, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    slots = torch.arange(S, dtype=torch.int32)
    ops.kv_write(kc, vc, k, v, slots)
    pt = torch.arange(npages, dtype=torch.int32).unsqueeze(0)
    out_d = torch.empty_like(q)
    ops.attn_decode(out_d, q, kc, vc, pt, torch.tensor([S], dtype=torch.int32),
                    scale)

    out_p = torch.empty(1, hq, dh, dtype=torch.bfloat16)
    ops.attn_prefill(out_p, q, k, v, scale, past_len=S - 1)
    assert torch.allclose(out_d.float(), out_p.float(), atol=2e-2, rtol=2e-2)


def test_sample_greedy_cpu():
    logits = torch.randn(3, 100)
    logits[:, 7] += 50
    out = torch.empty(3, dtype=torch.int32)
    ops.sample(out, logits, torch.zeros(3), torch.tensor([1, 2]))
    assert (out == 7).all()
