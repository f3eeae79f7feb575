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


def test_tile_plan_matches_measured_policy():
    """The (BN, splitk) chooser reproduces the r02 sweep winners
    (profiles/r02_tile_gemm.md) for every llama serving shape."""
    import calfkit_amd.ops as ops

    expect = {
        (512, 6144, 4096): (128, 2),     # 8b qkv
        (512, 4096, 4096): (128, 4),     # 8b o
        (512, 28672, 4096): (256, 1),    # 8b gateup
        (512, 4096, 14336): (128, 4),    # 8b down
        (512, 128256, 4096): (256, 1),   # 8b lm_head
        (512, 1280, 8192): (128, 8),     # 70b tp8 qkv
        (512, 8192, 1024): (128, 1),     # 70b tp8 o (K too shallow to split)
        (512, 7168, 8192): (128, 2),     # 70b tp8 gateup
        (512, 8192, 3584): (128, 2),     # 70b tp8 down
    }
    for (m, n, k), plan in expect.items():
        assert ops.tile_plan(m, n, k) == plan, (m, n, k)
    # unroutable: N not divisible by 128
    assert ops.tile_plan(512, 1000, 4096) is None


def test_grammar_transition_state_machine():
    from calfkit_amd import ops

    s = 0
    s = ops.grammar_transition(s, 99)            # free text stays free
    assert s == 0
    s = ops.grammar_transition(s, ops.G_TCALL)   # -> awaiting name
    assert s & 0xFF == 1
    s = ops.grammar_transition(s, 500)           # name consumed -> args
    assert s & 0xFF == 2
    for k in range(ops.G_ARG_TOKS):
        s = ops.grammar_transition(s, 100 + k)
    assert s & 0xFF == 2 + ops.G_ARG_TOKS
    s = ops.grammar_transition(s, ops.G_TEND)    # closed: free, calls=1
    assert s & 0xFF == 0 and s >> 8 == 1


def test_linear_indexed_falls_back_off_gpu():
    import torch

    import calfkit_amd.ops as ops

    x = torch.randn(4, 8, dtype=torch.float32)
    w = torch.randn(6, 8, dtype=torch.float32)
    out = ops.linear_indexed(x, w)
    assert torch.allclose(out, x @ w.T, atol=1e-5)
    # the algo table only admits numerics-ok entries with real speedups
    table = ops._algo_table()
    assert all(isinstance(k, tuple) and len(k) == 3 for k in table)


def test_small_plan_routes_only_the_measured_class():
    """Shallow-K wide-N (TP-sharded o projections) routes to the 64x128
    kernel; deep-K, narrow-N and skinny batches stay off it
    (profiles/r02_tile_gemm.md small-kernel sweep)."""
    assert ops.small_plan(512, 8192, 1024)       # o_70b_tp8: 1.51x vs lib
    assert ops.small_plan(512, 4096, 512)        # o_8b_tp8: 2.77x
    assert ops.small_plan(512, 4096, 1792)       # down_8b_tp8: 1.32x (ring)
    assert not ops.small_plan(512, 6144, 4096)   # qkv deep-K: big kernel's
    assert ops.small_plan(256, 4096, 512)        # TP-overlap half-chunk
    assert ops.small_plan(192, 4096, 1792)       # 96 blocks: measured 1.3x+
    assert not ops.small_plan(64, 8192, 1024)    # skinny: skinny_gemm's
    assert not ops.small_plan(512, 1000, 512)    # N % 128


def test_moe_route_cpu_matches_bucket_pipeline():
    """CPU fallback of the fused router == softmax/topk/renorm +
    moe_buckets, and inv inverts the stable bucket order so moe_combine
    reproduces index_add exactly."""
    torch.manual_seed(5)
    T, E, K = 16, 8, 2
    logits = torch.randn(T, E)
    tok, off, scale, inv = ops.moe_route(logits, E, K)
    w, e = torch.topk(torch.softmax(logits.float(), -1), K, dim=-1)
    w = w / w.sum(-1, keepdim=True)
    tok2, off2, scale2 = ops.moe_buckets(e, w, E)
    assert torch.equal(tok, tok2) and torch.equal(off, off2)
    assert torch.allclose(scale, scale2)
    assert torch.equal(torch.sort(inv).values.long(), torch.arange(T * K))
    y = torch.randn(T * K, 8, dtype=torch.bfloat16)
    out = torch.empty(T, 8, dtype=torch.bfloat16)
    ops.moe_combine(out, y, inv, K)
    ref = torch.zeros(T, 8)
    ref.index_add_(0, tok.long(), y.float())
    assert torch.allclose(out.float(), ref.to(torch.bfloat16).float())
