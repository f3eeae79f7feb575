#!/usr/bin/env python3
"""Per-op micro timings at the serving operating point (batch 512):
decode attention, fused_add_rmsnorm, silu_mul, rope_qkv_write.  JSON out."""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import calfkit_amd.ops as ops  # noqa: E402


def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    B, hq, hkv, dh, d, f = 512, 32, 8, 128, 4096, 14336
    slen = 192
    pages_per = (slen + 63) // 64
    npages = B * pages_per + 1
    kc = torch.randn(npages, hkv, 64, dh, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, hkv, 64, dh, dtype=torch.bfloat16, device="cuda")
    pt = torch.arange(B * pages_per, dtype=torch.int32,
                      device="cuda").reshape(B, pages_per)
    # shuffled pages (realistic indirection)
    perm = torch.randperm(B * pages_per, device="cuda").to(torch.int32)
    pt = perm.reshape(B, pages_per).contiguous()
    sl = torch.full((B,), slen, dtype=torch.int32, device="cuda")
    q = torch.randn(B, hq, dh, dtype=torch.bfloat16, device="cuda")
    o = torch.empty_like(q)
    t = bench(lambda: ops.attn_decode(o, q, kc, vc, pt, sl, 0.0883))
    kv_mb = B * slen * hkv * dh * 2 * 2 / 1e6
    print(json.dumps({"op": "attn_decode", "us": round(t, 1),
                      "kv_mb": round(kv_mb, 1),
                      "tbps": round(kv_mb / t / 1e3, 2)}), flush=True)

    h = torch.randn(B, d, dtype=torch.bfloat16, device="cuda")
    r = torch.randn(B, d, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(d, dtype=torch.bfloat16, device="cuda")
    t = bench(lambda: ops.fused_add_rmsnorm(h, r, w, 1e-5))
    mb = B * d * 2 * 3 / 1e6  # read h,r + write r,h (r re-read in L1)
    print(json.dumps({"op": "fused_add_rmsnorm", "us": round(t, 1),
                      "mb": round(mb, 1)}), flush=True)

    gu = torch.randn(B, 2 * f, dtype=torch.bfloat16, device="cuda")
    act = torch.empty(B, f, dtype=torch.bfloat16, device="cuda")
    t = bench(lambda: ops.silu_mul(act, gu))
    mb = B * f * 2 * 3 / 1e6
    print(json.dumps({"op": "silu_mul", "us": round(t, 1),
                      "mb": round(mb, 1),
                      "tbps": round(mb / t / 1e3, 2)}), flush=True)

    qkv = torch.randn(B, (hq + 2 * hkv) * dh, dtype=torch.bfloat16,
                      device="cuda")
    rope = ops.build_rope_table(8200, dh, device="cuda")
    pos = torch.full((B,), 150, dtype=torch.int32, device="cuda")
    slots = torch.arange(B, dtype=torch.int32, device="cuda")
    t = bench(lambda: ops.rope_qkv_write(qkv, kc, vc, rope, pos, slots,
                                         hq, hkv))
    print(json.dumps({"op": "rope_qkv_write", "us": round(t, 1)}), flush=True)


if __name__ == "__main__":
    main()
