#!/usr/bin/env python3
"""Grouped MoE decode kernels vs the weight-stream bound and the host loop.

Mixtral-8x7B MLP shapes (d=4096, F=14336, E=8, top-2).  At decode the MoE
MLP is weight-streaming bound: every ACTIVE expert's gate-up+down weights
are read once per step, so time >= active_bytes / ~8 TB/s.  This prints
the measured effective stream rate of ops.moe_gateup_silu + ops.moe_down
per batch size, next to the host-grouped (counts.tolist + per-expert
hipBLASLt) loop the grouped path replaced.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from calfkit_amd import ops

D, F, E, K = 4096, 14336, 8, 2
ITERS = int(os.environ.get("MOE_ITERS", "20"))


def host_loop(x, wg, wd, experts, weights):
    T = x.shape[0]
    flat_e = experts.reshape(-1)
    flat_t = torch.arange(T, device=x.device).repeat_interleave(K)
    order = torch.argsort(flat_e, stable=True)
    counts = torch.bincount(flat_e, minlength=E).tolist()
    out = torch.zeros_like(x)
    gathered = x[flat_t[order]]
    w_sorted = weights.reshape(-1)[order]
    start = 0
    for ei in range(E):
        n = counts[ei]
        if n == 0:
            continue
        xs = gathered[start:start + n]
        gu = torch.nn.functional.linear(xs, wg[ei])
        act = torch.empty(n, F, dtype=xs.dtype, device=xs.device)
        ops.silu_mul(act, gu)
        ys = torch.nn.functional.linear(act, wd[ei])
        out.index_add_(0, flat_t[order[start:start + n]],
                       (ys.float() * w_sorted[start:start + n, None])
                       .to(out.dtype))
        start += n
    return out


def grouped(x, wg, wd, experts, weights, work):
    T = x.shape[0]
    tok, off, scale = ops.moe_buckets(experts, weights, E)
    R = T * K
    act = torch.empty(R, F, dtype=x.dtype, device=x.device)
    ops.moe_gateup_silu(act, x, wg, tok, off, work)
    y = torch.empty(R, D, dtype=x.dtype, device=x.device)
    ops.moe_down(y, act, wd, off, scale)
    out32 = torch.zeros(T, D, dtype=torch.float32, device=x.device)
    out32.index_add_(0, tok.long(), y.float())
    return out32.to(x.dtype)


def bench(fn, *args):
    for _ in range(3):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        fn(*args)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / ITERS * 1e3


def main():
    assert torch.cuda.is_available() and ops.have_extension()
    dev = "cuda"
    torch.manual_seed(7)
    wg = torch.randn(E, 2 * F, D, dtype=torch.bfloat16, device=dev) * 0.01
    wd = torch.randn(E, D, F, dtype=torch.bfloat16, device=dev) * 0.01
    work = torch.empty(256 * K * 2 * F, dtype=torch.float32, device=dev)
    per_expert_bytes = (2 * F * D + D * F) * 2  # gate-up + down, bf16

    print(f"{'T':>4} {'active':>6} {'grouped ms':>11} {'host ms':>9} "
          f"{'speedup':>8} {'eff TB/s':>9} {'bound ms':>9}")
    for T in (8, 32, 64, 128, 256):
        x = torch.randn(T, D, dtype=torch.bfloat16, device=dev) * 0.1
        logits = torch.randn(T, E, device=dev)
        weights, experts = torch.topk(torch.softmax(logits, -1), K, dim=-1)
        weights = weights / weights.sum(-1, keepdim=True)
        active = int(torch.unique(experts).numel())
        g_ms = bench(grouped, x, wg, wd, experts, weights, work)
        h_ms = bench(host_loop, x, wg, wd, experts, weights)
        bytes_active = active * per_expert_bytes
        bound_ms = bytes_active / 8e12 * 1e3
        eff = bytes_active / (g_ms * 1e-3) / 1e12
        print(f"{T:>4} {active:>6} {g_ms:>11.3f} {h_ms:>9.3f} "
              f"{h_ms / g_ms:>8.2f} {eff:>9.2f} {bound_ms:>9.3f}")


if __name__ == "__main__":
    main()
