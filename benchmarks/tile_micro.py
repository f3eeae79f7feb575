#!/usr/bin/env python3
"""Tiled serving-batch GEMM (ops.tile_gemm) vs hipBLASLt on the llama
decode shapes at the 384/512 graph batches.

Prints one JSON line per shape: tile time, library time (heuristic and
searched-algo when profiles/gemm_algo_table.json has the shape), and the
achieved TFLOP/s.  --sweep additionally tries BN/splitk variants.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import calfkit_amd.ops as ops  # noqa: E402

SHAPES = [  # (name, M, K, N)
    ("qkv_8b_m512", 512, 4096, 6144),
    ("o_8b_m512", 512, 4096, 4096),
    ("gateup_8b_m512", 512, 4096, 28672),
    ("down_8b_m512", 512, 14336, 4096),
    ("lmhead_8b_m512", 512, 4096, 128256),
    ("qkv_8b_m384", 384, 4096, 6144),
    ("o_8b_m384", 384, 4096, 4096),
    ("gateup_8b_m384", 384, 4096, 28672),
    ("down_8b_m384", 384, 14336, 4096),
    ("o_8b_tp8_m512", 512, 512, 4096),
    ("down_8b_tp8_m512", 512, 1792, 4096),
    ("qkv_70b_tp8_m512", 512, 8192, 1280),
    ("o_70b_tp8_m512", 512, 1024, 8192),
    ("gateup_70b_tp8_m512", 512, 8192, 7168),
    ("down_70b_tp8_m512", 512, 3584, 8192),
]


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def algo_table() -> dict:
    path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "profiles", "gemm_algo_table.json")
    try:
        with open(path) as f:
            return {(v["M"], v["N"], v["K"]): v["algo_index"]
                    for v in json.load(f).values()}
    except Exception:
        return {}


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--sweep", action="store_true")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    algos = algo_table()
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    from calfkit_amd.ops import _K

    for name, M, K, N in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        w /= K ** 0.5
        out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
        out2 = torch.empty_like(out)
        flops = 2.0 * M * N * K

        t_lib = bench(lambda: torch.nn.functional.linear(x, w, out=out2),
                      args.iters)
        t_idx = None
        if (M, N, K) in algos:
            idx = algos[(M, N, K)]
            try:
                _K.hipblaslt_gemm_indexed(out2, x, w, idx, ws)
                t_idx = bench(
                    lambda: _K.hipblaslt_gemm_indexed(out2, x, w, idx, ws),
                    args.iters)
            except Exception:
                t_idx = None

        plans = []
        base = ops.tile_plan(M, N, K)
        if base:
            plans.append(base)
        if args.sweep:
            for bn in (256, 128):
                for sk in (1, 2, 4, 8):
                    if N % bn == 0 and K % sk == 0 and (K // sk) % 64 == 0 \
                            and (bn, sk) not in plans:
                        plans.append((bn, sk))
        best = None
        results = {}
        for bn, sk in plans:
            Mp = ((M + 255) // 256) * 256
            work = torch.empty(max(sk * Mp * N, 1), dtype=torch.float32,
                               device="cuda")
            ops.tile_gemm(out, x, w, work, bn, sk)
            err = (out.float() - x.float() @ w.float().T).abs().max().item()
            t = bench(lambda: ops.tile_gemm(out, x, w, work, bn, sk),
                      args.iters)
            results[f"bn{bn}_sk{sk}"] = round(t, 1)
            if err > 0.1:
                results[f"bn{bn}_sk{sk}_ERR"] = err
                continue
            if best is None or t < best[2]:
                best = (bn, sk, t)
        t_small = None
        if N % 128 == 0 and K % 64 == 0:
            ops.tile_gemm_small(out, x, w)
            err_s = (out.float() - x.float() @ w.float().T).abs().max().item()
            if err_s <= 0.1:
                t_small = bench(lambda: ops.tile_gemm_small(out, x, w),
                                args.iters)
            else:
                results["small_ERR"] = err_s
        line = {
            "shape": name, "M": M, "K": K, "N": N,
            "lib_us": round(t_lib, 1),
            "small_us": round(t_small, 1) if t_small else None,
            "small_vs_lib": round(t_lib / t_small, 3) if t_small else None,
            "lib_indexed_us": round(t_idx, 1) if t_idx else None,
            "tile_us": round(best[2], 1) if best else None,
            "tile_plan": f"bn{best[0]}_sk{best[1]}" if best else None,
            "tile_tf": round(flops / best[2] / 1e6, 1) if best else None,
            "lib_tf": round(flops / t_lib / 1e6, 1),
            "speedup_vs_lib": round(t_lib / best[2], 3) if best else None,
            "speedup_vs_indexed": (round(t_idx / best[2], 3)
                                   if best and t_idx else None),
        }
        if args.sweep:
            line["variants"] = results
        print(json.dumps(line), flush=True)


if __name__ == "__main__":
    main()
