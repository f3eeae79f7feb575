#!/usr/bin/env python3
"""Offline hipBLASLt algorithm search over the serving GEMM shapes.

The r32 profile (profiles/r01_bench512_kernelstats.md) shows the library
GEMMs at ~50-60 % MFU on the default heuristic picks — this sweep times
EVERY supported algorithm per shape through the in-tree binding
(ops._C.hipblaslt_algo_bench) and writes the winners' stable indices to
``profiles/gemm_algo_table.json``.  A dispatch layer can then replay the
winners with ops._C.hipblaslt_gemm_indexed (round-2 wiring).

Run on a GPU box:  python benchmarks/gemm_algo_search.py [--iters 20]
"""

import argparse
import json
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

# (label, M, N, K) — llama-3-8b serving shapes at the default operating
# point (decode batch 512, prefill chunks ~4k rows) plus the TP=8 70B
# decode shards and Mixtral prefill experts.
SHAPES = [
    ("8b decode qkv",    512, 6144, 4096),
    ("8b decode o",      512, 4096, 4096),
    ("8b decode gateup", 512, 28672, 4096),
    ("8b decode down",   512, 4096, 14336),
    ("8b decode lm_head", 512, 128256, 4096),
    ("8b prefill qkv",   4096, 6144, 4096),
    ("8b prefill gateup", 4096, 28672, 4096),
    ("8b prefill down",  4096, 4096, 14336),
    ("70b tp8 decode qkv", 512, 1280, 8192),
    ("70b tp8 decode gateup", 512, 7168, 8192),
    ("70b tp8 decode down", 512, 8192, 3584),
]


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--max-algos", type=int, default=200)
    ap.add_argument("--out", default="profiles/gemm_algo_table.json")
    args = ap.parse_args()

    assert torch.cuda.is_available(), "GPU required"
    from calfkit_amd import ops
    assert ops.have_extension()
    from calfkit_amd.ops import _C  # type: ignore

    table = {}
    for label, M, N, K in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") / math.sqrt(K)
        out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
        # default heuristic pick = what F.linear effectively uses
        ref = torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        import time
        for _ in range(3):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            torch.nn.functional.linear(x, w)
        torch.cuda.synchronize()
        default_ms = (time.perf_counter() - t0) / args.iters * 1e3

        timed = _C.hipblaslt_algo_bench(out, x, w, args.iters, args.max_algos)
        if not timed:
            print(f"{label}: no supported algos?!")
            continue
        timed.sort(key=lambda t: t[1])
        best_idx, best_ms = timed[0]
        # numerics check on the winner
        ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
        _C.hipblaslt_gemm_indexed(out, x, w, best_idx, ws)
        torch.cuda.synchronize()
        ok = torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2)
        tflops = 2 * M * N * K / (best_ms * 1e-3) / 1e12
        print(f"{label:24s} default {default_ms:8.3f} ms | best algo "
              f"{best_idx:6d} {best_ms:8.3f} ms ({default_ms/best_ms:5.2f}x, "
              f"{tflops:7.1f} TF/s) [{len(timed)} algos] "
              f"{'OK' if ok else 'NUMERICS MISMATCH'}")
        table[label] = {"M": M, "N": N, "K": K, "algo_index": int(best_idx),
                        "best_ms": round(best_ms, 4),
                        "default_ms": round(default_ms, 4),
                        "speedup": round(default_ms / best_ms, 3),
                        "numerics_ok": bool(ok)}

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(table, f, indent=1)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
