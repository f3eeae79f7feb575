#!/usr/bin/env python3
"""Standalone loop of one tile_gemm shape (for rocprofv3 PMC runs) with an
optional hipBLASLt arm for side-by-side counter comparison."""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import calfkit_amd.ops as ops  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--shape", default="512,4096,28672")  # M,K,N (gateup)
    ap.add_argument("--bn", type=int, default=256)
    ap.add_argument("--sk", type=int, default=1)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--lib", action="store_true")
    args = ap.parse_args()
    M, K, N = map(int, args.shape.split(","))
    torch.manual_seed(0)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") / K ** 0.5
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    Mp = ((M + 255) // 256) * 256
    work = torch.empty(max(args.sk * Mp * N, 1), dtype=torch.float32,
                       device="cuda")
    for _ in range(args.iters):
        if args.lib:
            torch.nn.functional.linear(x, w, out=out)
        else:
            ops.tile_gemm(out, x, w, work, args.bn, args.sk)
    torch.cuda.synchronize()
    print("ok")


if __name__ == "__main__":
    main()
