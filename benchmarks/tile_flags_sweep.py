#!/usr/bin/env python3
"""Within-probe A/B of tile_gemm FLAGS micro-structure variants."""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import calfkit_amd.ops as ops  # noqa: E402

SHAPES = [
    ("gateup", 512, 4096, 28672, 256, 1),
    ("qkv", 512, 4096, 6144, 128, 2),
    ("o", 512, 4096, 4096, 128, 4),
    ("lmhead", 512, 4096, 128256, 256, 1),
]


def bench(fn, iters=30):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    for name, M, K, N, bn, sk in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") / K ** 0.5
        out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
        work = torch.empty(max(sk * 512 * N, 1), dtype=torch.float32,
                           device="cuda")
        ref = x.float() @ w.float().T
        res = {}
        # interleaved rounds: flags cycled inside one loop (guide rule 24)
        for rnd in range(3):
            for flags in (31, 63, 95):
                os.environ["CALF_TILE_FLAGS"] = str(flags)
                ops.tile_gemm(out, x, w, work, bn, sk)
                err = (out.float() - ref).abs().max().item()
                t = bench(lambda: ops.tile_gemm(out, x, w, work, bn, sk), 20)
                key = f"f{flags}"
                res.setdefault(key, []).append(round(t, 1))
                if err > 0.1:
                    res[key + "_ERR"] = err
        os.environ["CALF_TILE_FLAGS"] = "0"
        print(json.dumps({"shape": name, "bn": bn, "sk": sk,
                          "med": {k: sorted(v)[len(v) // 2]
                                  for k, v in res.items()
                                  if not k.endswith("_ERR")},
                          "errs": {k: v for k, v in res.items()
                                   if k.endswith("_ERR")}}), flush=True)


if __name__ == "__main__":
    main()
