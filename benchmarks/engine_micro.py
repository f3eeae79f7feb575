#!/usr/bin/env python3
"""Engine microbench: isolates decode-step and prefill costs from the agent
plumbing.  Prints one JSON line with per-phase timings.

Usage: python benchmarks/engine_micro.py [--model llama-3-8b] [--batch 32]
"""

from __future__ import annotations

import argparse
import json
import statistics
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(
    __import__("os").path.abspath(__file__))))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--prompt", type=int, default=128)
    ap.add_argument("--decode-steps", type=int, default=64)
    ap.add_argument("--rounds", type=int, default=3)
    args = ap.parse_args()

    from calfkit_amd.engine.engine import GenRequest, InferenceEngine
    from calfkit_amd.models.config import get_config
    from calfkit_amd.models.llama import LlamaShard

    if not torch.cuda.is_available():
        print("needs a GPU", file=sys.stderr)
        sys.exit(1)
    cfg = get_config(args.model)
    model = LlamaShard(cfg, device="cuda")
    engine = InferenceEngine(model, max_batch=max(args.batch, 32),
                             max_model_len=1024)
    engine.start()
    try:
        # warmup round (includes graph capture)
        def round_once():
            futs = [engine.submit(GenRequest(
                prompt_ids=list(range(1, args.prompt + 1)),
                max_new_tokens=args.decode_steps))
                for _ in range(args.batch)]
            t0 = time.perf_counter()
            results = [f.result(600) for f in futs]
            wall = time.perf_counter() - t0
            return wall, results

        round_once()
        s0 = engine.stats
        walls = []
        for _ in range(args.rounds):
            wall, results = round_once()
            walls.append(wall)
        s1 = engine.stats
        dsteps = s1["decode_steps"] - s0["decode_steps"]
        dms = s1["decode_ms"] - s0["decode_ms"]
        pms = s1["prefill_ms"] - s0["prefill_ms"]
        pref = s1["prefills"] - s0["prefills"]
        # raw replay rate: GPU-only cost of one decode step (no host logic)
        raw_ms = None
        pad = engine._pad_batch(args.batch)
        g = engine._graphs.get((pad, 8)) or engine._graphs.get(pad)
        if g is not None:
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(50):
                g.replay()
            torch.cuda.synchronize()
            raw_ms = (time.perf_counter() - t0) / 50 * 1e3
        out = {
            "model": cfg.name,
            "raw_graph_replay_ms": round(raw_ms, 3) if raw_ms else None,
            "batch": args.batch,
            "wall_per_round_ms": round(statistics.median(walls) * 1e3, 1),
            "decode_ms_per_step": round(dms / max(1, dsteps), 3),
            "decode_steps": dsteps,
            "prefill_ms_per_batch": round(pms / max(1, pref) * args.batch, 2),
            "prefill_count": pref,
            "graph_replays": s1["graph_replays"] - s0["graph_replays"],
            "decode_tok_per_s": round(
                (s1["decode_tokens"] - s0["decode_tokens"]) / (dms / 1e3), 1),
        }
        print(json.dumps(out), flush=True)
    finally:
        engine.stop()


if __name__ == "__main__":
    main()
