"""Bitrot guards: the BASELINE benchmark scripts run end-to-end (tiny
settings, CPU) and print their JSON metric line.  These scripts ARE the
measurement deliverable — an API drift that breaks one must fail here,
not on the GPU box."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # repo root


def _run(script: str, env: dict, timeout: int = 300) -> dict:
    e = dict(os.environ, **env)
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "benchmarks", script)],
        capture_output=True, text=True, timeout=timeout, env=e, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, out.stdout[-2000:]
    return json.loads(lines[-1])


@pytest.mark.slow
def test_config1_plumbing_script():
    r = _run("config1_plumbing.py", {"C1_TURNS": "24", "C1_CONCURRENCY": "8"})
    assert r["unit"] == "turns/s"
    # catastrophic-regression floor only (measured ~1900 turns/s; a
    # machine-speed-safe 20x margin below the slowest plausible CI box)
    assert r["value"] > 20, r


@pytest.mark.slow
def test_config5_moe_team_script_cpu():
    r = _run("config5_moe_team.py", {"C5_STEPS": "1", "C5_WARMUP": "0"})
    assert r["unit"] == "team-turns/s" and r["value"] > 0


@pytest.mark.slow
def test_bench_default_contract_cpu():
    """bench.py sanity mode: one JSON line with every contract field."""
    r = _run("../bench.py", {"CALF_BENCH_TURNS": "8", "CALF_BENCH_PROMPT": "32",
                             "CALF_BENCH_DECODE": "8"})
    # driver-contract fields
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                  "dtype", "data", "config"):
        assert field in r, field
    assert r["n_gpus"] == 1 and r["scaling"] == "weak"


@pytest.mark.slow
def test_config3_toolcalls_script_cpu():
    r = _run("config3_toolcalls.py",
             {"C3_TURNS": "6", "C3_STEPS": "1", "C3_WARMUP": "0"})
    assert r["unit"] == "turns/s" and r["value"] > 0


@pytest.mark.slow
@pytest.mark.dist
def test_config4_tp_script_cpu_world2():
    """config4 (TP lockstep serving) under torchrun world=2 on gloo."""
    env = dict(os.environ, C4_TURNS="4", C4_STEPS="1", C4_WARMUP="0")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29547", os.path.join(ROOT, "benchmarks",
                                                "config4_tp70b.py")],
        capture_output=True, text=True, timeout=420, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, out.stdout[-2000:]
    r = json.loads(lines[-1])
    assert r["value"] > 0 and r["n_gpus"] == 2
