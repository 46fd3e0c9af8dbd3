#!/usr/bin/env python3
"""Engine wall-clock benchmark (parity:
/root/reference/tools/simulation_engine_benchmark.py:84-128 — fresh-run
wall-clock + RSS, schema simulation_engine_benchmark.v1-compatible).

Benchmarks the vectorized engine on CPU (and GPU if available) at several
env counts; a "run" is a fresh env + reset + `steps` vectorized steps.
"""
import argparse
import json
import resource
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import synthetic_ohlcv


def bench(n_envs, steps, runs, device):
    md = synthetic_ohlcv(steps + 64, seed=1, vol=3e-4)
    samples = []
    for r in range(runs):
        t0 = time.perf_counter()
        cfg = {"n_envs": n_envs, "device": device, "window_size": 8,
               "env_start_mode": "zero", "autoreset": True,
               "position_size": 1000.0, "seed": r}
        env = build_vec_environment(cfg, md)
        env.reset(seed=r)
        actions = torch.zeros(n_envs, dtype=torch.int64, device=env.device)
        for i in range(steps):
            actions.fill_(i % 3)
            env.step(actions)
        if env.device.type == "cuda":
            torch.cuda.synchronize()
        samples.append(time.perf_counter() - t0)
    return {
        "n_envs": n_envs, "steps": steps, "runs": runs, "device": device,
        "mean_seconds": statistics.mean(samples),
        "median_seconds": statistics.median(samples),
        "min_seconds": min(samples), "max_seconds": max(samples),
        "runs_per_second": 1.0 / statistics.mean(samples),
        "env_steps_per_second": n_envs * steps / statistics.mean(samples),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--runs", type=int, default=3)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    results = []
    for n in (1, 64, 1024):
        results.append(bench(n, args.steps, args.runs, "cpu"))
    if torch.cuda.is_available():
        for n in (1024, 4096, 16384):
            results.append(bench(n, args.steps, args.runs, "cuda"))
    out = {
        "schema": "simulation_engine_benchmark.v1",
        "engine": "gymfx_amd.vectorized",
        "results": results,
        "max_rss_mb": resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024,
        "note": ("fresh-run wall-clock incl. env construction; "
                 "synthetic OHLCV; cycling hold/long/short driver"),
    }
    text = json.dumps(out, indent=2)
    print(text)
    if args.out:
        Path(args.out).write_text(text)


if __name__ == "__main__":
    main()
