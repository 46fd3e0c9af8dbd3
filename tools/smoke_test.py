#!/usr/bin/env python3
"""Behavioral smoke test (parity: /root/reference/tools/smoke_test.py:108-155).

Four invariants on the vectorized engine:
 1. flat driver  -> final equity EXACTLY initial_cash
 2. buy_hold on a synthetic uptrend -> positive total_return
 3. seeded reset -> bit-identical first observation
 4. total_return == (final - initial) / initial

Writes golden summaries into examples/results/ when --write-goldens.
"""
import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import synthetic_ohlcv, load_csv


def _run(md, driver, steps=400, **cfg_extra):
    cfg = {"n_envs": 1, "device": "cpu", "window_size": 8,
           "initial_cash": 10000.0, "position_size": 1000.0,
           "env_start_mode": "zero", "seed": 0}
    cfg.update(cfg_extra)
    env = build_vec_environment(cfg, md)
    env.reset(seed=0)
    rng = np.random.default_rng(0)
    first_obs = env._obs.clone()
    for i in range(steps):
        if bool(env.st.terminated[0].item()):
            break
        if driver == "flat":
            a = 0
        elif driver == "buy_hold":
            a = 1
        else:
            a = int(rng.integers(0, 3))
        env.step(torch.tensor([a], dtype=torch.int64))
    final = float(env.st.equity[0].item())
    initial = cfg["initial_cash"]
    return {
        "driver": driver,
        "final_equity": final,
        "total_return": (final - initial) / initial,
        "trades": int(env.st.trade_count[0].item()),
        "steps": int(env.st.episode_step[0].item()),
    }, first_obs


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--write-goldens", action="store_true")
    ap.add_argument("--data", default=None, help="CSV path (default: synthetic)")
    args = ap.parse_args()

    if args.data:
        md = load_csv(args.data)
    else:
        md = synthetic_ohlcv(500, seed=1, vol=3e-4)
    up = synthetic_ohlcv(500, seed=2, vol=1e-4, drift=5e-4)

    failures = []
    flat, obs_a = _run(md, "flat")
    if flat["final_equity"] != 10000.0:
        failures.append(f"flat equity changed: {flat['final_equity']}")
    bh, _ = _run(up, "buy_hold")
    if bh["total_return"] <= 0:
        failures.append(f"buy_hold on uptrend not positive: {bh['total_return']}")
    _, obs_b = _run(md, "flat")
    if not torch.equal(obs_a, obs_b):
        failures.append("seeded reset first obs not reproducible")
    rnd, _ = _run(md, "random")
    expect = (rnd["final_equity"] - 10000.0) / 10000.0
    if abs(rnd["total_return"] - expect) > 1e-12:
        failures.append("total_return identity violated")

    results = {"flat": flat, "buy_hold_uptrend": bh, "random": rnd,
               "ok": not failures, "failures": failures}
    print(json.dumps(results, indent=2))
    if args.write_goldens:
        outdir = Path(__file__).resolve().parents[1] / "examples" / "results"
        outdir.mkdir(parents=True, exist_ok=True)
        for name, summary in [("flat_summary", flat),
                              ("buy_hold_summary", bh),
                              ("random_summary", rnd)]:
            (outdir / f"{name}.json").write_text(json.dumps(summary, indent=2))
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
