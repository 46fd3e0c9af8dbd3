#!/usr/bin/env python3
"""Training demonstration / stability evidence: N updates of the flagship
config on synthetic data with a drift regime, logging the learning curve
(mean step reward, entropy, KL, clipfrac) to JSON."""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.config import DEFAULT_VALUES


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--updates", type=int, default=150)
    ap.add_argument("--n-envs", type=int, default=4096)
    ap.add_argument("--policy", default="mlp")
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    cfg = {
        **DEFAULT_VALUES,
        "data_feed_plugin": "synthetic_data_feed",
        "synthetic_rows": 262_144,
        "synthetic_seed": 77,
        "synthetic_drift": 2e-5,
        "synthetic_extra_features": 3,
        "preprocessor_plugin": "feature_window_preprocessor",
        "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE", "VOLUME",
                            "FEAT_0", "FEAT_1", "FEAT_2"],
        "n_envs": args.n_envs,
        "device": "auto",
        "autoreset": True,
        "env_start_mode": "spread",
        "position_size": 1000.0,
        "commission": 2e-5,
        "slippage": 5e-6,
        "seed": 3,
    }
    env = build_vec_environment(cfg)
    env.reset(seed=3)
    pc = PPOConfig(seed=3, policy=args.policy)
    tr = PPOTrainer(env, pc)
    hist = []
    t0 = time.perf_counter()
    for u in range(args.updates):
        s = tr.train_update()
        s["update"] = u
        s["mean_step_reward"] = float(tr.rew_buf.mean())
        s["long_frac"] = float((tr.act_buf == 1).float().mean())
        hist.append(s)
    if env.device.type == "cuda":
        torch.cuda.synchronize()
    wall = time.perf_counter() - t0
    out = {
        "updates": args.updates,
        "policy": args.policy,
        "n_envs": args.n_envs,
        "wall_seconds": wall,
        "env_steps_per_sec": args.updates * 128 * args.n_envs / wall,
        "first": hist[0], "mid": hist[len(hist) // 2], "last": hist[-1],
        "entropy_curve": [round(h["entropy"], 4) for h in hist[::10]],
        "reward_curve": [round(h["mean_step_reward"], 8) for h in hist[::10]],
        "kl_curve": [round(h["approx_kl"], 6) for h in hist[::10]],
        "vec_summary": env.vec_summary(),
        "all_finite": all(
            all(v == v and abs(v) < 1e9 for v in h.values() if isinstance(v, float))
            for h in hist),
    }
    text = json.dumps(out, indent=2)
    print(text)
    if args.out:
        Path(args.out).write_text(text)


if __name__ == "__main__":
    main()
