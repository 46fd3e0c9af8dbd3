#!/usr/bin/env python3
"""Gym API compliance check (parity:
/root/reference/tools/check_gym_compliance.py:49-56 — the reference runs
gymnasium.utils.env_checker; gymnasium is not in this image, so this is the
equivalent in-house contract check against gymfx_amd.spaces)."""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np

from gymfx_amd import build_environment
from gymfx_amd.config import DEFAULT_VALUES
from gymfx_amd.plugins import load_plugin


def check_env(env) -> list:
    errors = []
    obs, info = env.reset(seed=3)
    if not isinstance(info, dict):
        errors.append("reset info is not a dict")
    if not env.observation_space.contains(obs):
        errors.append("reset obs not in observation_space")
    obs2, _ = env.reset(seed=3)
    for k in obs:
        if not np.array_equal(obs[k], obs2[k]):
            errors.append(f"seeded reset not deterministic for key {k}")
            break
    for _ in range(20):
        a = env.action_space.sample()
        obs, reward, terminated, truncated, info = env.step(a)
        if not isinstance(reward, float):
            errors.append(f"reward type {type(reward)} != float")
        if not isinstance(terminated, bool) or not isinstance(truncated, bool):
            errors.append("terminated/truncated not bool")
        if not env.observation_space.contains(obs):
            errors.append("step obs not in observation_space")
        if not isinstance(info, dict):
            errors.append("step info not dict")
        if errors or terminated:
            break
    return errors


def main():
    cfg = dict(DEFAULT_VALUES)
    cfg.update({"window_size": 8, "device": "cpu", "seed": 3,
                "data_feed_plugin": "synthetic_data_feed",
                "synthetic_rows": 300, "synthetic_seed": 3})
    plugins = {}
    for group, key in [("data_feed.plugins", "data_feed_plugin"),
                       ("broker.plugins", "broker_plugin"),
                       ("strategy.plugins", "strategy_plugin"),
                       ("preprocessor.plugins", "preprocessor_plugin"),
                       ("reward.plugins", "reward_plugin"),
                       ("metrics.plugins", "metrics_plugin")]:
        klass, _ = load_plugin(group, cfg[key])
        plugins[key] = klass(cfg)
    env = build_environment(
        config=cfg,
        data_feed_plugin=plugins["data_feed_plugin"],
        broker_plugin=plugins["broker_plugin"],
        strategy_plugin=plugins["strategy_plugin"],
        preprocessor_plugin=plugins["preprocessor_plugin"],
        reward_plugin=plugins["reward_plugin"],
        metrics_plugin=plugins["metrics_plugin"],
    )
    errors = check_env(env)
    print(json.dumps({"ok": not errors, "errors": errors}, indent=2))
    sys.exit(1 if errors else 0)


if __name__ == "__main__":
    main()
