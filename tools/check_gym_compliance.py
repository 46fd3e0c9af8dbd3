#!/usr/bin/env python3
"""Gym API compliance check (parity:
/root/reference/tools/check_gym_compliance.py:49-56 — the reference runs
gymnasium.utils.env_checker; gymnasium is not installable here, so this
runs the PORTED checker assertions in gymfx_amd/gym_check.py, which defer
to the real gymnasium automatically when it is importable)."""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from gymfx_amd import build_environment
from gymfx_amd.config import DEFAULT_VALUES
from gymfx_amd.gym_check import check_env as _check_env
from gymfx_amd.plugins import load_plugin


def check_env(env) -> list:
    try:
        _check_env(env)
        return []
    except AssertionError as exc:
        return [str(exc)]


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
