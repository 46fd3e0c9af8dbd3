"""Gymnasium API contract: the ported env_checker assertions
(gymfx_amd/gym_check.py — VERDICT r1 #7) run against GymFxEnv in CI, so
any divergence from the gymnasium contract fails the suite."""
import numpy as np
import pytest

from gymfx_amd import build_environment
from gymfx_amd.config import DEFAULT_VALUES
from gymfx_amd.gym_check import check_env, data_equivalence
from gymfx_amd.plugins import load_plugin


def _env(**over):
    cfg = dict(DEFAULT_VALUES)
    cfg.update({"window_size": 8, "device": "cpu", "seed": 3,
                "data_feed_plugin": "synthetic_data_feed",
                "synthetic_rows": 400, "synthetic_seed": 3})
    cfg.update(over)
    plugins = {}
    for group, key in [("data_feed.plugins", "data_feed_plugin"),
                       ("broker.plugins", "broker_plugin"),
                       ("strategy.plugins", "strategy_plugin"),
                       ("preprocessor.plugins", "preprocessor_plugin"),
                       ("reward.plugins", "reward_plugin"),
                       ("metrics.plugins", "metrics_plugin")]:
        klass, _ = load_plugin(group, cfg[key])
        plugins[key] = klass(cfg)
    return build_environment(config=cfg, **plugins)


def test_env_passes_ported_gymnasium_checker():
    check_env(_env())


def test_env_passes_checker_continuous_actions():
    check_env(_env(action_space_mode="continuous"))


def test_checker_catches_violations():
    """The ported checker must actually detect contract breaks, not just
    pass everything (guard against a vacuous checker)."""
    env = _env()

    class BrokenReset:
        observation_space = env.observation_space
        action_space = env.action_space

        def reset(self, *, seed=None, options=None):
            return env.reset(seed=seed, options=options)[0]  # no info

        def step(self, action):
            return env.step(action)

    with pytest.raises(AssertionError, match="tuple"):
        check_env(BrokenReset())


def test_data_equivalence_semantics():
    a = {"x": np.zeros(3, dtype=np.float32), "y": 1}
    b = {"x": np.zeros(3, dtype=np.float32), "y": 1}
    assert data_equivalence(a, b)
    assert not data_equivalence(a, {"x": np.zeros(3, dtype=np.float64), "y": 1})
    assert not data_equivalence(a, {"x": np.zeros(3, dtype=np.float32), "y": 2})
