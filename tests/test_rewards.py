"""Vectorized rewards vs the scalar reference-contract plugins, driven over
a real episode (pnl / rolling-sharpe / dd-penalized)."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.plugins.rewards import DdPenalizedReward, PnlReward, SharpeReward


@pytest.mark.parametrize(
    "reward_plugin,scalar_cls,extra",
    [
        ("pnl_reward", PnlReward, {"reward_scale": 2.0}),
        ("sharpe_reward", SharpeReward, {"window": 16}),
        ("dd_penalized_reward", DdPenalizedReward, {"penalty_lambda": 0.5}),
    ],
)
def test_vec_reward_matches_scalar_plugin(reward_plugin, scalar_cls, extra):
    md = synthetic_ohlcv(300, seed=11, vol=5e-4)
    cfg = {
        "n_envs": 1,
        "device": "cpu",
        "window_size": 8,
        "initial_cash": 10000.0,
        "position_size": 1000.0,
        "reward_plugin": reward_plugin,
        **extra,
    }
    env = build_vec_environment(cfg, md)
    env.reset()
    scalar = scalar_cls(cfg)
    rng = np.random.default_rng(5)
    for k in range(200):
        a = int(rng.integers(0, 3))
        out = env.step(torch.tensor([a]))
        bs = env.bridge_state(0)
        expected = scalar.compute_reward(
            prev_equity=bs["prev_equity"],
            new_equity=bs["equity"],
            step=bs["bar_index"],
            config=cfg,
        )
        got = float(out["reward"][0])
        assert got == pytest.approx(expected, rel=2e-4, abs=1e-7), f"step {k}"
        if bool(out["terminated"][0]):
            break
