"""Stage-B force-close reward penalty (reference
tests/test_force_close_reward_penalty.py semantics, app/env.py:639-665):
an open position inside the pre-Friday-close window (or the force-close
zone itself) pays coef x |sign(position)| per step; flat positions,
out-of-window bars, and the disabled config pay nothing."""
import numpy as np
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import MarketData

COEF = 2e-4
DAY = 86400
# 2024-01-05 is a Friday; epoch day of 2024-01-05 00:00 UTC
FRI = 1704412800


def _md_at(start_epoch, n=8, step_s=3600):
    px = np.full(n, 1.1)
    return MarketData(
        columns={"OPEN": px.copy(), "HIGH": px + 1e-4, "LOW": px - 1e-4,
                 "CLOSE": px.copy(), "VOLUME": np.zeros(n)},
        timestamps=start_epoch + np.arange(n, dtype=np.int64) * step_s,
    )


def _env(md, **over):
    cfg = {"n_envs": 1, "device": "cpu", "window_size": 2,
           "initial_cash": 10_000.0, "position_size": 100.0,
           "commission": 0.0, "slippage": 0.0,
           "timeframe_hours": 1.0,
           "stage_b_force_close_obs": True,
           "stage_b_force_close_reward_penalty": True,
           "force_close_exposure_penalty_coef": COEF,
           "force_close_exposure_penalty_window_hours": 4.0,
           "force_close_dow": 4, "force_close_hour": 20}
    cfg.update(over)
    env = build_vec_environment(cfg, md)
    env.reset()
    return env


def _step_penalties(env, actions):
    pens = []
    for a in actions:
        out = env.step(torch.tensor([a]))
        pens.append(float(out["force_close_reward_penalty"][0]))
    return pens


def test_penalty_applies_before_friday_close():
    # bars start Friday 15:00 UTC; entry fills, then holding at 17:00+
    # is within 4h of the 20:00 close -> penalized
    env = _env(_md_at(FRI + 15 * 3600))
    pens = _step_penalties(env, [1, 1, 1, 1])
    assert pens[0] == 0.0            # entry pending, still flat
    assert any(p == COEF for p in pens[1:]), pens


def test_penalty_applies_inside_force_close_zone():
    # bars start Friday 20:00 UTC (inside the close zone); short position
    env = _env(_md_at(FRI + 20 * 3600))
    pens = _step_penalties(env, [2, 2, 2])
    assert pens[-1] == COEF, pens


def test_penalty_skips_flat_or_outside_window():
    # flat all the way inside the window -> zero
    env = _env(_md_at(FRI + 15 * 3600))
    assert _step_penalties(env, [0, 0, 0]) == [0.0, 0.0, 0.0]
    # long position on Wednesday noon (days from the window) -> zero
    env2 = _env(_md_at(FRI - 2 * DAY))
    assert all(p == 0.0 for p in _step_penalties(env2, [1, 1, 1]))


def test_penalty_is_config_gated():
    env = _env(_md_at(FRI + 15 * 3600),
               stage_b_force_close_reward_penalty=False)
    assert all(p == 0.0 for p in _step_penalties(env, [1, 1, 1, 1]))


def test_penalty_reduces_published_reward():
    """reward = base_reward - penalty (env.step contract; the penalty is
    also published separately so training metrics can split it out)."""
    env = _env(_md_at(FRI + 15 * 3600))
    env.step(torch.tensor([1]))
    out = env.step(torch.tensor([1]))
    r = float(out["reward"][0])
    b = float(out["base_reward"][0])
    p = float(out["force_close_reward_penalty"][0])
    assert p == COEF
    assert abs((b - p) - r) < 1e-12
