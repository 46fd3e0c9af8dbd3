"""FX rollover financing (parity: FXRolloverInterestModule wiring
nautilus_adapter.py:363-368 + bakeoff.py:179-210 overnight fixture +
monthly-rate schema bakeoff.py:104-113)."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.calendar import compute_rollover_schedule
from gymfx_amd.data.feed import synthetic_ohlcv

RATES = [
    {"LOCATION": "EA19", "TIME": "2024-01", "Value": 5.0},
    {"LOCATION": "USA", "TIME": "2024-01", "Value": 4.0},
]


def test_schedule_units_and_rate():
    # Tue 2024-01-02 21:58 / 22:01 / 22:02 UTC (the reference fixture times)
    ts = [1704232680, 1704232860, 1704232920]
    s = compute_rollover_schedule(ts, "EUR_USD", RATES)
    assert s[0] == 0.0 and s[2] == 0.0
    assert s[1] == pytest.approx((5.0 - 4.0) / 100 / 365)
    # Wednesday triple rollover
    ts_wed = [t + 86400 for t in ts]
    s3 = compute_rollover_schedule(ts_wed, "EUR_USD", RATES)
    assert s3[1] == pytest.approx(3 * (5.0 - 4.0) / 100 / 365)


def test_overnight_position_accrues_interest():
    md = synthetic_ohlcv(60 * 30, seed=2, vol=0.0,
                         start="2024-01-02 21:00:00")  # flat prices
    cfg = {
        "n_envs": 2, "device": "cpu", "window_size": 4,
        "position_size": 1000.0, "env_start_mode": "zero",
        "financing_enabled": True, "rollover_rate_data": RATES,
        "commission": 0.0, "slippage": 0.0, "seed": 0,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=0)
    # env 0 goes long before 22:00 and holds; env 1 stays flat
    acts = torch.tensor([1, 0], dtype=torch.int64)
    hold = torch.tensor([0, 0], dtype=torch.int64)
    env.step(acts)
    for _ in range(120):
        env.step(hold)
    eq0 = float(env.st.equity[0].item())
    eq1 = float(env.st.equity[1].item())
    price = float(md.columns["CLOSE"][0])
    expected = 1000.0 * price * (5.0 - 4.0) / 100 / 365
    assert eq1 == pytest.approx(10000.0)          # flat env: no financing
    assert eq0 - 10000.0 == pytest.approx(expected, rel=1e-5)


def test_financing_requires_rate_data():
    md = synthetic_ohlcv(200, seed=2)
    cfg = {"n_envs": 1, "device": "cpu", "window_size": 4,
           "financing_enabled": True}
    with pytest.raises(ValueError, match="rollover_rate_data"):
        build_vec_environment(cfg, md)


def test_engine_alias_gym_contract_with_profile_and_rate_file(tmp_path):
    """The reference's nautilus-bridge contract test
    (tests/test_nautilus_gym_bridge.py): simulation_engine='nautilus'
    (an alias of the native engine here) + an execution-cost profile +
    financing_rate_data_file CSV must still give a conforming Gymnasium
    step tuple with the usual info keys."""
    from pathlib import Path

    from gymfx_amd import build_environment
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.data.feed import write_csv
    from gymfx_amd.plugins import load_plugin

    root = Path(__file__).resolve().parents[1]
    md = synthetic_ohlcv(200, seed=9, vol=2e-4, start="2024-01-02 21:00:00")
    data = tmp_path / "px.csv"
    write_csv(md, str(data))
    cfg = {
        **DEFAULT_VALUES,
        "simulation_engine": "nautilus",
        "execution_cost_profile": str(
            root / "examples/config/execution_cost_profiles/default.json"),
        "financing_rate_data_file": str(
            root / "examples/data/fx_rollover_rates_smoke.csv"),
        "financing_enabled": True,
        "input_data_file": str(data),
        "window_size": 4, "initial_cash": 10000.0,
        "position_size": 1000.0, "quiet_mode": True,
    }
    plugins = {}
    for group, key in [("data_feed.plugins", "data_feed_plugin"),
                       ("broker.plugins", "broker_plugin"),
                       ("strategy.plugins", "strategy_plugin"),
                       ("preprocessor.plugins", "preprocessor_plugin"),
                       ("reward.plugins", "reward_plugin"),
                       ("metrics.plugins", "metrics_plugin")]:
        klass, _ = load_plugin(group, cfg[key])
        plugins[key] = klass(cfg)
    env = build_environment(config=cfg, **plugins)
    obs, info = env.reset(seed=1)
    assert env.observation_space.contains(obs)
    out = env.step(1)
    assert len(out) == 5
    obs2, reward, terminated, truncated, info2 = out
    assert isinstance(reward, float) and truncated is False
    assert "equity" in info2 and "execution_diagnostics" in info2
    env.close()
