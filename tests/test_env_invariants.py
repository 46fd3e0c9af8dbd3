"""Behavioral smoke invariants (the reference's own checks,
tools/smoke_test.py:108-155): flat driver leaves equity exactly unchanged;
buy&hold on an uptrend earns; seeded resets reproduce the first observation;
total_return identity; spaces contain observations."""
import math

import numpy as np
import pytest

from gymfx_amd.config import DEFAULT_VALUES
from gymfx_amd.data.feed import uptrend_ohlcv, write_csv
from gymfx_amd.main import run_env


@pytest.fixture(scope="module")
def data_files(tmp_path_factory):
    d = tmp_path_factory.mktemp("data")
    from gymfx_amd.data.feed import synthetic_ohlcv

    write_csv(synthetic_ohlcv(500, seed=7, vol=2e-4), str(d / "sample.csv"))
    write_csv(uptrend_ohlcv(500), str(d / "uptrend.csv"))
    return d


def _cfg(data_files, driver, **kw):
    cfg = {
        **DEFAULT_VALUES,
        "driver_mode": driver,
        "steps": 480,
        "input_data_file": str(data_files / "sample.csv"),
        "quiet_mode": True,
    }
    cfg.update(kw)
    return cfg


def test_flat_driver_equity_unchanged(data_files):
    s = run_env(_cfg(data_files, "flat"))
    assert math.isclose(s["final_equity"], s["initial_cash"], rel_tol=1e-9, abs_tol=1e-3)
    assert math.isclose(s["total_return"], 0.0, abs_tol=1e-6)
    assert s["trades_total"] == 0


def test_buy_hold_uptrend_positive(data_files):
    s = run_env(
        _cfg(data_files, "buy_hold", input_data_file=str(data_files / "uptrend.csv"))
    )
    assert s["total_return"] > 0.0
    expected = (s["final_equity"] - s["initial_cash"]) / s["initial_cash"]
    assert math.isclose(s["total_return"], expected, rel_tol=1e-9, abs_tol=1e-9)


def test_seeded_reset_reproducible(data_files):
    from gymfx_amd import build_environment
    from gymfx_amd.plugins import load_plugin

    cfg = _cfg(data_files, "flat")

    def build():
        plugins = {}
        for group, key in [
            ("data_feed.plugins", "data_feed_plugin"),
            ("broker.plugins", "broker_plugin"),
            ("strategy.plugins", "strategy_plugin"),
            ("preprocessor.plugins", "preprocessor_plugin"),
            ("reward.plugins", "reward_plugin"),
            ("metrics.plugins", "metrics_plugin"),
        ]:
            klass, _ = load_plugin(group, cfg[key])
            plugins[key] = klass(cfg)
        return build_environment(config=cfg, **plugins)

    env_a, env_b = build(), build()
    obs_a, _ = env_a.reset(seed=123)
    obs_b, _ = env_b.reset(seed=123)
    assert set(obs_a) == set(obs_b)
    for k in obs_a:
        assert np.allclose(obs_a[k], obs_b[k]), f"seed 123 not reproducible for {k}"
    # obs are contained in the declared observation space
    assert env_a.observation_space.contains(obs_a)
    env_a.close()
    env_b.close()


def test_gym_api_shapes(data_files):
    from gymfx_amd import build_environment
    from gymfx_amd.plugins import load_plugin

    cfg = _cfg(data_files, "flat")
    plugins = {}
    for group, key in [
        ("data_feed.plugins", "data_feed_plugin"),
        ("broker.plugins", "broker_plugin"),
        ("strategy.plugins", "strategy_plugin"),
        ("preprocessor.plugins", "preprocessor_plugin"),
        ("reward.plugins", "reward_plugin"),
        ("metrics.plugins", "metrics_plugin"),
    ]:
        klass, _ = load_plugin(group, cfg[key])
        plugins[key] = klass(cfg)
    env = build_environment(config=cfg, **plugins)
    obs, info = env.reset()
    assert env.action_space.contains(1)
    assert obs["prices"].shape == (32,)
    assert obs["returns"].shape == (32,)
    for k in ("position", "equity_norm", "unrealized_pnl_norm", "steps_remaining_norm"):
        assert obs[k].shape == (1,)
        assert obs[k].dtype == np.float32
    obs2, reward, terminated, truncated, info = env.step(1)
    assert isinstance(reward, float)
    assert truncated is False
    assert "equity" in info and "action_diagnostics" in info
    env.close()


def test_continuous_action_mode(data_files):
    from gymfx_amd import build_vec_environment
    import torch

    cfg = _cfg(data_files, "flat", action_space_mode="continuous", n_envs=3, device="cpu")
    env = build_vec_environment(cfg)
    env.reset()
    out = env.step(torch.tensor([0.5, -0.5, 0.1]))
    a = out["coerced_action"]
    assert a.tolist() == [1, 2, 0]
    d = env.action_diagnostics(2)
    assert d["continuous_deadband_actions"] == 1


def test_gym_api_compliance():
    """In-house env-checker (the reference runs gymnasium's check_env,
    tools/check_gym_compliance.py; gymnasium is absent in this image)."""
    import subprocess
    import sys
    from pathlib import Path

    r = subprocess.run(
        [sys.executable, str(Path(__file__).resolve().parents[1] / "tools" /
                             "check_gym_compliance.py")],
        capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stdout + r.stderr


def test_equity_accounting_identity_property():
    """Hypothesis property over arbitrary action strings: at every step,
    equity == cash + margin + unrealized PnL, commission_paid only grows,
    and a flat position carries zero margin (the engine-side accounting
    identity of bt_bridge.py:239-251, checked continuously)."""
    import torch
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(400, seed=41, vol=6e-4)

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.integers(min_value=0, max_value=2),
                    min_size=5, max_size=60))
    def check(actions):
        env = build_vec_environment(
            {"n_envs": 2, "device": "cpu", "window_size": 8,
             "env_start_mode": "spread", "position_size": 1000.0,
             "commission": 2e-5, "slippage": 5e-6, "seed": 1}, md)
        env.reset(seed=1)
        prev_comm = 0.0
        for a in actions:
            if bool(env.st.terminated.all()):
                break
            env.step(torch.tensor([a, a], dtype=torch.int64))
            st_ = env.st
            for n in range(2):
                if bool(st_.terminated[n]):
                    continue
                bar = max(0, int(st_.cursor[n]) - 1)
                px = float(env.mt.price[min(bar, env.mt.T - 1)])
                unreal = float(st_.pos[n]) * (px - float(st_.avg_entry[n]))
                ident = float(st_.cash[n]) + float(st_.margin_used[n]) + unreal
                assert abs(ident - float(st_.equity[n])) <= 1e-6 * 10000.0
                if float(st_.pos[n]) == 0.0:
                    assert float(st_.margin_used[n]) == 0.0
            comm = float(st_.commission_paid.sum())
            assert comm >= prev_comm - 1e-12
            prev_comm = comm

    check()


def test_fused_rnn_state_reset_matches_mask_reset(data_files):
    """env.step(rnn_h=, rnn_c=) zeros exactly the terminated rows — the
    same semantics as api.mask_reset after the step (the fused form the
    recurrent rollout uses; ppo.py:_rollout_half)."""
    import torch

    from gymfx_amd import build_vec_environment
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.ops import api

    from gymfx_amd.data.feed import synthetic_ohlcv, write_csv
    tiny = data_files / "tiny.csv"
    write_csv(synthetic_ohlcv(20, seed=9, vol=2e-4), str(tiny))
    cfg = {**DEFAULT_VALUES, "n_envs": 8, "device": "cpu",
           "autoreset": True, "quiet_mode": True, "window_size": 8,
           "input_data_file": str(tiny)}
    env_a = build_vec_environment(cfg)
    env_b = build_vec_environment(cfg)
    env_a.reset(seed=3)
    env_b.reset(seed=3)
    N, H = 8, 16
    g = torch.Generator().manual_seed(5)
    h_a = torch.randn(N, H, generator=g).to(torch.bfloat16)
    c_a = torch.randn(N, H, generator=g)
    h_b, c_b = h_a.clone(), c_a.clone()
    acts = torch.ones(N, dtype=torch.int64)
    hit = False
    for _ in range(25):
        out_a = env_a.step(acts, rnn_h=h_a, rnn_c=c_a)
        out_b = env_b.step(acts)
        api.mask_reset(h_b, c_b, out_b["terminated"])
        assert torch.equal(out_a["terminated"], out_b["terminated"])
        assert torch.equal(h_a, h_b) and torch.equal(c_a, c_b)
        hit = hit or bool(out_a["terminated"].any())
    assert hit, "20-bar data must terminate at least once in 25 steps"
    assert (h_a[out_a["terminated"]] == 0).all()


def test_plugin_apply_error_graceful_fallback(data_files):
    """A strategy plugin that raises mid-run must not crash the driver:
    the step falls back to hold and the summary reports
    plugin_apply_errors (reference bt_bridge.py:191-201 semantics)."""
    from gymfx_amd.main import run_env
    from gymfx_amd.plugins.strategies import DefaultStrategy

    calls = {"n": 0}
    orig = DefaultStrategy.decide_action

    def flaky(self, obs, info, step):
        calls["n"] += 1
        if step in (2, 5):
            raise RuntimeError("plugin exploded")
        return orig(self, obs, info, step)

    DefaultStrategy.decide_action = flaky
    try:
        s = run_env(_cfg(data_files, "flat", steps=10))
    finally:
        DefaultStrategy.decide_action = orig
    assert calls["n"] == 10
    assert s["plugin_apply_errors"] == 2
    assert s["trades_total"] == 0  # fallback action is hold


def test_bracket_audit_jsonl(data_files, tmp_path):
    """GYMFX_BRACKET_AUDIT parity (direct_atr_sltp.py:40-50): bracket
    arms and exits are appended as JSONL records."""
    import json

    audit = tmp_path / "audit.jsonl"
    s = run_env(_cfg(
        data_files, "random",
        strategy_plugin="direct_fixed_sltp",
        sl_pips=2.0, tp_pips=2.0, seed=5, steps=200,
        bracket_audit_file=str(audit)))
    assert audit.exists(), "audit file must be written"
    recs = [json.loads(l) for l in audit.read_text().splitlines()]
    kinds = {r["event"] for r in recs}
    assert "bracket_armed" in kinds, kinds
    armed = [r for r in recs if r["event"] == "bracket_armed"]
    assert all("sl" in r and "tp" in r and "bar_index" in r for r in armed)
