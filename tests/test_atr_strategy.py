"""ATR-bracket strategy: warmup gating, ATR math, risk-mode SL/TP shrink
(reference oracle direct_atr_sltp.py:133-311), sizing modes, session filter."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import MarketData
from gymfx_amd.plugins.strategies import DirectAtrSLTP


def _md(n=60, price=100.0, rng_vol=1.0, seed=0, timestamps=None):
    rng = np.random.default_rng(seed)
    close = price + np.cumsum(rng.normal(0, rng_vol, n))
    open_ = np.concatenate([[price], close[:-1]])
    high = np.maximum(open_, close) + rng_vol
    low = np.minimum(open_, close) - rng_vol
    if timestamps is None:
        timestamps = 1700000000 + np.arange(n, dtype=np.int64) * 60
    return MarketData(
        columns={
            "OPEN": open_,
            "HIGH": high,
            "LOW": low,
            "CLOSE": close,
            "VOLUME": np.zeros(n),
        },
        timestamps=timestamps,
    )


BASE = {
    "n_envs": 1,
    "device": "cpu",
    "window_size": 2,
    "initial_cash": 100000.0,
    "position_size": 10.0,
    "strategy_plugin": "direct_atr_sltp",
    "atr_period": 5,
    "k_sl": 2.0,
    "k_tp": 3.0,
    "min_sltp_frac": None,
    "max_sltp_frac": None,
}


def test_atr_warmup_blocks_entries():
    env = build_vec_environment(dict(BASE), _md())
    env.reset()
    for _ in range(3):  # fewer than atr_period bars seen
        env.step(torch.tensor([1]))
    d = env.execution_diagnostics(0)
    assert d["blocked_atr_warmup"] >= 1
    assert env.bridge_state(0)["position"] == 0


def test_atr_entry_after_warmup_places_bracket():
    env = build_vec_environment(dict(BASE), _md())
    env.reset()
    for _ in range(6):
        env.step(torch.tensor([0]))
    env.step(torch.tensor([1]))
    assert env.execution_diagnostics(0)["entry_orders_submitted"] == 1
    # ATR oracle: average of the last 5 true ranges, exactly as the plugin
    st = env.st
    P = 5
    n = min(int(st.tr_count[0]), P)
    atr_vec = float(st.tr_sum[0]) / n
    sl = float(st.pend_sl[0])
    tp = float(st.pend_tp[0])
    bs = env.bridge_state(0)
    assert sl == pytest.approx(bs["price"] - 2.0 * atr_vec, rel=1e-5)
    assert tp == pytest.approx(bs["price"] + 3.0 * atr_vec, rel=1e-5)


def test_risk_mode_shrink_math_matches_plugin_oracle():
    plugin = DirectAtrSLTP()
    cfg = {
        "sltp_risk_mode": "rel_volume_aware_atr",
        "rel_volume": 0.30,
        "k_sl": 2.0,
        "k_tp": 3.0,
        "baseline_rel_volume": 0.05,
        "max_risk_rel_volume": 0.50,
    }
    k_sl_eff, k_tp_eff = plugin.effective_sltp_multiples(cfg)
    prog = (0.30 - 0.05) / (0.50 - 0.05)
    assert k_sl_eff == pytest.approx(max(1.0, 2.0 * (1 - 0.35 * prog)))
    assert k_tp_eff == pytest.approx(max(3.0 * (1 - 0.20 * prog), k_sl_eff * 1.0))
    # baseline point preserved
    cfg["rel_volume"] = 0.05
    assert plugin.effective_sltp_multiples(cfg) == (2.0, 3.0)


def test_rel_volume_sizing_uses_free_cash():
    cfg = dict(BASE)
    cfg.update({"rel_volume": 0.1, "leverage": 5.0, "min_order_volume": 0.0})
    env = build_vec_environment(cfg, _md())
    env.reset()
    for _ in range(6):
        env.step(torch.tensor([0]))
    env.step(torch.tensor([1]))
    size = float(env.st.pend_open_size[0])
    cash = float(env.st.cash[0])
    assert size == pytest.approx(cash * 0.1 * 5.0, rel=1e-6)


def test_session_filter_blocks_and_force_closes():
    # timestamps: Saturday (outside entry window Mon12:00..Fri20:00)
    sat = 1699660800  # 2023-11-11 00:00:00 UTC, Saturday
    ts = sat + np.arange(60, dtype=np.int64) * 60
    cfg = dict(BASE)
    cfg["session_filter"] = True
    env = build_vec_environment(cfg, _md(timestamps=ts))
    env.reset()
    for _ in range(8):
        env.step(torch.tensor([1]))
    d = env.execution_diagnostics(0)
    assert d["blocked_session_filter"] >= 1
    assert env.bridge_state(0)["position"] == 0


def test_sltp_frac_clamps():
    cfg = dict(BASE)
    cfg.update({"min_sltp_frac": 0.05, "max_sltp_frac": 0.08})
    env = build_vec_environment(cfg, _md(rng_vol=0.01))  # tiny ATR -> floor binds
    env.reset()
    for _ in range(6):
        env.step(torch.tensor([0]))
    env.step(torch.tensor([1]))
    bs = env.bridge_state(0)
    sl_dist = bs["price"] - float(env.st.pend_sl[0])
    assert sl_dist == pytest.approx(0.05 * bs["price"], rel=1e-5)
