"""Broker fill arithmetic against hand-computed ledgers: next-bar-open
fills, % slippage per fill, % commission on notional, long<->short flip =
two commissions, leverage margin ledger, min_equity bust
(reference semantics: broker_plugins/default_broker.py + bt_bridge.py:203-237)."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import MarketData


def _flat_market(prices, highs=None, lows=None, opens=None):
    prices = np.asarray(prices, dtype=np.float64)
    n = len(prices)
    return MarketData(
        columns={
            "OPEN": np.asarray(opens, dtype=np.float64) if opens is not None else prices.copy(),
            "HIGH": np.asarray(highs, dtype=np.float64) if highs is not None else prices + 1e-9,
            "LOW": np.asarray(lows, dtype=np.float64) if lows is not None else prices - 1e-9,
            "CLOSE": prices.copy(),
            "VOLUME": np.zeros(n),
        },
        timestamps=1700000000 + np.arange(n, dtype=np.int64) * 60,
    )


BASE = {
    "n_envs": 1,
    "device": "cpu",
    "window_size": 4,
    "initial_cash": 1000.0,
    "position_size": 10.0,
    "commission": 0.001,
    "slippage": 0.0,
    "leverage": 1.0,
}


def _env(md, **kw):
    cfg = dict(BASE)
    cfg.update(kw)
    env = build_vec_environment(cfg, md)
    env.reset()
    return env


def step(env, a):
    return env.step(torch.tensor([a]))


def test_buy_fill_next_bar_open_with_commission():
    # bars: close = [10,10,11,12,...]; open[t] == close[t]
    md = _flat_market([10.0, 10.0, 11.0, 12.0, 13.0, 14.0, 15.0, 16.0])
    env = _env(md)
    step(env, 1)           # decision on bar 1 (no advance: first step)
    step(env, 0)           # bar 2 (row 1): fill at open=10.0, close=10.0
    bs = env.bridge_state(0)
    # commission = 10 units * 10.0 * 0.001 = 0.1
    assert bs["commission_paid"] == pytest.approx(0.1)
    assert bs["equity"] == pytest.approx(1000.0 - 0.1)
    assert bs["position"] == 1
    step(env, 0)           # bar 3 (row 2): close=11.0, mark-to-market
    bs = env.bridge_state(0)
    assert bs["equity"] == pytest.approx(1000.0 - 0.1 + 10 * (11.0 - 10.0))


def test_slippage_applied_per_side():
    md = _flat_market([10.0] * 8)
    env = _env(md, slippage=0.01, commission=0.0)
    step(env, 1)
    step(env, 0)  # buy fills at 10 * 1.01 = 10.1
    bs = env.bridge_state(0)
    assert bs["equity"] == pytest.approx(1000.0 + 10 * (10.0 - 10.1))
    step(env, 2)  # decision: flip to short
    step(env, 0)  # close fills at 10*0.99 (sell), open short at 10*0.99
    bs = env.bridge_state(0)
    # realized on close: 10*(9.9 - 10.1) = -2.0 ; short entry at 9.9
    # short unrealized at close 10.0: -10*(10.0-9.9) = -1.0
    assert bs["equity"] == pytest.approx(1000.0 - 2.0 - 1.0)
    assert bs["position"] == -1


def test_flip_charges_two_commissions():
    md = _flat_market([10.0] * 10)
    env = _env(md, commission=0.001)
    step(env, 1)
    step(env, 0)   # long open: comm 0.1
    step(env, 2)   # decision flip
    step(env, 0)   # close (comm 0.1) + open short (comm 0.1)
    bs = env.bridge_state(0)
    assert bs["commission_paid"] == pytest.approx(0.3)
    assert bs["trade_count"] == 1
    d = env.execution_diagnostics(0)
    assert d["default_orders_submitted"] == 3  # 1 open + (close+open)


def test_leverage_margin_ledger_and_free_cash():
    md = _flat_market([100.0] * 8)
    env = _env(md, leverage=10.0, commission=0.0, position_size=50.0,
               initial_cash=1000.0)
    step(env, 1)
    step(env, 0)  # buy 50 units @100 -> notional 5000, margin 500
    st = env.st
    assert float(st.margin_used[0]) == pytest.approx(500.0)
    assert float(st.cash[0]) == pytest.approx(500.0)
    bs = env.bridge_state(0)
    assert bs["equity"] == pytest.approx(1000.0)  # no price move, no commission


def test_min_equity_bust_terminates():
    # price collapses; long position busts through min_equity
    closes = [100.0, 100.0, 100.0, 5.0, 5.0, 5.0]
    md = _flat_market(closes)
    env = _env(md, position_size=11.0, commission=0.0, min_equity=100.0,
               initial_cash=1000.0)
    step(env, 1)
    step(env, 0)   # row1: fill at 100
    step(env, 0)   # row2: close still 100
    out = step(env, 0)  # row3: close 5 -> equity = 1000 + 11*(5-100) = -45
    assert bool(out["terminated"][0])
    bs = env.bridge_state(0)
    assert bs["equity"] == pytest.approx(1000.0 + 11 * (5.0 - 100.0))


def test_hold_same_direction_no_new_orders():
    md = _flat_market([10.0] * 8)
    env = _env(md, commission=0.001)
    step(env, 1)
    step(env, 1)
    step(env, 1)
    bs = env.bridge_state(0)
    assert bs["commission_paid"] == pytest.approx(0.1)  # only one open
    assert bs["position"] == 1


def test_data_exhaustion_terminates():
    md = _flat_market([10.0] * 6)  # T=6
    env = _env(md, window_size=4)
    step(env, 0)
    terminated = False
    for _ in range(10):
        out = step(env, 0)
        if bool(out["terminated"][0]):
            terminated = True
            break
    assert terminated
    assert env.bridge_state(0)["bar_index"] == 6


def test_margin_preflight_denies_oversized_order():
    """enforce_margin_preflight: an order whose margin+commission exceeds
    free cash is denied with a diagnostics count, the episode continues
    (nautilus_gym.py:128-171 / bakeoff.py:166-176 margin-rejection fixture)."""
    import torch

    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(100, seed=4, start_price=1.1)
    base = {"n_envs": 1, "device": "cpu", "window_size": 4,
            "initial_cash": 100.0, "position_size": 1_000_000.0,
            "leverage": 1.0, "env_start_mode": "zero", "seed": 0}
    # preflight ON: denied
    env = build_vec_environment({**base, "enforce_margin_preflight": True}, md)
    env.reset(seed=0)
    env.step(torch.tensor([1]))
    env.step(torch.tensor([0]))   # fill bar
    assert float(env.st.pos[0]) == 0.0
    assert env.execution_diagnostics(0)["margin_preflight_denied"] == 1
    assert not bool(env.st.terminated[0])
    # preflight OFF: fills (cash goes deeply negative -> bust path)
    env2 = build_vec_environment(base, md)
    env2.reset(seed=0)
    env2.step(torch.tensor([1]))
    env2.step(torch.tensor([0]))
    assert float(env2.st.pos[0]) != 0.0
