"""Deterministic replay + independent reconciliation oracle
(capability parity: simulation_engines/nautilus_adapter.py event-log hashes,
bakeoff.py:228-303 reconcile_fills, tests/test_nautilus_bakeoff.py idioms)."""
import numpy as np
import pytest

from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.replay import ReplayAdapter


def _cfg(**kw):
    cfg = {
        "window_size": 8,
        "initial_cash": 10000.0,
        "position_size": 1000.0,
        "commission": 2e-5,
        "slippage": 5e-6,
        "device": "cpu",
        "seed": 0,
    }
    cfg.update(kw)
    return cfg


def _actions(n, seed=3):
    rng = np.random.default_rng(seed)
    return rng.integers(0, 3, size=n).tolist()


def test_replay_reconciles_direct_strategy():
    md = synthetic_ohlcv(400, seed=9, vol=3e-4)
    res = ReplayAdapter().run(_cfg(), md, _actions(300))
    assert res["reconciled"], res["reconciliation"]
    assert res["steps"] == 300
    fills = [e for e in res["events"] if e["type"] == "order_filled"]
    assert len(fills) >= 2
    assert res["engine"]["trade_count"] == res["oracle"]["trade_count"]


def test_replay_reconciles_bracket_strategy():
    md = synthetic_ohlcv(400, seed=10, vol=6e-4)
    res = ReplayAdapter().run(
        _cfg(strategy_plugin="direct_fixed_sltp", sl_pips=8.0, tp_pips=16.0),
        md, _actions(300, seed=4))
    assert res["reconciled"], res["reconciliation"]
    kinds = {e["type"] for e in res["events"]}
    assert "bracket_sl_fill" in kinds or "bracket_tp_fill" in kinds


def test_replay_hash_determinism_and_sensitivity():
    md = synthetic_ohlcv(300, seed=11, vol=4e-4)
    acts = _actions(200, seed=5)
    r1 = ReplayAdapter().run(_cfg(), md, acts)
    r2 = ReplayAdapter().run(_cfg(), md, acts)
    assert r1["event_hash"] == r2["event_hash"]
    assert r1["result_hash"] == r2["result_hash"]
    acts2 = list(acts)
    acts2[50] = (acts2[50] + 1) % 3
    r3 = ReplayAdapter().run(_cfg(), md, acts2)
    assert r3["result_hash"] != r1["result_hash"]


def test_replay_no_future_leakage():
    """Mutating bars strictly after the replayed range must not change the
    result (test_nautilus_bakeoff.py:124-156 idiom)."""
    md = synthetic_ohlcv(400, seed=12, vol=4e-4)
    acts = _actions(100, seed=6)
    r1 = ReplayAdapter().run(_cfg(), md, acts)
    for col in ("OPEN", "HIGH", "LOW", "CLOSE"):
        md.columns[col] = md.columns[col].copy()
        md.columns[col][-50:] *= 5.0
    r2 = ReplayAdapter().run(_cfg(), md, acts)
    assert r1["event_hash"] == r2["event_hash"]
    assert r1["engine"] == r2["engine"]


@pytest.mark.parametrize("trial", range(8))
def test_replay_reconciles_under_random_configs(trial):
    """Property sweep: random market/cost/strategy configs must all
    reconcile engine vs the independent ledger (robustness currency of
    the reference's bakeoff)."""
    rng = np.random.default_rng(100 + trial)
    md = synthetic_ohlcv(int(rng.integers(150, 400)),
                         seed=int(rng.integers(0, 1000)),
                         vol=float(rng.uniform(1e-4, 1e-3)),
                         drift=float(rng.uniform(-2e-4, 2e-4)))
    cfg = _cfg(
        commission=float(rng.choice([0.0, 1e-5, 5e-5])),
        slippage=float(rng.choice([0.0, 5e-6, 2e-5])),
        position_size=float(rng.choice([100.0, 1000.0, 5000.0])),
        leverage=float(rng.choice([1.0, 10.0, 50.0])),
    )
    if rng.random() < 0.5:
        cfg.update(strategy_plugin="direct_fixed_sltp",
                   sl_pips=float(rng.uniform(2, 15)),
                   tp_pips=float(rng.uniform(2, 25)))
    acts = rng.integers(0, 3, size=int(rng.integers(80, 250))).tolist()
    res = ReplayAdapter().run(cfg, md, acts)
    assert res["reconciled"], (trial, res["reconciliation"],
                               res["engine"], res["oracle"])


def test_replay_reconciles_hypothesis_fuzz():
    """Hypothesis-driven action-sequence fuzz: the engine and the
    independent ledger must reconcile for ARBITRARY action strings
    (incl. pathological runs of flips / force-closes), not just uniform
    random ones."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    md = synthetic_ohlcv(300, seed=77, vol=6e-4)

    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.integers(min_value=0, max_value=3),
                    min_size=10, max_size=120),
           st.booleans())
    def check(actions, brackets):
        cfg = _cfg()
        if brackets:
            cfg.update(strategy_plugin="direct_fixed_sltp",
                       sl_pips=8.0, tp_pips=12.0)
        res = ReplayAdapter().run(cfg, md, actions)
        assert res["reconciled"], (actions[:20], res["reconciliation"])

    check()
