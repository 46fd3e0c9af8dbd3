"""Bracket SL/TP lifecycle: worst-case intrabar collision ordering (the
reference's signature nautilus invariant — with a path touching both, the
long bracket must fill its STOP, not the TP: tests/test_nautilus_bakeoff.py:63-79
concept), gap-through fills at open, and TP fills."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import MarketData


def _md(bars):
    """bars: list of (open, high, low, close)."""
    arr = np.asarray(bars, dtype=np.float64)
    return MarketData(
        columns={
            "OPEN": arr[:, 0].copy(),
            "HIGH": arr[:, 1].copy(),
            "LOW": arr[:, 2].copy(),
            "CLOSE": arr[:, 3].copy(),
            "VOLUME": np.zeros(len(arr)),
        },
        timestamps=1700000000 + np.arange(len(arr), dtype=np.int64) * 60,
    )


BASE = {
    "n_envs": 1,
    "device": "cpu",
    "window_size": 2,
    "initial_cash": 10000.0,
    "position_size": 100.0,
    "commission": 0.0,
    "slippage": 0.0,
    "strategy_plugin": "direct_fixed_sltp",
    "sl_pips": 50.0,     # 0.005 at pip 0.0001
    "tp_pips": 50.0,
    "pip_size": 0.0001,
}


def _run(md, actions, **kw):
    cfg = dict(BASE)
    cfg.update(kw)
    env = build_vec_environment(cfg, md)
    env.reset()
    outs = [env.step(torch.tensor([a])) for a in actions]
    return env, outs


def test_worst_case_collision_long_fills_stop():
    # entry decision at bar1 close 1.0 -> SL 0.995, TP 1.005
    # bar2: parent fills at open 1.0 (children not active yet)
    # bar3: path touches BOTH (high 1.006, low 0.994) -> STOP must fill
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.006, 0.994, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
    ])
    env, _ = _run(md, [1, 0, 0, 0])
    d = env.execution_diagnostics(0)
    assert d["bracket_sl_fills"] == 1
    assert d["bracket_tp_fills"] == 0
    bs = env.bridge_state(0)
    assert bs["position"] == 0
    # realized = 100 * (0.995 - 1.0) = -0.5
    assert bs["equity"] == pytest.approx(10000.0 - 0.5, abs=1e-6)


def test_tp_only_fills_limit_price():
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.006, 0.998, 1.0),   # only TP touched
        (1.0, 1.001, 0.999, 1.0),
    ])
    env, _ = _run(md, [1, 0, 0])
    d = env.execution_diagnostics(0)
    assert d["bracket_tp_fills"] == 1
    assert env.bridge_state(0)["equity"] == pytest.approx(10000.0 + 100 * 0.005, abs=1e-6)


def test_gap_through_stop_fills_at_open():
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (0.990, 0.992, 0.988, 0.991),  # gaps below SL 0.995 -> fill at open 0.990
        (1.0, 1.001, 0.999, 1.0),
    ])
    env, _ = _run(md, [1, 0, 0])
    bs = env.bridge_state(0)
    assert env.execution_diagnostics(0)["bracket_sl_fills"] == 1
    assert bs["equity"] == pytest.approx(10000.0 + 100 * (0.990 - 1.0), abs=1e-6)


def test_short_bracket_mirrored():
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.006, 0.994, 1.0),   # both touched: short SL above (1.005) fills
        (1.0, 1.001, 0.999, 1.0),
    ])
    env, _ = _run(md, [2, 0, 0])
    d = env.execution_diagnostics(0)
    assert d["bracket_sl_fills"] == 1
    # short entered at 1.0, stopped at 1.005 -> loss 0.5
    assert env.bridge_state(0)["equity"] == pytest.approx(10000.0 - 0.5, abs=1e-6)


def test_children_not_active_on_entry_bar():
    # the entry bar itself touches the SL level, but children arm next bar
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.994, 1.0),   # parent fills at open; low < SL but not armed
        (1.0, 1.001, 0.9995, 1.0),
        (1.0, 1.001, 0.999, 1.0),
    ])
    env, _ = _run(md, [1, 0, 0])
    assert env.execution_diagnostics(0)["bracket_sl_fills"] == 0
    assert env.bridge_state(0)["position"] == 1


def test_agent_exit_cancels_bracket():
    md = _md([
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.001, 0.999, 1.0),
        (1.0, 1.006, 0.994, 1.0),  # would hit brackets, but position closed
        (1.0, 1.001, 0.999, 1.0),
    ])
    # enter long at step0; at step2 the overlay-style flat (action 3 via
    # event overlay is tested elsewhere) — here flip to short then back:
    env, _ = _run(md, [1, 0, 2, 0])
    d = env.execution_diagnostics(0)
    # flip closed the long before bar4; the long's brackets must be gone
    assert d["bracket_sl_fills"] + d["bracket_tp_fills"] <= 1  # only short's own brackets may fire
