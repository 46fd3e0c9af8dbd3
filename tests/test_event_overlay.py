"""Event-context execution overlay: block-new-entries and force-flat action
rewrites + diagnostics counters (reference semantics app/env.py:394-440,
test idiom from tests/test_event_context_execution_overlay.py)."""
import numpy as np
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import MarketData


def _md_with_no_trade(n, active_rows):
    prices = np.full(n, 10.0)
    nt = np.zeros(n)
    nt[list(active_rows)] = 1.0
    return MarketData(
        columns={
            "OPEN": prices.copy(),
            "HIGH": prices + 0.01,
            "LOW": prices - 0.01,
            "CLOSE": prices.copy(),
            "VOLUME": np.zeros(n),
            "event_no_trade_window_active": nt,
        },
        timestamps=1700000000 + np.arange(n, dtype=np.int64) * 60,
    )


BASE = {
    "n_envs": 1,
    "device": "cpu",
    "window_size": 2,
    "initial_cash": 10000.0,
    "position_size": 10.0,
    "event_context_execution_overlay": True,
}


def test_blocked_entry_when_flat():
    md = _md_with_no_trade(10, active_rows=range(10))
    env = build_vec_environment(dict(BASE), md)
    env.reset()
    out = env.step(torch.tensor([1]))  # entry attempt while overlay active
    assert int(out["coerced_action"][0]) == 0
    d = env.execution_diagnostics(0)
    assert d["event_context_blocked_entries"] == 1
    assert d["event_context_action_overrides"] == 1
    assert env.bridge_state(0)["position"] == 0


def test_force_flat_rewrites_to_close():
    md = _md_with_no_trade(12, active_rows=range(5, 12))
    cfg = dict(BASE)
    cfg["event_context_force_flat"] = True
    env = build_vec_environment(cfg, md)
    env.reset()
    env.step(torch.tensor([1]))   # decision bar1 (overlay inactive rows 0-4)
    env.step(torch.tensor([0]))   # fill at bar2
    assert env.bridge_state(0)["position"] == 1
    # advance until overlay row becomes active (bar_index >= 5)
    for _ in range(4):
        out = env.step(torch.tensor([0]))
    d = env.execution_diagnostics(0)
    assert d["event_context_forced_flat_actions"] >= 1
    assert d["event_context_forced_flat_orders"] >= 1
    # close order filled on the following bar
    env.step(torch.tensor([0]))
    assert env.bridge_state(0)["position"] == 0


def test_overlay_inactive_without_flag():
    md = _md_with_no_trade(10, active_rows=range(10))
    cfg = dict(BASE)
    cfg["event_context_execution_overlay"] = False
    env = build_vec_environment(cfg, md)
    env.reset()
    out = env.step(torch.tensor([1]))
    assert int(out["coerced_action"][0]) == 1
    assert env.execution_diagnostics(0)["event_context_blocked_entries"] == 0


def test_gym_wrapper_event_context_info_fields():
    """Per-step event_context_* info block (env.py:383-440 field parity)."""
    import numpy as np
    import torch

    from gymfx_amd import build_environment
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.data.feed import synthetic_ohlcv
    from gymfx_amd.plugins import load_plugin

    md = synthetic_ohlcv(200, seed=3, extra_feature_columns=1)
    md.columns["EV_NO_TRADE"] = np.ones(md.n_rows, dtype=np.float64)
    cfg = dict(DEFAULT_VALUES)
    cfg.update({
        "window_size": 8, "device": "cpu", "seed": 3,
        "data_feed_plugin": "synthetic_data_feed",
        "event_context_execution_overlay": True,
        "event_context_block_new_entries": True,
        "event_context_no_trade_column": "EV_NO_TRADE",
        "event_context_no_trade_threshold": 0.5,
    })
    plugins = {}
    for group, key in [("data_feed.plugins", "data_feed_plugin"),
                       ("broker.plugins", "broker_plugin"),
                       ("strategy.plugins", "strategy_plugin"),
                       ("preprocessor.plugins", "preprocessor_plugin"),
                       ("reward.plugins", "reward_plugin"),
                       ("metrics.plugins", "metrics_plugin")]:
        klass, _ = load_plugin(group, cfg[key])
        plugins[key] = klass(cfg)

    class MdFeed:
        plugin_params = {}
        def set_params(self, **kw): pass
        def load_data(self, config): return md
        def build_market(self, data, config): return data

    env = build_environment(
        config=cfg, data_feed_plugin=MdFeed(),
        broker_plugin=plugins["broker_plugin"],
        strategy_plugin=plugins["strategy_plugin"],
        preprocessor_plugin=plugins["preprocessor_plugin"],
        reward_plugin=plugins["reward_plugin"],
        metrics_plugin=plugins["metrics_plugin"])
    env.reset(seed=3)
    obs, r, term, trunc, info = env.step(1)  # entry attempt under no-trade
    assert info["event_context_no_trade_active"] == 1.0
    assert info["event_context_action_before_overlay"] == 1
    assert info["event_context_action_after_overlay"] == 0
    assert info["event_context_blocked_entry"] is True
    assert info["event_context_spread_stress_multiplier"] == 1.0
    # the kernel really blocked it: no position after fill bar
    env.step(0)
    assert info["event_context_position_before_overlay"] == 0
