"""Calendar: scalar predicates (reference semantics app/oanda_calendar.py)
and scalar-vs-vectorized-table equivalence across DST boundaries."""
import datetime as dt

import numpy as np
import pytest

from gymfx_amd.calendar import (
    CALENDAR_FEATURE_KEYS,
    broker_market_open,
    compute_fx_calendar_features,
    compute_fx_calendar_table,
    is_force_flat_window,
    is_no_new_position_window,
    _to_ny,
)

NY = dt.timezone.utc  # inputs below are UTC epoch values


def _epoch(y, m, d, hh, mm):
    return int(dt.datetime(y, m, d, hh, mm, tzinfo=dt.timezone.utc).timestamp())


def test_scalar_predicates():
    # 2024-01-05 is a Friday. 19:30 UTC = 14:30 NY (EST, UTC-5).
    ts = _to_ny(_epoch(2024, 1, 5, 19, 30))
    assert ts.weekday() == 4
    assert is_no_new_position_window(ts)
    assert not is_force_flat_window(ts)
    # 20:50 UTC = 15:50 NY -> force flat
    ts2 = _to_ny(_epoch(2024, 1, 5, 20, 50))
    assert is_force_flat_window(ts2)
    # Saturday closed
    assert not broker_market_open(_to_ny(_epoch(2024, 1, 6, 12, 0)))
    # Sunday 22:30 UTC = 17:30 NY -> open
    assert broker_market_open(_to_ny(_epoch(2024, 1, 7, 22, 30)))
    # Monday daily break 21:59-22:05 UTC (EST): 22:00 UTC = 17:00 NY
    assert not broker_market_open(_to_ny(_epoch(2024, 1, 8, 22, 0)))


@pytest.mark.parametrize(
    "start",
    [
        _epoch(2024, 1, 3, 0, 0),    # EST
        _epoch(2024, 3, 9, 12, 0),   # spring-forward DST transition window
        _epoch(2024, 11, 2, 12, 0),  # fall-back DST transition window
        _epoch(2024, 7, 1, 0, 0),    # EDT
    ],
)
def test_table_matches_scalar(start):
    ts = start + np.arange(0, 96 * 3600, 1800, dtype=np.int64)  # 4 days, 30-min grid
    table = compute_fx_calendar_table(ts, timeframe_hours=1.0)
    for i in range(0, len(ts), 7):
        scalar = compute_fx_calendar_features(int(ts[i]), timeframe_hours=1.0)
        for j, key in enumerate(CALENDAR_FEATURE_KEYS):
            assert table[i, j] == pytest.approx(scalar[key], abs=1e-4), (
                f"{key} mismatch at ts={ts[i]}"
            )


def test_neutral_on_unparseable():
    out = compute_fx_calendar_features("not-a-date", timeframe_hours=1.0)
    assert all(v == 0.0 for v in out.values())


def test_calendar_feature_ranges_property():
    """Hypothesis property over arbitrary epoch timestamps: every feature
    is finite and in range (binary flags in {0,1}, countdowns >= 0, the
    force-flat window implies the no-new-position window)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from gymfx_amd.calendar import CALENDAR_FEATURE_KEYS, compute_fx_calendar_features

    flags = {"is_friday_risk_reduction_window", "is_no_new_position_window",
             "is_force_flat_window", "is_broker_daily_break_near",
             "broker_market_open", "is_no_trade_window"}

    @settings(max_examples=120, deadline=None)
    @given(st.integers(min_value=0, max_value=2_500_000_000))
    def check(ts):
        f = compute_fx_calendar_features(ts, timeframe_hours=1.0)
        for k in CALENDAR_FEATURE_KEYS:
            v = f[k]
            assert v == v and abs(v) != float("inf"), (k, v)
            if k in flags:
                assert v in (0.0, 1.0), (k, v)
            else:
                assert v >= 0.0, (k, v)
        if f["is_force_flat_window"]:
            assert f["is_no_new_position_window"] == 1.0

    check()


def test_window_start_boundaries_exact_ny_times():
    """The reference pins its Friday windows to exact NY wall-clock starts
    (oanda_calendar.py:30-48; tests test_friday_*_window_starts_at_*):
    no-new-position at 14:00, risk-reduction at 15:00, force-flat at
    15:45 — one minute before each start the flag must be off, at the
    start it must be on.  Checked in BOTH EST (Jan, UTC-5) and EDT
    (Jul, UTC-4) so a fixed-UTC-offset implementation fails."""
    from gymfx_amd.calendar import (is_broker_daily_break_near,
                                    is_friday_risk_reduction_window,
                                    is_no_trade_window)

    for friday, off in ((dt.datetime(2024, 1, 5), 5),   # EST
                        (dt.datetime(2024, 7, 5), 4)):  # EDT
        def ny(hh, mm):
            return _to_ny(_epoch(friday.year, friday.month, friday.day,
                                 hh + off, mm))

        for pred, (hh, mm) in ((is_no_new_position_window, (14, 0)),
                               (is_friday_risk_reduction_window, (15, 0)),
                               (is_force_flat_window, (15, 45))):
            before = ny(hh, mm - 1) if mm else ny(hh - 1, 59)
            assert not pred(before), (pred.__name__, "one minute early", off)
            assert pred(ny(hh, mm)), (pred.__name__, "at start", off)
        # daily break near 16:59 NY and the 16:50-17:10 no-trade window
        assert is_broker_daily_break_near(ny(16, 59))
        assert is_no_trade_window(ny(16, 55))
        assert is_no_trade_window(ny(17, 5))
        assert not is_no_trade_window(ny(16, 45))
        assert not is_no_trade_window(ny(17, 15))


def test_calendar_bars_scale_with_timeframe():
    """bars_to_* features are hour-countdowns divided by the timeframe:
    halving timeframe_hours doubles the bar counts (reference
    test_feature_dict_keys_complete_and_bars_scale_with_timeframe)."""
    ts = _epoch(2024, 1, 3, 12, 0)  # Wednesday, mid-session
    f1 = compute_fx_calendar_features(ts, timeframe_hours=1.0)
    f2 = compute_fx_calendar_features(ts, timeframe_hours=0.5)
    assert set(f1) == set(CALENDAR_FEATURE_KEYS)
    for k in CALENDAR_FEATURE_KEYS:
        if k.startswith("bars_"):
            assert f2[k] == pytest.approx(2.0 * f1[k], rel=1e-9), k
            assert f1[k] > 0.0


def test_broker_profile_auto_enables_calendar_obs():
    """broker_profile='oanda_us_fx' flips the calendar-obs flag on without
    setting oanda_fx_calendar_obs explicitly (reference
    test_oanda_fx_broker_profile_auto_enables_calendar_obs), and the info
    dict carries the calendar block + broker metadata; without either
    flag the calendar keys must be absent."""
    import torch

    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(60, seed=3, vol=1e-4)
    base = {"n_envs": 1, "device": "cpu", "window_size": 4,
            "timeframe_hours": 1.0}
    on = build_vec_environment({**base, "broker_profile": "oanda_us_fx"}, md)
    assert on.params.oanda_fx_calendar_obs
    off = build_vec_environment(dict(base), md)
    assert not off.params.oanda_fx_calendar_obs

    import tempfile

    from gymfx_amd import build_environment
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.data.feed import write_csv
    from gymfx_amd.plugins import load_plugin

    d = tempfile.mkdtemp()
    write_csv(md, d + "/cal.csv")

    def gym_env(extra):
        cfg = {**DEFAULT_VALUES, **base, "input_data_file": d + "/cal.csv",
               "quiet_mode": True, **extra}
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

    env = gym_env({"broker_profile": "oanda_us_fx"})
    env.reset()
    _, _, _, _, info = env.step(0)
    assert "hours_to_friday_close" in info
    assert info["broker_profile"] == "oanda_us_fx"
    assert "margin_available_norm" in info
    env.close()
    env2 = gym_env({})
    env2.reset()
    _, _, _, _, info2 = env2.step(0)
    assert "hours_to_friday_close" not in info2
    env2.close()
