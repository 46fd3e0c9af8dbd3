"""Preprocessor cross-validation: the numpy plugin (reference semantics)
vs the vectorized torch observation builder, plus the reference's signature
no-future-leakage poisoning test
(tests/test_feature_window_preprocessor.py:113-139 idiom)."""
import numpy as np
import pytest
import torch

from gymfx_amd.data.feed import MarketData, synthetic_ohlcv
from gymfx_amd.envs.market import build_market_tensors
from gymfx_amd.envs.params import EnvParams
from gymfx_amd.envs.reference_step import build_obs_torch
from gymfx_amd.envs.state import alloc_state
from gymfx_amd.plugins.preprocessors import DefaultPreprocessor, FeatureWindowPreprocessor

FEATURES = ["CLOSE", "FEAT_0", "FEAT_1", "FEAT_2"]


def _vec_obs_at_steps(md, config, steps, equity=10000.0, pos_sign=0.0):
    """Build the torch obs with each env's cursor pinned at a given step."""
    config = dict(config)
    config["n_envs"] = len(steps)
    params = EnvParams.from_config(config, timeframe_hours=md.timeframe_hours())
    device = torch.device("cpu")
    mt = build_market_tensors(md, params, device)
    st = alloc_state(params, device)
    st.cursor = torch.tensor(steps, dtype=torch.int32)
    st.equity = torch.full((len(steps),), equity, dtype=torch.float64)
    st.pos = torch.full((len(steps),), pos_sign, dtype=torch.float64)
    return build_obs_torch(st, mt, params), params


@pytest.mark.parametrize("scaling", ["none", "rolling_zscore", "expanding_zscore"])
def test_feature_window_torch_matches_numpy(sample_market, scaling):
    md = sample_market
    config = {
        "window_size": 16,
        "feature_columns": FEATURES,
        "feature_binary_columns": ["FEAT_2"],
        "feature_scaling": scaling,
        "feature_scaling_window": 64,
        "feature_clip": 10.0,
        "preprocessor_plugin": "feature_window_preprocessor",
        "include_price_window": True,
        "initial_cash": 10000.0,
        "position_size": 1.0,
    }
    plugin = FeatureWindowPreprocessor(config)
    steps = [1, 2, 5, 15, 16, 17, 100, 300, 599]
    obs_vec, params = _vec_obs_at_steps(md, config, steps)
    sl = params.obs_slices()
    W, F = 16, len(FEATURES)
    for i, step in enumerate(steps):
        ref = plugin.make_observation(
            data=md,
            step=step,
            bridge_state={
                "position": 0,
                "equity": 10000.0,
                "initial_cash": 10000.0,
                "price": float(md.close[min(step - 1, md.n_rows - 1)]),
                "bar_index": step,
                "total_bars": md.n_rows,
            },
            config=config,
        )
        got_feat = obs_vec[i, sl["features"]].numpy().reshape(W, F)
        np.testing.assert_allclose(got_feat, ref["features"], rtol=2e-4, atol=2e-4)
        got_prices = obs_vec[i, sl["prices"]].numpy()
        np.testing.assert_allclose(got_prices, ref["prices"], rtol=1e-6)
        got_returns = obs_vec[i, sl["returns"]].numpy()
        np.testing.assert_allclose(got_returns, ref["returns"], rtol=1e-4, atol=1e-6)
        got_agent = obs_vec[i, sl["agent_state"]].numpy()
        np.testing.assert_allclose(got_agent[1], ref["equity_norm"][0], atol=1e-6)
        np.testing.assert_allclose(got_agent[3], ref["steps_remaining_norm"][0], atol=1e-6)


def test_default_preprocessor_matches(sample_market):
    md = sample_market
    config = {"window_size": 32, "initial_cash": 10000.0, "position_size": 1.0}
    plugin = DefaultPreprocessor(config)
    steps = [1, 4, 31, 32, 33, 250]
    obs_vec, params = _vec_obs_at_steps(md, config, steps)
    sl = params.obs_slices()
    for i, step in enumerate(steps):
        ref = plugin.make_observation(
            data=md,
            step=step,
            bridge_state={
                "position": 0,
                "equity": 10000.0,
                "initial_cash": 10000.0,
                "price": float(md.close[min(step - 1, md.n_rows - 1)]),
                "bar_index": step,
                "total_bars": md.n_rows,
            },
            config=config,
        )
        np.testing.assert_allclose(obs_vec[i, sl["prices"]].numpy(), ref["prices"], rtol=1e-6)
        np.testing.assert_allclose(
            obs_vec[i, sl["returns"]].numpy(), ref["returns"], rtol=1e-4, atol=1e-6
        )


def test_no_future_leakage_poisoning(sample_market):
    """Poison all rows >= step with 1e6; observation must be bit-identical."""
    md = sample_market
    step = 300
    config = {
        "window_size": 16,
        "feature_columns": FEATURES,
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 64,
        "preprocessor_plugin": "feature_window_preprocessor",
        "include_price_window": True,
    }
    obs_clean, params = _vec_obs_at_steps(md, config, [step])

    poisoned_cols = {k: v.copy() for k, v in md.columns.items()}
    for c in set(FEATURES) | {"OPEN", "HIGH", "LOW", "CLOSE"}:
        poisoned_cols[c][step:] = 1e6
    md_poisoned = MarketData(
        columns=poisoned_cols, timestamps=md.timestamps, timeframe=md.timeframe
    )
    obs_poisoned, _ = _vec_obs_at_steps(md_poisoned, config, [step])
    np.testing.assert_array_equal(obs_clean.numpy(), obs_poisoned.numpy())


def test_binary_passthrough_and_clip(sample_market):
    md = sample_market
    cols = {k: v.copy() for k, v in md.columns.items()}
    cols["BIN"] = (np.arange(md.n_rows) % 2).astype(np.float64)
    cols["HUGE"] = np.where(np.arange(md.n_rows) == 250, 1e9, 1.0)
    md2 = MarketData(columns=cols, timestamps=md.timestamps, timeframe=md.timeframe)
    config = {
        "window_size": 8,
        "feature_columns": ["BIN", "HUGE"],
        "feature_binary_columns": ["BIN"],
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 32,
        "feature_clip": 5.0,
        "preprocessor_plugin": "feature_window_preprocessor",
        "include_price_window": False,
        "include_agent_state": False,
    }
    obs, params = _vec_obs_at_steps(md2, config, [251, 260])
    feat = obs.numpy().reshape(2, 8, 2)
    # binary column passes through unscaled
    assert set(np.unique(feat[:, :, 0])) <= {0.0, 1.0}
    # huge spike is clipped
    assert feat.max() <= 5.0 + 1e-6
    assert feat.min() >= -5.0 - 1e-6


def test_warmup_zero_history(sample_market):
    """Fewer than 2 history rows -> neutral zeros (not raw leaked levels)."""
    md = sample_market
    config = {
        "window_size": 8,
        "feature_columns": ["CLOSE"],
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 64,
        "preprocessor_plugin": "feature_window_preprocessor",
        "include_price_window": False,
        "include_agent_state": False,
    }
    obs, _ = _vec_obs_at_steps(md, config, [1])
    assert np.all(obs.numpy() == 0.0)


def test_load_csv_case_insensitive_and_nan_fill(tmp_path):
    """Reference data_handler semantics (app/data_handler.py:38-64):
    'date_time' matches case-insensitively (duplicates dropped), the
    price column matches case-insensitively, and NaN cells in numeric
    columns fill with 0."""
    from gymfx_amd.data.feed import load_csv

    p = tmp_path / "mixed.csv"
    p.write_text(
        "date_time,close,Volume,NOTE\n"
        "2024-01-02 00:00:00,1.10,100,alpha\n"
        "2024-01-02 00:01:00,,200,beta\n"
        "2024-01-02 00:02:00,1.12,,gamma\n")
    md = load_csv(str(p), date_column="DATE_TIME", price_column="CLOSE")
    assert md.timestamps is not None and len(md.timestamps) == 3
    assert "close" in md.columns
    assert md.columns["close"][1] == 0.0       # NaN -> 0 (reference fill)
    assert md.columns["Volume"][2] == 0.0
    assert "NOTE" not in md.columns            # fully non-numeric: auxiliary


def test_feature_aware_obs_space_excludes_disabled_raw_prices():
    """With feature_columns configured, the observation space drops the
    raw prices/returns blocks unless include_price_window is explicitly
    on (reference test_feature_aware_observation_space_excludes_disabled_
    raw_prices / test_legacy_observation_space_keeps_prices_returns)."""
    from gymfx_amd.envs.gym_env import build_base_observation_space

    feat = build_base_observation_space(
        {"feature_columns": FEATURES}, window_size=8)
    assert "features" in feat.spaces
    assert "prices" not in feat.spaces and "returns" not in feat.spaces
    both = build_base_observation_space(
        {"feature_columns": FEATURES, "include_price_window": True},
        window_size=8)
    assert {"features", "prices", "returns"} <= set(both.spaces)
    legacy = build_base_observation_space({}, window_size=8)
    assert "prices" in legacy.spaces and "features" not in legacy.spaces


def test_missing_and_empty_feature_columns_raise(sample_market):
    """Configured-but-absent feature columns and an empty feature list
    both fail loudly (reference test_missing_columns_raises /
    test_empty_feature_list_raises)."""
    pre = FeatureWindowPreprocessor({})
    with pytest.raises(ValueError, match="missing from data"):
        pre.make_observation(
            data=sample_market, step=4, bridge_state={},
            config={"feature_columns": ["CLOSE", "NO_SUCH_COL"],
                    "window_size": 4})
    with pytest.raises(ValueError, match="non-empty"):
        pre.make_observation(
            data=sample_market, step=4, bridge_state={},
            config={"feature_columns": [], "window_size": 4})
