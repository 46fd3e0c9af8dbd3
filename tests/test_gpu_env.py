"""GPU: the fused HIP env_step/build_obs kernels vs the torch oracle
(envs/reference_step.py), elementwise, across strategy/reward/preprocessor
configurations and many steps."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import synthetic_ohlcv

pytestmark = pytest.mark.gpu

N = 64
STEPS = 220


def _market(pairs=1):
    if pairs == 1:
        return synthetic_ohlcv(2000, seed=5, vol=4e-4, extra_feature_columns=3)
    from gymfx_amd.data.feed import concat_markets
    return concat_markets([
        synthetic_ohlcv(700 + 100 * i, seed=5 + i, vol=4e-4,
                        extra_feature_columns=3, instrument=f"PAIR_{i}")
        for i in range(pairs)
    ])


BASE = {
    "n_envs": N,
    "window_size": 16,
    "initial_cash": 10000.0,
    "position_size": 1000.0,
    "commission": 2e-5,
    "slippage": 1e-5,
    "env_start_mode": "spread",
}

CONFIGS = {
    "default": {},
    "fixed_sltp": {"strategy_plugin": "direct_fixed_sltp", "sl_pips": 10.0, "tp_pips": 20.0},
    "atr": {
        "strategy_plugin": "direct_atr_sltp",
        "rel_volume": 0.1,
        "leverage": 10.0,
        "atr_period": 7,
    },
    "sharpe": {"reward_plugin": "sharpe_reward", "window": 16},
    "dd": {"reward_plugin": "dd_penalized_reward", "penalty_lambda": 0.5},
    "feature_window": {
        "preprocessor_plugin": "feature_window_preprocessor",
        "feature_columns": ["OPEN", "CLOSE", "FEAT_0", "FEAT_1"],
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 64,
    },
    "autoreset": {"autoreset": True, "strategy_plugin": "direct_fixed_sltp",
                  "sl_pips": 3.0, "tp_pips": 3.0},
    "continuous": {"action_space_mode": "continuous"},
    "multipair": {"_pairs": 3, "autoreset": True,
                  "strategy_plugin": "direct_fixed_sltp",
                  "sl_pips": 6.0, "tp_pips": 9.0},
    "preflight": {"enforce_margin_preflight": True, "leverage": 1.0,
                  "position_size": 50000.0, "initial_cash": 10000.0},
    "financing": {"financing_enabled": True,
                  "rollover_rate_data": [
                      {"LOCATION": "EA19", "TIME": "2024-01", "Value": 5.0},
                      {"LOCATION": "USA", "TIME": "2024-01", "Value": 4.0}]},
    # execution-realism tier: one GPU parity config per
    # policy against the same torch oracle
    "collision_ohlc": {"strategy_plugin": "direct_fixed_sltp",
                       "sl_pips": 6.0, "tp_pips": 6.0, "autoreset": True,
                       "intrabar_collision_policy": "ohlc"},
    "collision_adaptive": {"strategy_plugin": "direct_fixed_sltp",
                           "sl_pips": 6.0, "tp_pips": 6.0, "autoreset": True,
                           "intrabar_collision_policy": "adaptive"},
    "limit_cross": {"strategy_plugin": "direct_fixed_sltp",
                    "sl_pips": 8.0, "tp_pips": 4.0, "autoreset": True,
                    "limit_fill_policy": "cross"},
    "limit_conservative": {"strategy_plugin": "direct_fixed_sltp",
                           "sl_pips": 8.0, "tp_pips": 4.0, "autoreset": True,
                           "limit_fill_policy": "conservative"},
    "latency": {"latency_ms": 120_000.0, "strategy_plugin": "direct_fixed_sltp",
                "sl_pips": 10.0, "tp_pips": 10.0},
    "margin_standard": {"margin_model": "standard", "margin_init_rate": 0.04},
}


@pytest.mark.parametrize("name", sorted(CONFIGS))
def test_kernel_matches_torch_oracle(name):
    cfg = {**BASE, **CONFIGS[name]}
    md = _market(cfg.pop("_pairs", 1))
    env_g = build_vec_environment({**cfg, "device": "cuda"}, md, use_native=True)
    env_c = build_vec_environment({**cfg, "device": "cuda"}, md, use_native=False)
    env_g.reset(seed=0)
    env_c.reset(seed=0)
    rng = np.random.default_rng(9)
    for k in range(STEPS):
        if cfg.get("action_space_mode") == "continuous":
            a = torch.from_numpy(rng.uniform(-1, 1, N).astype(np.float32)).cuda()
        else:
            a = torch.from_numpy(rng.integers(0, 3, N)).cuda()
        out_g = env_g.step(a)
        out_c = env_c.step(a)
        torch.cuda.synchronize()
        for fld in ("equity", "cash", "pos", "avg_entry", "commission_paid"):
            g = getattr(env_g.st, fld).cpu().numpy()
            c = getattr(env_c.st, fld).cpu().numpy()
            np.testing.assert_allclose(g, c, rtol=1e-9, atol=1e-9,
                                       err_msg=f"{name} step {k} field {fld}")
        np.testing.assert_array_equal(
            env_g.st.cursor.cpu().numpy(), env_c.st.cursor.cpu().numpy()
        )
        np.testing.assert_array_equal(
            env_g.st.terminated.cpu().numpy(), env_c.st.terminated.cpu().numpy()
        )
        # sharpe's ring mean/var differ by f32 summation ORDER between the
        # kernel (sequential) and torch (pairwise) — tolerance, not equality.
        np.testing.assert_allclose(
            out_g["reward"].cpu().numpy(),
            out_c["reward"].cpu().float().numpy(),
            rtol=5e-4, atol=1e-6, err_msg=f"{name} step {k} reward",
        )
        np.testing.assert_allclose(
            out_g["obs"].cpu().numpy(), out_c["obs"].cpu().numpy(),
            rtol=3e-5, atol=3e-5, err_msg=f"{name} step {k} obs",
        )
        np.testing.assert_array_equal(
            env_g.st.exec_diag.cpu().numpy(), env_c.st.exec_diag.cpu().numpy(),
            err_msg=f"{name} step {k} exec_diag",
        )


def test_gpu_determinism_bitwise():
    """Same run twice on GPU -> bit-identical equity trajectory."""
    md = _market()
    cfg = {**BASE, "strategy_plugin": "direct_fixed_sltp", "autoreset": True,
           "device": "cuda"}

    def run():
        env = build_vec_environment(cfg, md, use_native=True)
        env.reset(seed=0)
        rng = np.random.default_rng(3)
        tail = []
        for _ in range(150):
            a = torch.from_numpy(rng.integers(0, 3, N)).cuda()
            env.step(a)
        return env.st.equity.cpu().numpy().copy(), env.st.trade_count.cpu().numpy().copy()

    e1, t1 = run()
    e2, t2 = run()
    np.testing.assert_array_equal(e1, e2)
    np.testing.assert_array_equal(t1, t2)


def test_native_required_on_gpu():
    """GPU default path must use the HIP engine (no silent eager fallback)."""
    md = _market()
    env = build_vec_environment({**BASE, "device": "cuda"}, md)
    assert env._native is not None


def test_replay_reconciles_hip_engine_vs_scalar_ledger():
    """The fused HIP engine's ledger vs the independent pure-Python scalar
    oracle (replay.py; bakeoff.py:228-303 idiom) — on device."""
    import numpy as np

    from gymfx_amd.replay import ReplayAdapter

    md = synthetic_ohlcv(400, seed=9, vol=5e-4)
    rng = np.random.default_rng(3)
    actions = rng.integers(0, 3, size=250).tolist()
    cfg = {"window_size": 8, "initial_cash": 10000.0, "position_size": 1000.0,
           "commission": 2e-5, "slippage": 5e-6, "device": "cuda", "seed": 0,
           "strategy_plugin": "direct_fixed_sltp", "sl_pips": 8.0,
           "tp_pips": 16.0}
    res = ReplayAdapter().run(cfg, md, actions)
    assert res["reconciled"], res["reconciliation"]
    kinds = {e["type"] for e in res["events"]}
    assert "order_filled" in kinds


def test_update_speed_regression_guard():
    """Catastrophic-regression guard: one full PPO update (4096 envs,
    T=128) must stay well under 3x the measured ~23 ms."""
    import time

    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer

    md = synthetic_ohlcv(65536, seed=5, vol=4e-4, extra_feature_columns=3)
    cfg = {**BASE, "n_envs": 4096, "device": "cuda", "autoreset": True,
           "preprocessor_plugin": "feature_window_preprocessor",
           "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE",
                               "FEAT_0", "FEAT_1", "FEAT_2"]}
    env = build_vec_environment(cfg, md)
    env.reset(seed=0)
    tr = PPOTrainer(env, PPOConfig(seed=0))
    for _ in range(3):
        tr.train_update(with_stats=False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        tr.train_update(with_stats=False)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 5 * 1000
    assert ms < 80.0, f"PPO update took {ms:.1f} ms (expected ~23 ms)"


def test_fused_obs_equals_two_launch():
    """env_step_obs_kernel (one env per wave: lane-0 step + wave obs build)
    must be bitwise identical to the separate env_step + build_obs launches."""
    md = synthetic_ohlcv(3000, seed=11, vol=4e-4)
    def make():
        cfg = {"n_envs": 193, "device": "cuda", "window_size": 16,
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 3,
               "strategy_plugin": "direct_atr_sltp"}
        env = build_vec_environment(cfg, md)
        env.reset(seed=3)
        return env
    e1, e2 = make(), make()
    g = torch.Generator().manual_seed(44)
    for _ in range(60):
        acts = torch.randint(0, 3, (193,), generator=g).cuda()
        o1 = e1.step(acts, fuse_obs=True)
        o2 = e2.step(acts, fuse_obs=False)
        assert torch.equal(o1["obs"], o2["obs"])
        assert torch.equal(o1["reward"], o2["reward"])
        assert torch.equal(o1["terminated"], o2["terminated"])
    assert torch.equal(e1.st.equity, e2.st.equity)
    assert torch.equal(e1.st.cursor, e2.st.cursor)
    assert torch.equal(e1.st.trade_count, e2.st.trade_count)
