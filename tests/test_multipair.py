"""Multi-instrument market tensor (BASELINE config #5): per-env instrument
blocks, episode bounds, per-env pip size, and no cross-instrument leakage."""
import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import concat_markets, synthetic_ohlcv


def _pairs(n=3, rows=300):
    mds = []
    for i in range(n):
        m = synthetic_ohlcv(rows + 40 * i, seed=20 + i, vol=4e-4,
                            instrument=f"P{i}", start_price=1.0 + 0.3 * i)
        m.meta["pip_size"] = 0.0001 if i < 2 else 0.01
        mds.append(m)
    return mds


def _cfg(**kw):
    cfg = {"n_envs": 6, "device": "cpu", "window_size": 8,
           "position_size": 1000.0, "env_start_mode": "zero", "autoreset": True,
           "seed": 0}
    cfg.update(kw)
    return cfg


def test_blocks_assigned_round_robin():
    env = build_vec_environment(_cfg(), _pairs())
    assert env.st.inst_id.tolist() == [0, 1, 2, 0, 1, 2]
    blocks = env.instrument_blocks
    assert [b["instrument"] for b in blocks] == ["P0", "P1", "P2"]
    assert blocks[1]["lo"] == 300 and blocks[1]["end"] == 640
    assert torch.equal(env.st.lo_bar[:3], torch.tensor([0, 300, 640],
                                                       dtype=torch.int32))
    assert env.st.pip_env[2].item() == pytest.approx(0.01)


def test_episodes_stay_inside_blocks():
    env = build_vec_environment(_cfg(), _pairs())
    env.reset(seed=0)
    for i in range(900):
        env.step(torch.zeros(6, dtype=torch.int64))
        t = torch.maximum(env.st.cursor - 1, env.st.lo_bar)
        assert bool((t >= env.st.lo_bar).all())
        assert bool((t < env.st.end_bar).all())


def test_obs_never_reads_across_boundary():
    """Poison instrument P0's data; envs on P1/P2 must be unaffected
    (cross-instrument variant of the reference's leakage poisoning test)."""
    mds = _pairs()
    env1 = build_vec_environment(_cfg(), [m for m in mds])
    env1.reset(seed=0)
    obs1 = []
    for _ in range(100):
        out = env1.step(torch.zeros(6, dtype=torch.int64))
        obs1.append(out["obs"].clone())

    mds2 = _pairs()
    for col in ("OPEN", "HIGH", "LOW", "CLOSE"):
        mds2[0].columns[col] = mds2[0].columns[col] * 7.0
    env2 = build_vec_environment(_cfg(), mds2)
    env2.reset(seed=0)
    for k in range(100):
        out = env2.step(torch.zeros(6, dtype=torch.int64))
        # envs 1,2,4,5 ride P1/P2 -> identical
        for e in (1, 2, 4, 5):
            assert torch.equal(obs1[k][e], out["obs"][e]), (k, e)
        # envs 0,3 ride the poisoned P0 -> must differ somewhere
    assert not torch.equal(obs1[-1][0], out["obs"][0])


def test_fixed_bracket_uses_per_env_pip():
    mds = _pairs()
    env = build_vec_environment(
        _cfg(strategy_plugin="direct_fixed_sltp", sl_pips=10.0, tp_pips=20.0),
        mds)
    env.reset(seed=0)
    env.step(torch.ones(6, dtype=torch.int64))   # request long
    env.step(torch.zeros(6, dtype=torch.int64))  # fill
    # env 2 rides P2 with pip 0.01 -> 100x wider bracket than env 0
    w0 = float(env.st.br_tp[0] - env.st.br_sl[0])
    w2 = float(env.st.br_tp[2] - env.st.br_sl[2])
    assert w2 == pytest.approx(w0 * 100.0 * (1.0), rel=1e-3) or \
        w2 / w0 == pytest.approx(100.0, rel=1e-3)


def test_multipair_trainer_runs():
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer

    env = build_vec_environment(_cfg(n_envs=8, autoreset=True,
                                     env_start_mode="spread"), _pairs())
    env.reset(seed=0)
    pc = PPOConfig(rollout_steps=8, minibatches=2, ppo_epochs=1, seed=0,
                   hidden=16)
    tr = PPOTrainer(env, pc)
    stats = tr.train_update()
    assert np.isfinite(list(stats.values())).all()


def test_concat_markets_block_meta_invariants():
    """instrument_blocks tile [0, total) contiguously with the right names,
    pip sizes, and column concatenation."""
    import numpy as np

    mds = [synthetic_ohlcv(100 + 30 * i, seed=i, instrument=f"I{i}")
           for i in range(4)]
    for i, m in enumerate(mds):
        m.meta["pip_size"] = 0.0001 * (i + 1)
    cat = concat_markets(mds)
    blocks = cat.meta["instrument_blocks"]
    assert len(blocks) == 4
    off = 0
    for i, b in enumerate(blocks):
        assert b["lo"] == off
        assert b["end"] == off + (100 + 30 * i)
        assert b["instrument"] == f"I{i}"
        assert b["pip_size"] == 0.0001 * (i + 1)
        off = b["end"]
    assert off == cat.n_rows == sum(100 + 30 * i for i in range(4))
    # column data concatenated in order
    np.testing.assert_array_equal(
        cat.columns["CLOSE"][blocks[2]["lo"]:blocks[2]["end"]],
        mds[2].columns["CLOSE"])


def test_concat_markets_rejects_mismatched_columns():
    a = synthetic_ohlcv(50, seed=1)
    b = synthetic_ohlcv(50, seed=2, extra_feature_columns=2)
    with pytest.raises(ValueError, match="same columns"):
        concat_markets([a, b])
