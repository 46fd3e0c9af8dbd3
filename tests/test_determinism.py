"""Determinism as hash equality (reference currency:
tests/test_nautilus_bakeoff.py:35-42, tools/nautilus_parallel_smoke.py):
two identical runs produce bit-identical state trajectories."""
import hashlib

import numpy as np
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.data.feed import synthetic_ohlcv


def _run_hash(seed_actions=7, n_envs=8):
    md = synthetic_ohlcv(400, seed=3, vol=4e-4)
    cfg = {
        "n_envs": n_envs,
        "device": "cpu",
        "window_size": 8,
        "initial_cash": 10000.0,
        "position_size": 1000.0,
        "commission": 1e-5,
        "slippage": 1e-5,
        "strategy_plugin": "direct_fixed_sltp",
        "sl_pips": 10.0,
        "tp_pips": 20.0,
        "env_start_mode": "spread",
        "autoreset": True,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=0)
    rng = np.random.default_rng(seed_actions)
    h = hashlib.sha256()
    for _ in range(300):
        a = torch.from_numpy(rng.integers(0, 3, size=n_envs))
        out = env.step(a)
        h.update(out["obs"].numpy().tobytes())
        h.update(out["reward"].numpy().tobytes())
        h.update(env.st.equity.numpy().tobytes())
    return h.hexdigest()


def test_repeat_runs_bit_identical():
    assert _run_hash() == _run_hash()


def test_different_actions_differ():
    assert _run_hash(seed_actions=7) != _run_hash(seed_actions=8)


def test_cross_process_determinism():
    """Spawn-pool hash equality (tools/nautilus_parallel_smoke.py:40-51 idiom)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    with ctx.Pool(2) as pool:
        hashes = pool.map(_worker, [0, 1])
    assert hashes[0] == hashes[1] == _run_hash()


def _worker(_):
    return _run_hash()
