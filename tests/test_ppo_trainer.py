"""PPO trainer on the CPU oracle path: Feistel shuffle properties, fused
minibatch gather coverage, end-to-end determinism, and data-parallel
equivalence over gloo (world_size=2) — the CPU stand-in for the RCCL path
(reference has no trainer; BASELINE.json north-star configs #2/#3)."""
import os

import numpy as np
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.ops import api


# ---------------------------------------------------------------------------
# Feistel permutation
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n", [7, 64, 100, 1000, 4096])
def test_feistel_is_a_bijection(n):
    perm = api.feistel_perm(n, api.feistel_key(123, 5, 0))
    assert perm.shape == (n,)
    assert torch.equal(torch.sort(perm).values, torch.arange(n))


def test_feistel_keys_differ():
    a = api.feistel_perm(1024, api.feistel_key(1, 0, 0))
    b = api.feistel_perm(1024, api.feistel_key(1, 0, 1))
    c = api.feistel_perm(1024, api.feistel_key(1, 128, 0))
    assert not torch.equal(a, b)
    assert not torch.equal(a, c)
    # deterministic
    assert torch.equal(a, api.feistel_perm(1024, api.feistel_key(1, 0, 0)))


def test_mb_gather_epoch_covers_all_rows():
    n, mbs = 96, 4
    M = n // mbs
    D = 8
    obs = torch.arange(n * D, dtype=torch.float32).view(n, D).to(torch.bfloat16)
    act = torch.arange(n, dtype=torch.int64)
    f = torch.arange(n, dtype=torch.float32)
    obs_mb = torch.empty(M, D, dtype=torch.bfloat16)
    act_mb = torch.empty(M, dtype=torch.int64)
    f_mb = torch.empty(M, dtype=torch.float32)
    step_base = torch.tensor(7, dtype=torch.int64)
    seen = []
    for mb in range(mbs):
        mb_ctr = torch.tensor(mb, dtype=torch.int64)  # epoch 0
        api.mb_gather(obs, act, f, f, f, obs_mb, act_mb, f_mb, f_mb.clone(),
                      f_mb.clone(), seed=9, minibatches=mbs,
                      step_base=step_base, mb_ctr=mb_ctr)
        seen.append(act_mb.clone())
        # gathered rows are consistent across tensors
        assert torch.equal(obs_mb[:, 0].to(torch.float32),
                           (act_mb * D).to(torch.bfloat16).to(torch.float32))
    allseen = torch.cat(seen)
    assert torch.equal(torch.sort(allseen).values, torch.arange(n))


# ---------------------------------------------------------------------------
# trainer end-to-end (CPU oracle path)
# ---------------------------------------------------------------------------

def _make_trainer(rank=0, world_size=1, pg=None, seed=11):
    md = synthetic_ohlcv(600, seed=5, vol=4e-4)
    cfg = {
        "n_envs": 16,
        "device": "cpu",
        "window_size": 8,
        "env_start_mode": "spread",
        "autoreset": True,
        "position_size": 1000.0,
        "seed": seed + rank,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=seed + rank)
    pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=seed,
                   hidden=32)
    return PPOTrainer(env, pc, rank=rank, world_size=world_size,
                      process_group=pg)


def test_trainer_runs_and_is_deterministic():
    t1 = _make_trainer()
    t2 = _make_trainer()
    for _ in range(2):
        s1 = t1.train_update()
        s2 = t2.train_update()
    assert torch.equal(t1.model.params, t2.model.params)
    assert s1 == s2
    assert np.isfinite(list(s1.values())).all()
    assert t1.global_step == 32


def test_trainer_stats_sane():
    t = _make_trainer()
    s = t.train_update()
    assert set(s) == {"pi_loss", "v_loss", "entropy", "approx_kl", "clipfrac"}
    assert 0.0 <= s["clipfrac"] <= 1.0
    assert 0.0 < s["entropy"] <= np.log(3) + 1e-5


# ---------------------------------------------------------------------------
# data-parallel gloo world_size=2 (CPU stand-in for RCCL/xGMI)
# ---------------------------------------------------------------------------

def _ddp_worker(rank, world_size, port, out_q):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        t = _make_trainer(rank=rank, world_size=world_size,
                          pg=dist.group.WORLD)
        for _ in range(2):
            t.train_update()
        out_q.put((rank, t.model.params.numpy().tobytes()))
    finally:
        dist.destroy_process_group()


def test_ddp_gloo_ws2_ranks_stay_in_sync():
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29731
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, params = q.get(timeout=300)
        results[rank] = params
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # replicas start identical and all-reduce every minibatch -> identical
    assert results[0] == results[1]
    # and the data-parallel result differs from a single-rank run (it saw
    # different rollouts on rank 1)
    solo = _make_trainer()
    for _ in range(2):
        solo.train_update()
    assert results[0] != solo.model.params.numpy().tobytes()


def test_ppo_learns_to_go_long_on_uptrend():
    """End-to-end training efficacy: on a strongly trending market the
    policy must shift toward long and the mean step reward must rise
    (behavioral upgrade of the reference's buy_hold>0 smoke invariant,
    tools/smoke_test.py:108-155)."""
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(3000, seed=8, vol=1e-4, drift=4e-4)
    cfg = {"n_envs": 64, "device": "cpu", "window_size": 8,
           "env_start_mode": "spread", "autoreset": True,
           "position_size": 1000.0, "seed": 5}
    env = build_vec_environment(cfg, md)
    env.reset(seed=5)
    pc = PPOConfig(rollout_steps=32, minibatches=4, ppo_epochs=4, seed=5,
                   hidden=32, lr=3e-3, ent_coef=0.003)
    tr = PPOTrainer(env, pc)
    tr.train_update()
    first = float(tr.rew_buf.mean())
    for _ in range(19):
        tr.train_update()
    last = float(tr.rew_buf.mean())
    long_frac = float((tr.act_buf == 1).float().mean())
    assert last > max(first * 5, 2e-5), (first, last)
    assert long_frac > 0.5, long_frac


def test_ppo_config_bool_string_coercion():
    """String 'false' from a raw config dict must not coerce to True
    (bool('false') is True; from_config routes through convert_type)."""
    from gymfx_amd.algo.ppo import PPOConfig

    assert PPOConfig.from_config({"fuse_sample": "false"}).fuse_sample is False
    assert PPOConfig.from_config({"use_graphs": "false"}).use_graphs is False
    assert PPOConfig.from_config({"shuffle_rows": "true"}).shuffle_rows is True


def test_feistel_bijection_property():
    """Hypothesis property: the cycle-walked Feistel permutation is a
    bijection on [0, n) for ARBITRARY n (odd, prime, tiny, non-power-of-2)
    and key — the correctness backbone of the minibatch shuffle."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=50, deadline=None)
    @given(st.integers(min_value=2, max_value=5000),
           st.integers(min_value=0, max_value=2**31),
           st.integers(min_value=0, max_value=64),
           st.integers(min_value=0, max_value=7))
    def check(n, seed, step, epoch):
        perm = api.feistel_perm(n, api.feistel_key(seed, step, epoch))
        assert torch.equal(torch.sort(perm).values, torch.arange(n))

    check()


def test_ddp_gloo_ws3_odd_world_size():
    """Odd world size: the 1/world_size gradient averaging and the
    rank-keyed rollout seeds must keep 3 replicas bit-identical too
    (guards the scale run at any N)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ddp_worker, args=(r, 3, 29737, q))
             for r in range(3)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(3):
        rank, params = q.get(timeout=300)
        results[rank] = params
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert results[0] == results[1] == results[2]


# ---------------------------------------------------------------------------
# failure path: a dead peer rank must surface as a timely error, not a hang
# (failure-detection stance of SURVEY.md §5.3)
# ---------------------------------------------------------------------------

def _ddp_dying_worker(rank, world_size, port, out_q):
    import time as _time

    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    if rank == 1:
        # die mid-job WITHOUT destroying the pg — simulates a crashed rank
        os._exit(17)
    t = _make_trainer(rank=rank, world_size=world_size, pg=dist.group.WORLD)
    t._reducer.timeout_s = 5.0
    t0 = _time.perf_counter()
    try:
        t.train_update()
        out_q.put((rank, "no-error", _time.perf_counter() - t0))
    except RuntimeError as exc:
        out_q.put((rank, f"raised: {exc}", _time.perf_counter() - t0))


def test_ddp_dead_rank_surfaces_watchdog_error():
    """Kill rank 1 before its first collective: rank 0's GradAllReducer
    watchdog must raise within its timeout instead of hanging forever."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ddp_dying_worker, args=(r, 2, 29739, q))
             for r in range(2)]
    for p in procs:
        p.start()
    rank, outcome, elapsed = q.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        p.kill()  # belt-and-braces; join above should have reaped both
    assert rank == 0
    assert outcome.startswith("raised:"), outcome
    # raised promptly (5 s watchdog + slack), not after some giant default
    assert elapsed < 60.0, elapsed


# ---------------------------------------------------------------------------
# optimization mode with policy-driven trials
# ---------------------------------------------------------------------------

def test_optimize_policy_driver_uses_checkpoint(tmp_path):
    """driver_mode=policy scores trials with a trained policy's greedy
    actions instead of scripted random/buy-hold rollouts."""
    from gymfx_amd.algo.optimize import optimize_from_config
    from gymfx_amd.utils.checkpoint import save_checkpoint

    t = _make_trainer()
    t.train_update()
    ckpt = str(tmp_path / "p.pt")
    save_checkpoint(t, ckpt)
    cfg = {
        "n_envs": 16, "device": "cpu", "window_size": 8,
        "data_feed_plugin": "synthetic_data_feed", "synthetic_rows": 400,
        "strategy_plugin": "direct_atr_sltp", "seed": 5,
        "optimization_trials": 2, "optimization_steps": 24,
        "driver_mode": "policy", "checkpoint_file": ckpt,
        "position_size": 1000.0, "quiet_mode": True,
    }
    out = optimize_from_config(cfg)
    assert out["trials"] == 2 and "best_params" in out


def test_optimize_train_driver_trains_per_trial():
    """driver_mode=train runs a short PPO fit per trial and scores the
    trained policy — optimization can now tune reward/strategy hparams the
    PPO path cares about."""
    from gymfx_amd.algo.optimize import optimize_from_config

    cfg = {
        "n_envs": 16, "device": "cpu", "window_size": 8,
        "data_feed_plugin": "synthetic_data_feed", "synthetic_rows": 500,
        "strategy_plugin": "direct_atr_sltp", "seed": 6,
        "optimization_trials": 2, "optimization_steps": 16,
        "driver_mode": "train", "optimization_train_updates": 1,
        "hidden_size": 16, "position_size": 1000.0, "quiet_mode": True,
    }
    out = optimize_from_config(cfg)
    assert out["trials"] == 2 and "best_params" in out


def test_mode_training_under_torchrun_ws2_synchronizes(tmp_path):
    """python -m gymfx_amd.main --mode training under torchrun ws=2 must
    join the data-parallel group (gloo here, RCCL on GPUs): both ranks'
    models see all-reduced gradients, only rank 0 writes the checkpoint
    and results, and the checkpoint stays loadable."""
    import json
    import socket
    import subprocess
    import sys
    from pathlib import Path

    import torch

    repo = Path(__file__).resolve().parents[1]
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.data.feed import synthetic_ohlcv, write_csv

    write_csv(synthetic_ohlcv(600, seed=5, vol=2e-4), str(tmp_path / "px.csv"))
    cfg = {**DEFAULT_VALUES, "mode": "training",
           "input_data_file": str(tmp_path / "px.csv"),
           "n_envs": 8, "window_size": 6, "device": "cpu",
           "train_updates": 2, "rollout_steps": 8, "minibatches": 2,
           "ppo_epochs": 1, "hidden_size": 16, "seed": 3,
           "quiet_mode": True,
           "checkpoint_file": str(tmp_path / "ck.pt"),
           "results_file": str(tmp_path / "results.json"),
           "save_config": str(tmp_path / "config_out.json"),
           "save_log": None}
    cfg_path = tmp_path / "train.json"
    cfg_path.write_text(json.dumps(cfg))
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "-m", "gymfx_amd.main",
         "--load_config", str(cfg_path)],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert r.returncode == 0, r.stderr[-2500:]
    res = json.loads((tmp_path / "results.json").read_text())
    assert res["world_size"] == 2 and res["rank"] == 0
    sd = torch.load(tmp_path / "ck.pt", map_location="cpu",
                    weights_only=True)
    assert sd["policy"] == "mlp"


def test_trainer_rejects_continuous_action_mode(tmp_path):
    """PPO's head is 3-way discrete; a continuous-mode env must be
    refused at construction with a clear message, not a kernel dtype
    error mid-rollout."""
    import pytest as _pytest
    import torch

    from gymfx_amd import build_vec_environment
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(200, seed=2, vol=1e-4)
    env = build_vec_environment(
        {"n_envs": 4, "device": "cpu", "window_size": 4,
         "action_space_mode": "continuous"}, md)
    with _pytest.raises(ValueError, match="discrete"):
        PPOTrainer(env, PPOConfig(rollout_steps=4, minibatches=2,
                                  ppo_epochs=1, hidden=16))
