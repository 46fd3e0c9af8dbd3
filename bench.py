#!/usr/bin/env python3
"""Flagship benchmark: PPO on EURUSD-1m, 4096 vectorized envs per GPU.

BASELINE.json metric: env-steps/sec (whole node), PPO EURUSD-1m, 4096
vec-envs, feature_window_preprocessor, MLP(256,256) bf16 — measured on
synthetic OHLCV ticks (no network for datasets) with random-init weights.

One "step" = one full PPO update: a T=128-step on-device rollout across all
envs (fused HIP env kernels + MFMA policy forward + sampler) followed by
GAE, advantage normalization and 4 epochs x 8 minibatches of clipped-PPO
updates (MFMA fwd/bwd + fused Adam), gradients all-reduced over RCCL for
world_size > 1 (weak scaling: per-GPU work fixed).

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.config import DEFAULT_VALUES

N_ENVS = 4096
ROLLOUT_T = 128
SYN_ROWS = 262_144  # ~6 months of 1-minute bars
FEATURES = ["OPEN", "HIGH", "LOW", "CLOSE", "VOLUME", "FEAT_0", "FEAT_1", "FEAT_2"]


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--n-envs", type=int, default=N_ENVS)
    ap.add_argument("--rollout", type=int, default=ROLLOUT_T)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--policy", type=str, default="mlp", choices=["mlp", "lstm"],
                    help="actor-critic architecture (BASELINE config #2 / #4)")
    ap.add_argument("--reward", type=str, default=None,
                    help="reward plugin (default: BASELINE config for the run shape)")
    ap.add_argument("--strategy", type=str, default=None,
                    help="strategy plugin (default: BASELINE config for the run shape)")
    ap.add_argument("--pairs", type=int, default=1,
                    help="instruments in the market tensor (BASELINE config #5)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))

    have_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    else:
        device = torch.device(f"cuda:{local_rank}") if have_gpu else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(device)

    pg = None
    # init the process group whenever a launcher provided rendezvous env
    # (even at world_size 1): the driver's torchrun N=1 run then exercises
    # RCCL communicator init on hardware, not just the ws>1 path.
    if world_size > 1 or ("MASTER_ADDR" in os.environ and "MASTER_PORT" in os.environ):
        import torch.distributed as dist

        backend = "nccl" if device.type == "cuda" else "gloo"
        dist.init_process_group(backend=backend)
        pg = dist.group.WORLD

    cfg = {
        **DEFAULT_VALUES,
        "data_feed_plugin": "synthetic_data_feed",
        "synthetic_rows": SYN_ROWS,
        "synthetic_seed": 1234,
        "synthetic_pairs": args.pairs,
        "synthetic_extra_features": 3,
        "synthetic_bar_minutes": 1,
        "instrument": "EUR_USD",
        "timeframe": "M1",
        "preprocessor_plugin": "feature_window_preprocessor",
        # BASELINE.json named configs: #4 (LSTM) dd_penalized +
        # direct_atr_sltp.  The MLP scale runs use pnl_reward at EVERY N so
        # the driver's weak-scaling efficiency compares identical work
        # (config #3's sharpe_reward variant is measured separately:
        # profiles/bench_mlp_sharpe.json, ~8% env-kernel cost).
        # Overridable via --reward/--strategy.
        "reward_plugin": args.reward or (
            "dd_penalized_reward" if args.policy == "lstm" else "pnl_reward"),
        "strategy_plugin": args.strategy or (
            "direct_atr_sltp" if args.policy == "lstm" else "default_strategy"),
        "feature_columns": FEATURES,
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 256,
        "window_size": 32,
        "n_envs": args.n_envs,
        "device": str(device),
        "autoreset": True,
        "env_start_mode": "spread",
        "position_size": 1000.0,
        "commission": 2e-5,
        "slippage": 5e-6,
        "seed": 1000 + rank,
    }
    env = build_vec_environment(cfg)
    env.reset(seed=1000 + rank)
    if device.type == "cuda":
        assert env._native is not None, "HIP engine must be active on GPU"

    ppo = PPOConfig(
        rollout_steps=args.rollout,
        ppo_epochs=4,
        minibatches=8,
        seed=1000,
        policy=args.policy,
    )
    trainer = PPOTrainer(env, ppo, rank=rank, world_size=world_size, process_group=pg)

    def barrier_sync():
        if world_size > 1:
            import torch.distributed as dist

            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        trainer.train_update(with_stats=False)
    if trainer.use_graphs and not trainer._graphs_ready:
        trainer._capture_graphs()  # warmup=0 safety: keep capture untimed
    # per-phase attribution: CUDA events bracket the
    # rollout and update phases of every timed step — no host syncs inside
    # the timed region; elapsed_time is read after the closing barrier.
    use_ev = device.type == "cuda"
    if use_ev:
        evs = [tuple(torch.cuda.Event(enable_timing=True) for _ in range(3))
               for _ in range(args.steps)]
    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        if use_ev:
            e0, e1, e2 = evs[i]
            e0.record()
            trainer.collect_rollout()
            e1.record()
            trainer.update(with_stats=False)
            e2.record()
        else:
            trainer.collect_rollout()
            trainer.update(with_stats=False)
    barrier_sync()
    elapsed = time.perf_counter() - t0
    rollout_ms = update_ms = None
    if use_ev:
        rollout_ms = sum(e0.elapsed_time(e1) for e0, e1, _ in evs) / args.steps
        update_ms = sum(e1.elapsed_time(e2) for _, e1, e2 in evs) / args.steps

    # MAX over ranks (driver contract)
    if world_size > 1:
        import torch.distributed as dist

        e = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if device.type == "cuda" else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    env_steps_per_update = args.rollout * args.n_envs
    total_env_steps = args.steps * env_steps_per_update * world_size
    value = total_env_steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "env-steps/sec (whole node) PPO EURUSD-1m 4096 vec-envs",
            "value": value,
            "unit": "env-steps/sec",
            "n_gpus": world_size if have_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "phases_ms": {"rollout": rollout_ms, "update": update_ms},
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": ("PPO LSTM(256) actor-critic bf16 MFMA" if args.policy == "lstm"
                          else "PPO MLP(256,256) actor-critic bf16 MFMA"),
                "global_batch": env_steps_per_update * world_size,
                "seq_len": 32,
                "parallelism": f"dp{world_size}",
                "n_envs_per_gpu": args.n_envs,
                "rollout_steps": args.rollout,
                "obs_dim": env.obs_dim,
                "preprocessor": "feature_window_preprocessor",
                "reward": cfg["reward_plugin"],
                "strategy": cfg["strategy_plugin"],
                "pairs": args.pairs,
            },
        }))

    if pg is not None:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
