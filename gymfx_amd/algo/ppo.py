"""PPO trainer: on-device rollout -> GAE -> minibatch clipped-PPO updates.

The whole loop is device-resident: env steps are the fused HIP kernels
(VecFxEnv native path), the policy is the MFMA MLP (models/mlp.py), GAE /
advantage-normalize / loss-backward / Adam are the ops/api kernels.  On CPU
the same code runs against the torch oracle implementations (tested by the
CPU suite, gloo multi-rank included).

Data parallelism (BASELINE config #3): one process per GPU; gradients live
in ONE flat bucket (model.grads) all-reduced per minibatch over RCCL/xGMI
(parallel/ddp.py) — gradient volume is ~0.4 MB so the collective is
latency-bound and a single fused bucket is optimal on this topology.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import torch

from ..envs.vec_env import VecFxEnv
from ..models.mlp import ActorCriticMLP
from ..ops import api


@dataclass
class PPOConfig:
    rollout_steps: int = 128
    ppo_epochs: int = 4
    minibatches: int = 8
    gamma: float = 0.99
    gae_lambda: float = 0.95
    clip_eps: float = 0.2
    ent_coef: float = 0.01
    vf_coef: float = 0.5
    lr: float = 3e-4
    max_grad_norm: float = 0.5
    seed: int = 0
    hidden: int = 256
    normalize_adv: bool = True
    shuffle_rows: bool = True

    @classmethod
    def from_config(cls, cfg: Dict[str, Any]) -> "PPOConfig":
        out = cls()
        mapping = {
            "rollout_steps": "rollout_steps",
            "ppo_epochs": "ppo_epochs",
            "minibatches": "minibatches",
            "gamma": "gamma",
            "gae_lambda": "gae_lambda",
            "clip_eps": "clip_eps",
            "ent_coef": "ent_coef",
            "vf_coef": "vf_coef",
            "lr": "learning_rate",
            "max_grad_norm": "max_grad_norm",
            "hidden": "hidden_size",
            "normalize_adv": "normalize_adv",
            "shuffle_rows": "shuffle_rows",
        }
        for attr, key in mapping.items():
            if cfg.get(key) is not None:
                cur = getattr(out, attr)
                setattr(out, attr, type(cur)(cfg[key]))
        if cfg.get("seed") is not None:
            out.seed = int(cfg["seed"])
        return out


class PPOTrainer:
    def __init__(self, env: VecFxEnv, cfg: PPOConfig, *, rank: int = 0,
                 world_size: int = 1, process_group=None):
        self.env = env
        self.cfg = cfg
        self.rank = rank
        self.world_size = world_size
        self.pg = process_group
        self.device = env.device

        N = env.n_envs
        T = cfg.rollout_steps
        D = env.obs_dim
        self.N, self.T, self.D = N, T, D
        n_actions = 3

        self.model = ActorCriticMLP(
            D, n_actions, cfg.hidden, device=self.device, seed=cfg.seed + rank
        )

        dev = self.device
        self.obs_buf = torch.empty(T, N, D, dtype=torch.bfloat16, device=dev)
        self.act_buf = torch.empty(T, N, dtype=torch.int64, device=dev)
        self.logp_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.val_buf = torch.empty(T + 1, N, dtype=torch.float32, device=dev)
        self.rew_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.done_buf = torch.empty(T, N, dtype=torch.bool, device=dev)
        self.adv_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self.ret_buf = torch.empty(T, N, dtype=torch.float32, device=dev)
        self._adv_part = torch.empty(512, dtype=torch.float32, device=dev)

        self.acts_rollout = self.model.alloc_acts(N)
        self.obs_bf16_step = torch.empty(N, D, dtype=torch.bfloat16, device=dev)
        self.step_actions = torch.empty(N, dtype=torch.int64, device=dev)
        self.step_logp = torch.empty(N, dtype=torch.float32, device=dev)
        self.step_value = torch.empty(N, dtype=torch.float32, device=dev)

        M = (T * N) // cfg.minibatches
        self.mb_rows = M
        self.acts_train = self.model.alloc_acts(M)
        self.scratch = self.model.alloc_scratch(M)
        self.dhead = torch.empty(M, n_actions + 1, dtype=torch.bfloat16, device=dev)
        self.losses = torch.zeros(5, dtype=torch.float32, device=dev)
        self._perm_gen = torch.Generator(device="cpu").manual_seed(cfg.seed * 9973 + rank)

        self.global_step = 0   # env steps taken (per rank)
        self.update_count = 0

    # ------------------------------------------------------------------
    def collect_rollout(self) -> None:
        env, model, cfg = self.env, self.model, self.cfg
        T, N = self.T, self.N
        for t in range(T):
            api.f32_to_bf16(env._obs, self.obs_bf16_step)
            self.obs_buf[t].copy_(self.obs_bf16_step)
            head = model.forward(self.obs_bf16_step, self.acts_rollout)
            api.sample_head(
                head, cfg.seed * 1_000_003 + self.rank, self.global_step + t,
                self.step_actions, self.step_logp, self.step_value,
            )
            self.act_buf[t].copy_(self.step_actions)
            self.logp_buf[t].copy_(self.step_logp)
            self.val_buf[t].copy_(self.step_value)
            out = env.step(self.step_actions)
            self.rew_buf[t].copy_(out["reward"])
            self.done_buf[t].copy_(out["terminated"])
        # bootstrap value
        api.f32_to_bf16(env._obs, self.obs_bf16_step)
        head = model.forward(self.obs_bf16_step, self.acts_rollout)
        self.val_buf[T].copy_(head[:, -1])
        self.global_step += T

    def compute_advantages(self) -> None:
        api.gae(
            self.rew_buf, self.val_buf, self.done_buf, self.adv_buf,
            self.ret_buf, self.cfg.gamma, self.cfg.gae_lambda,
        )
        if self.cfg.normalize_adv:
            api.adv_normalize(self.adv_buf.view(-1), self._adv_part)

    def update(self) -> Dict[str, float]:
        cfg, model = self.cfg, self.model
        TN = self.T * self.N
        M = self.mb_rows
        obs_flat = self.obs_buf.view(TN, self.D)
        act_flat = self.act_buf.view(TN)
        logp_flat = self.logp_buf.view(TN)
        adv_flat = self.adv_buf.view(TN)
        ret_flat = self.ret_buf.view(TN)
        self.losses.zero_()
        inv_count = 1.0 / (M * cfg.ppo_epochs * cfg.minibatches)

        for _ in range(cfg.ppo_epochs):
            if cfg.shuffle_rows:
                perm = torch.randperm(TN, generator=self._perm_gen).to(self.device)
                obs_e = obs_flat[perm]
                act_e = act_flat[perm]
                logp_e = logp_flat[perm]
                adv_e = adv_flat[perm]
                ret_e = ret_flat[perm]
            else:
                obs_e, act_e, logp_e, adv_e, ret_e = (
                    obs_flat, act_flat, logp_flat, adv_flat, ret_flat
                )
            for mb in range(cfg.minibatches):
                sl = slice(mb * M, (mb + 1) * M)
                obs_mb = obs_e[sl].contiguous()
                model.zero_grad()
                head = model.forward(obs_mb, self.acts_train)
                api.ppo_loss_bwd(
                    head, act_e[sl].contiguous(), logp_e[sl].contiguous(),
                    adv_e[sl].contiguous(), ret_e[sl].contiguous(), self.dhead,
                    clip_eps=cfg.clip_eps, ent_coef=cfg.ent_coef,
                    vf_coef=cfg.vf_coef, inv_count=1.0 / M,
                    losses=self.losses,
                )
                model.backward(obs_mb, self.acts_train, self.dhead, self.scratch)
                self._allreduce_grads()
                model.adam(cfg.lr, max_grad_norm=cfg.max_grad_norm)
        self.update_count += 1
        n_mb = cfg.ppo_epochs * cfg.minibatches
        lv = (self.losses / n_mb).cpu()
        return {
            "pi_loss": float(lv[0]),
            "v_loss": float(lv[1]),
            "entropy": float(lv[2]),
            "approx_kl": float(lv[3]),
            "clipfrac": float(lv[4]),
        }

    def _allreduce_grads(self) -> None:
        if self.world_size <= 1:
            return
        import torch.distributed as dist

        dist.all_reduce(self.model.grads, op=dist.ReduceOp.SUM, group=self.pg)
        self.model.grads.mul_(1.0 / self.world_size)

    def train_update(self) -> Dict[str, float]:
        """One full PPO update (rollout + GAE + epochs)."""
        self.collect_rollout()
        self.compute_advantages()
        return self.update()


# ---------------------------------------------------------------------------
# config-driven entry (mode=training in the CLI runner)
# ---------------------------------------------------------------------------

def train_from_config(config: Dict[str, Any]) -> Dict[str, Any]:
    from .. import build_vec_environment

    cfg = PPOConfig.from_config(config)
    vec_cfg = dict(config)
    vec_cfg.setdefault("autoreset", True)
    vec_cfg.setdefault("env_start_mode", "spread")
    env = build_vec_environment(vec_cfg)
    env.reset(seed=cfg.seed)
    trainer = PPOTrainer(env, cfg)
    updates = int(config.get("train_updates", 10))
    t0 = time.perf_counter()
    history: List[Dict[str, float]] = []
    for u in range(updates):
        stats = trainer.train_update()
        history.append(stats)
        if not config.get("quiet_mode") and (u % max(1, updates // 10) == 0):
            print(f"update {u}: {stats}")
    wall = time.perf_counter() - t0
    steps = trainer.global_step * env.n_envs
    summary = {
        "mode": "training",
        "updates": updates,
        "env_steps": steps,
        "wall_seconds": wall,
        "env_steps_per_sec": steps / wall if wall > 0 else 0.0,
        "final_stats": history[-1] if history else {},
        "vec_summary": env.vec_summary(),
    }
    return summary
