"""Vectorized policy evaluation: checkpoint -> greedy rollout -> metrics.

The serving-side counterpart of mode=training (the reference's inference
mode drives a single env with a scripted strategy, app/main.py:57-66; here
a trained policy runs greedily over thousands of device-resident envs and
the fleet's trading metrics are aggregated).
"""
from __future__ import annotations

import time
from typing import Any, Dict

import torch

from ..ops import api


def evaluate_from_config(config: Dict[str, Any]) -> Dict[str, Any]:
    from .. import build_vec_environment
    from .ppo import PPOConfig

    cfg = PPOConfig.from_config(config)
    vec_cfg = dict(config)
    vec_cfg.setdefault("autoreset", False)
    vec_cfg.setdefault("env_start_mode", "spread")
    env = build_vec_environment(vec_cfg)
    env.reset(seed=cfg.seed)
    ckpt = config.get("checkpoint_file")
    if ckpt:
        # inference-only load: evaluation needs the model weights, not the
        # training-time n_envs or env cursor (a 4096-env checkpoint must
        # evaluate on any fleet size)
        from ..utils.checkpoint import load_model_for_inference

        model, meta = load_model_for_inference(ckpt, env.device)
        if meta["obs_dim"] != env.obs_dim:
            raise ValueError(
                f"checkpoint obs_dim {meta['obs_dim']} != eval env obs_dim "
                f"{env.obs_dim} (feature/window config mismatch)")
        recurrent = meta["policy"] == "lstm"
        policy_name = meta["policy"]
    else:
        if cfg.policy == "lstm":
            from ..models.lstm import ActorCriticLSTM

            model = ActorCriticLSTM(env.obs_dim, 3, cfg.hidden,
                                    device=env.device, seed=cfg.seed)
        else:
            from ..models.mlp import ActorCriticMLP

            model = ActorCriticMLP(env.obs_dim, 3, cfg.hidden,
                                   device=env.device, seed=cfg.seed)
        recurrent = cfg.policy == "lstm"
        policy_name = cfg.policy
    steps = int(config.get("eval_steps", config.get("steps", 500)))
    N, D = env.n_envs, env.obs_dim
    dev = env.device
    obs_bf16 = torch.empty(N, D, dtype=torch.bfloat16, device=dev)
    actions = torch.empty(N, dtype=torch.int64, device=dev)
    logp = torch.empty(N, dtype=torch.float32, device=dev)
    acts = model.alloc_acts(N)
    state = model.alloc_state(N) if recurrent else None
    total_reward = torch.zeros(N, dtype=torch.float64, device=dev)
    t0 = time.perf_counter()
    n_steps = 0
    for t in range(steps):
        api.f32_to_bf16(env._obs, obs_bf16)
        if recurrent:
            head = model.step_forward(obs_bf16, state, acts)
        else:
            head = model.forward(obs_bf16, acts)
        api.sample_head(head, 0, t, actions, logp, greedy=True)
        out = env.step(actions)
        total_reward += out["reward"].to(torch.float64)
        if recurrent:
            api.mask_reset(state["h"], state["c"], out["terminated"])
        n_steps += 1
        if bool(env.st.terminated.all()):
            break
    if dev.type == "cuda":
        torch.cuda.synchronize()
    wall = time.perf_counter() - t0
    ic = env.params.initial_cash
    eq = env.st.equity
    # fleet-level trading-metrics digest through the CONFIGURED metrics
    # plugin (the reference's inference mode reports the metrics plugin's
    # summary, app/main.py:66-69 -> env.summary())
    from ..plugins import load_plugin

    mname = str(config.get("metrics_plugin", "trading_metrics"))
    klass, _ = load_plugin("metrics.plugins", mname)
    metrics = klass(config).summarize(
        initial_cash=ic, final_equity=float(eq.mean()),
        analyzers=env.fleet_analyzers(), config=config)
    return {
        "mode": "inference",
        "policy_model": policy_name,
        "checkpoint_file": ckpt,
        "eval_steps": n_steps,
        "n_envs": N,
        "env_steps_per_sec": n_steps * N / wall if wall > 0 else 0.0,
        "mean_final_equity": float(eq.mean()),
        "mean_total_return": float(((eq - ic) / ic).mean()),
        "mean_step_reward": float((total_reward / max(n_steps, 1)).mean()),
        "terminated_envs": int(env.st.terminated.sum()),
        "total_trades": int(env.st.trade_count.sum()),
        "metrics_plugin": mname,
        "metrics": metrics,
        "vec_summary": env.vec_summary(),
    }
