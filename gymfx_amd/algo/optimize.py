"""mode=optimization: hyperparameter search over a strategy's hparam_schema.

The reference only validates the mode and exposes ``hparam_schema()`` on
strategies for an EXTERNAL GA tuner (/root/reference/strategy_plugins/
direct_atr_sltp.py:344-350, app/main.py:82-83).  Here the tuner is built in:
deterministic random search (optionally evolutionary halving) over the
schema, each trial a short vectorized run on device, scored by the metrics
plugin's risk-adjusted return — thousands of envs per trial make this cheap
on an MI355X.
"""
from __future__ import annotations


from typing import Any, Dict, List, Sequence, Tuple

import numpy as np
import torch

Schema = Sequence[Tuple[str, float, float, str]]


def sample_params(schema: Schema, rng: np.random.Generator) -> Dict[str, Any]:
    out: Dict[str, Any] = {}
    for name, lo, hi, kind in schema:
        if kind == "int":
            out[name] = int(rng.integers(int(lo), int(hi) + 1))
        else:
            out[name] = float(rng.uniform(float(lo), float(hi)))
    return out


def _policy_driver(cfg: Dict[str, Any], env) -> "callable":
    """Greedy-action driver from a trained policy (driver_mode=policy/train):
    trials then score what the POLICY would do under the trial's
    hyperparameters, so optimization can tune anything the PPO path cares
    about."""
    from ..utils.checkpoint import load_model_for_inference

    ckpt = cfg.get("checkpoint_file")
    if not ckpt:
        raise ValueError("driver_mode=policy requires checkpoint_file")
    model, meta = load_model_for_inference(ckpt, env.device)
    if meta["obs_dim"] != env.obs_dim:
        raise ValueError(
            f"policy obs_dim {meta['obs_dim']} != trial env {env.obs_dim}")
    return _greedy_driver(model, meta["policy"] == "lstm", env)


def _greedy_driver(model, recurrent: bool, env) -> "callable":
    acts = model.alloc_acts(env.n_envs)
    state = model.alloc_state(env.n_envs) if recurrent else None

    def drive(obs_f32: torch.Tensor) -> torch.Tensor:
        obs_bf16 = obs_f32.to(torch.bfloat16)
        if recurrent:
            head = model.step_forward(obs_bf16, state, acts)
        else:
            head = model.forward(obs_bf16, acts)
        return head[:, :-1].argmax(dim=1)

    return drive


def _score_trial(config: Dict[str, Any], trial: Dict[str, Any]) -> Dict[str, Any]:
    from .. import build_vec_environment

    cfg = dict(config)
    cfg.update(trial)
    driver = str(cfg.get("driver_mode", "random"))
    cfg.setdefault("autoreset", driver == "train")
    cfg.setdefault("env_start_mode", "spread")
    env = build_vec_environment(cfg)
    env.reset(seed=int(cfg.get("seed") or 0))
    steps = int(cfg.get("optimization_steps", 256))
    rng = np.random.default_rng(int(cfg.get("seed") or 0))
    N = env.n_envs
    policy_drive = None
    if driver == "policy":
        policy_drive = _policy_driver(cfg, env)
    elif driver == "train":
        # short PPO run per trial, then score the TRAINED policy greedily
        from .ppo import PPOConfig, PPOTrainer

        pc = PPOConfig.from_config(cfg)
        pc.rollout_steps = int(cfg.get("optimization_train_rollout", 32))
        pc.minibatches = int(cfg.get("optimization_train_minibatches", 4))
        pc.ppo_epochs = int(cfg.get("optimization_train_epochs", 2))
        trainer = PPOTrainer(env, pc)
        for _ in range(int(cfg.get("optimization_train_updates", 4))):
            trainer.train_update(with_stats=False)
        policy_drive = _greedy_driver(trainer.model, trainer.recurrent, env)
        env.reset(seed=int(cfg.get("seed") or 0) + 1)  # fresh eval episode
    obs = env._obs
    for i in range(steps):
        if policy_drive is not None:
            a = policy_drive(obs)
        elif driver == "buy_hold":
            a = torch.ones(N, dtype=torch.int64, device=env.device)
        elif driver == "flat":
            a = torch.zeros(N, dtype=torch.int64, device=env.device)
        else:
            a = torch.from_numpy(rng.integers(0, 3, size=N)).to(env.device)
        info = env.step(a)
        obs = info.get("obs", env._obs)
        if bool(env.st.terminated.all()):
            break
    ic = env.params.initial_cash
    eq = env.st.equity
    total_return = float((eq.mean().item() - ic) / ic)
    dd_frac = float((env.st.max_dd_pct / 100.0).mean().item())
    lam = float(cfg.get("dd_penalty_lambda", 1.0))
    rap = total_return - lam * dd_frac
    return {
        "params": trial,
        "total_return": total_return,
        "mean_drawdown_frac": dd_frac,
        "rap": rap,
        "mean_trades": float(env.st.trade_count.float().mean().item()),
    }


def optimize_from_config(config: Dict[str, Any]) -> Dict[str, Any]:
    """Random search (optionally with halving refinement around the best)."""
    from ..plugins import load_plugin

    strat_name = str(config.get("strategy_plugin", "default_strategy"))
    klass, _ = load_plugin("strategy.plugins", strat_name)
    inst = klass(config)
    schema_fn = getattr(inst, "hparam_schema", None)
    if schema_fn is None or not schema_fn():
        raise ValueError(
            f"strategy '{strat_name}' exposes no hparam_schema to optimize")
    schema = list(schema_fn())
    trials = int(config.get("optimization_trials", 16))
    rng = np.random.default_rng(int(config.get("seed") or 0))
    results: List[Dict[str, Any]] = []
    for t in range(trials):
        trial = sample_params(schema, rng)
        res = _score_trial(config, trial)
        res["trial"] = t
        results.append(res)
        if not config.get("quiet_mode"):
            print(f"trial {t}: rap={res['rap']:.6f} {trial}")
    results.sort(key=lambda r: r["rap"], reverse=True)
    # optional refinement: re-sample around the incumbent best inside a
    # narrowed range (+-25% of each parameter's original span)
    refine = int(config.get("optimization_refine_trials", 0) or 0)
    for t in range(refine):
        center = results[0]["params"]
        narrowed: List[Tuple[str, float, float, str]] = []
        for name, lo, hi, kind in schema:
            span = (float(hi) - float(lo)) * 0.25
            c = float(center[name])
            narrowed.append((name, max(float(lo), c - span),
                             min(float(hi), c + span), kind))
        trial = sample_params(narrowed, rng)
        res = _score_trial(config, trial)
        res["trial"] = trials + t
        res["refined"] = True
        results.append(res)
        results.sort(key=lambda r: r["rap"], reverse=True)
        if not config.get("quiet_mode"):
            print(f"refine {t}: rap={res['rap']:.6f} {trial}")
    best = results[0]
    return {
        "mode": "optimization",
        "strategy_plugin": strat_name,
        "schema": [list(s) for s in schema],
        "trials": trials + refine,
        "best_params": best["params"],
        "best": best,
        "top5": results[:5],
    }
