"""Checkpoint / resume.

The reference has no model checkpointing (it has no model); what it persists
is config diffs (/root/reference/app/config_handler.py:11-24) and action
replays.  Here a checkpoint is the full training state: policy/optimizer
tensors, the vectorized-env SoA state (every tensor of EnvState), the
device RNG/schedule counters, and the trainer's host counters — enough that
``resume + K updates`` is bit-identical to ``no interruption`` (verified by
tests/test_checkpoint.py).

Layout is plain ``torch.save`` of a nested dict (schema gymfx.ckpt.v1).
"""
from __future__ import annotations

from typing import Any, Dict

import torch

SCHEMA = "gymfx.ckpt.v1"


def trainer_state_dict(trainer) -> Dict[str, Any]:
    env = trainer.env
    out: Dict[str, Any] = {
        "schema": SCHEMA,
        "model": trainer.model.state_dict(),
        "env_state": {k: v.detach().cpu() for k, v in env.st.to_dict().items()},
        "env_obs": env._obs.detach().cpu(),
        "step_base": int(trainer.step_base.item()),
        "mb_ctr": int(trainer.mb_ctr.item()),
        "global_step": trainer.global_step,
        "update_count": trainer.update_count,
        "policy": trainer.cfg.policy,
        "n_envs": env.n_envs,
        "obs_dim": env.obs_dim,
    }
    if trainer.recurrent:
        out["rnn_h"] = trainer.rnn_state["h"].detach().cpu()
        out["rnn_c"] = trainer.rnn_state["c"].detach().cpu()
    return out


def load_trainer_state_dict(trainer, sd: Dict[str, Any]) -> None:
    if sd.get("schema") != SCHEMA:
        raise ValueError(f"unknown checkpoint schema: {sd.get('schema')!r}")
    if sd["policy"] != trainer.cfg.policy:
        raise ValueError(
            f"checkpoint policy {sd['policy']!r} != trainer {trainer.cfg.policy!r}")
    if sd["n_envs"] != trainer.env.n_envs or sd["obs_dim"] != trainer.env.obs_dim:
        raise ValueError("checkpoint env shape mismatch")
    trainer.model.load_state_dict(sd["model"])
    st = trainer.env.st.to_dict()
    for k, v in sd["env_state"].items():
        st[k].copy_(v.to(st[k].device))
    trainer.env._obs.copy_(sd["env_obs"].to(trainer.env.device))
    trainer.step_base.fill_(sd["step_base"])
    trainer.mb_ctr.fill_(sd["mb_ctr"])
    trainer.global_step = int(sd["global_step"])
    trainer.update_count = int(sd["update_count"])
    if trainer.recurrent:
        trainer.rnn_state["h"].copy_(sd["rnn_h"].to(trainer.env.device))
        trainer.rnn_state["c"].copy_(sd["rnn_c"].to(trainer.env.device))


def save_checkpoint(trainer, path: str, *, extra: Dict[str, Any] = None) -> str:
    """Atomic save: write to a sibling temp file, then rename — a crash
    mid-save never corrupts an existing resume file."""
    import os

    sd = trainer_state_dict(trainer)
    if extra:
        sd["extra"] = extra
    tmp = f"{path}.tmp.{os.getpid()}"
    torch.save(sd, tmp)
    os.replace(tmp, path)
    return path


def load_checkpoint(trainer, path: str) -> Dict[str, Any]:
    # weights_only: the payload is tensors / scalars / plain dicts only, so
    # the restricted unpickler suffices — an untrusted checkpoint file can
    # not execute code on load.
    sd = torch.load(path, map_location="cpu", weights_only=True)
    load_trainer_state_dict(trainer, sd)
    return sd.get("extra", {})


def load_model_for_inference(path: str, device) -> tuple:
    """Weights-only load for serving / evaluation.

    Unlike :func:`load_trainer_state_dict` this does NOT require the
    checkpoint's ``n_envs`` to match anything — a policy trained with
    n_envs=4096 serves from a 16-env template just fine.  Only the model
    shape (policy kind, obs_dim, hidden) is taken from the checkpoint.

    Returns ``(model, meta)`` where ``meta`` has ``policy``, ``obs_dim``,
    ``hidden``, ``n_envs`` (the training-time value, informational) and
    ``update_count``.
    """
    sd = torch.load(path, map_location="cpu", weights_only=True)
    if sd.get("schema") != SCHEMA:
        raise ValueError(f"unknown checkpoint schema: {sd.get('schema')!r}")
    policy = sd["policy"]
    obs_dim = int(sd["obs_dim"])
    msd = sd["model"]
    hidden = int(msd["hidden"].item()) if "hidden" in msd else 256
    if policy == "lstm":
        from ..models.lstm import ActorCriticLSTM

        model = ActorCriticLSTM(obs_dim, 3, hidden, device=device, seed=0)
    else:
        from ..models.mlp import ActorCriticMLP

        model = ActorCriticMLP(obs_dim, 3, hidden, device=device, seed=0)
    model.load_state_dict(msd)
    meta = {
        "policy": policy,
        "obs_dim": obs_dim,
        "hidden": hidden,
        "n_envs": int(sd.get("n_envs", 0)),
        "update_count": int(sd.get("update_count", 0)),
    }
    return model, meta
