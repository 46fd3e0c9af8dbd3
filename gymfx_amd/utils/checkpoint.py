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
    sd = torch.load(path, map_location="cpu", weights_only=False)
    load_trainer_state_dict(trainer, sd)
    return sd.get("extra", {})
