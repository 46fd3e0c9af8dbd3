"""Checkpoint/resume: `resume + K updates` must be bit-identical to an
uninterrupted run (MI355X upgrade of the reference's config-diff persistence,
/root/reference/app/config_handler.py:11-24 — SURVEY.md §5.4)."""
import pytest
import torch

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.utils.checkpoint import load_checkpoint, save_checkpoint


def _make(policy="mlp", seed=31):
    md = synthetic_ohlcv(600, seed=6, vol=4e-4)
    cfg = {
        "n_envs": 16,
        "device": "cpu",
        "window_size": 8,
        "env_start_mode": "spread",
        "autoreset": True,
        "position_size": 1000.0,
        "seed": seed,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=seed)
    pc = PPOConfig(rollout_steps=8, minibatches=2, ppo_epochs=2, seed=seed,
                   hidden=16, policy=policy, bptt_len=4)
    return PPOTrainer(env, pc)


@pytest.mark.parametrize("policy", ["mlp", "lstm"])
def test_resume_bit_identical(tmp_path, policy):
    path = str(tmp_path / "ckpt.pt")
    # uninterrupted: 3 + 2 updates
    t_full = _make(policy)
    for _ in range(3):
        t_full.train_update()
    save_checkpoint(t_full, path, extra={"note": "after-3"})
    for _ in range(2):
        t_full.train_update()

    # resumed: fresh trainer, load after-3, 2 more updates
    t_res = _make(policy)
    extra = load_checkpoint(t_res, path)
    assert extra == {"note": "after-3"}
    assert t_res.update_count == 3
    for _ in range(2):
        t_res.train_update()

    assert torch.equal(t_full.model.params, t_res.model.params)
    assert torch.equal(t_full.env.st.equity, t_res.env.st.equity)
    assert t_full.global_step == t_res.global_step


def test_checkpoint_rejects_mismatch(tmp_path):
    path = str(tmp_path / "ckpt.pt")
    t = _make("mlp")
    save_checkpoint(t, path)
    other = _make("lstm")
    with pytest.raises(ValueError, match="policy"):
        load_checkpoint(other, path)
