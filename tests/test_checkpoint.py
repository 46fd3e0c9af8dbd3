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


def test_train_then_evaluate_roundtrip(tmp_path):
    """mode=training with a checkpoint, then vectorized greedy evaluation
    from that checkpoint (the serving-side counterpart of training)."""
    import json

    from gymfx_amd.main import main as cli_main

    ckpt = str(tmp_path / "m.pt")
    train_res = tmp_path / "train.json"
    eval_res = tmp_path / "eval.json"
    common = [
        "--quiet_mode", "true",
        "--data_feed_plugin", "synthetic_data_feed",
        "--synthetic_rows", "2000",
        "--n_envs", "16",
        "--window_size", "8",
        "--position_size", "1000.0",
        "--rollout_steps", "16",
        "--minibatches", "2",
        "--ppo_epochs", "1",
        "--hidden_size", "16",
        "--seed", "0",
        "--checkpoint_file", ckpt,
        "--save_config", "",
    ]
    cli_main(["--mode", "training", "--train_updates", "2",
              "--results_file", str(train_res), *common])
    out = json.loads(train_res.read_text())
    assert out["updates"] == 2 and out["checkpoint_file"] == ckpt
    cli_main(["--mode", "inference", "--eval_steps", "64",
              "--results_file", str(eval_res), *common])
    ev = json.loads(eval_res.read_text())
    assert ev["mode"] == "inference"
    assert ev["eval_steps"] == 64 and ev["n_envs"] == 16
    assert ev["mean_final_equity"] > 0
    assert ev["total_trades"] >= 0


def test_trace_file_phase_timings(tmp_path):
    """trace_file emits one JSONL record per update with rollout/update
    phase wall-clock (SURVEY §5.1 observability upgrade)."""
    import json

    from gymfx_amd.main import main as cli_main

    trace = tmp_path / "trace.jsonl"
    cli_main([
        "--mode", "training", "--quiet_mode", "true",
        "--data_feed_plugin", "synthetic_data_feed",
        "--synthetic_rows", "1200",
        "--n_envs", "16", "--window_size", "8",
        "--rollout_steps", "16", "--minibatches", "2", "--ppo_epochs", "1",
        "--hidden_size", "16", "--seed", "0", "--train_updates", "3",
        "--trace_file", str(trace),
        "--results_file", str(tmp_path / "r.json"), "--save_config", "",
    ])
    lines = [json.loads(l) for l in trace.read_text().splitlines()]
    assert len(lines) == 3
    for i, rec in enumerate(lines):
        assert rec["update"] == i
        assert rec["phases_ms"]["rollout"] > 0
        assert rec["phases_ms"]["update"] > 0
        assert "pi_loss" in rec


def test_checkpoint_save_is_atomic(tmp_path, monkeypatch):
    """A crash during torch.save must leave any pre-existing checkpoint
    intact (save goes to a temp sibling, then os.replace)."""
    import torch as _torch

    from gymfx_amd.utils import checkpoint as ck

    path = str(tmp_path / "c.pt")
    t = _make("mlp")
    save_checkpoint(t, path)
    good = open(path, "rb").read()

    real_save = _torch.save

    def boom(obj, f, *a, **kw):
        # write partial garbage then die, like an OOM/kill mid-serialize
        with open(f, "wb") as fh:
            fh.write(b"partial")
        raise RuntimeError("crash mid-save")

    monkeypatch.setattr(_torch, "save", boom)
    with pytest.raises(RuntimeError):
        ck.save_checkpoint(t, path)
    assert open(path, "rb").read() == good  # original untouched
    monkeypatch.setattr(_torch, "save", real_save)
    t2 = _make("mlp")
    load_checkpoint(t2, path)  # still loadable


def test_load_model_for_inference_ignores_n_envs(tmp_path):
    """Inference load takes only the model shape from the checkpoint —
    training-time n_envs is informational (serving from a
    4096-env checkpoint with a 16-env template must work)."""
    from gymfx_amd.utils.checkpoint import load_model_for_inference

    path = str(tmp_path / "ckpt.pt")
    tr = _make("mlp")
    tr.train_update()
    save_checkpoint(tr, path)
    model, meta = load_model_for_inference(path, torch.device("cpu"))
    assert meta["policy"] == "mlp" and meta["n_envs"] == 16
    assert meta["obs_dim"] == tr.env.obs_dim and meta["hidden"] == 16
    assert torch.equal(model.params, tr.model.params)


def test_checkpoint_loads_with_weights_only(tmp_path):
    """The checkpoint payload must stay within torch.load(weights_only=True)
    territory — no pickled code objects in the file."""
    path = str(tmp_path / "ckpt.pt")
    tr = _make("lstm")
    tr.train_update()
    save_checkpoint(tr, path, extra={"config": {"seed": 31, "note": "x"}})
    sd = torch.load(path, map_location="cpu", weights_only=True)
    assert sd["schema"] == "gymfx.ckpt.v1"


def test_periodic_checkpoint_interval(tmp_path):
    """checkpoint_interval=N saves inside the update loop, so a crash after
    update k>=N resumes from the last interval, not from zero."""
    from gymfx_amd.algo.ppo import train_from_config
    from gymfx_amd.config import DEFAULT_VALUES

    path = str(tmp_path / "ckpt.pt")
    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed", "synthetic_rows": 600,
           "n_envs": 8, "window_size": 8, "device": "cpu", "seed": 3,
           "rollout_steps": 8, "minibatches": 2, "ppo_epochs": 1,
           "hidden_size": 16, "train_updates": 5, "checkpoint_interval": 2,
           "checkpoint_file": path, "quiet_mode": True}
    saves = []
    import gymfx_amd.utils.checkpoint as ck
    orig = ck.save_checkpoint

    def spy(trainer, p, **kw):
        saves.append(trainer.update_count)
        return orig(trainer, p, **kw)

    ck.save_checkpoint = spy
    try:
        train_from_config(cfg)
    finally:
        ck.save_checkpoint = orig
    # interval saves after updates 2 and 4, final save after 5
    assert saves == [2, 4, 5]


def test_evaluate_loads_checkpoint_from_different_n_envs(tmp_path):
    """mode=inference evaluation only needs model weights: a 16-env
    checkpoint must evaluate on a different fleet size (same ADVICE class
    as serving)."""
    from gymfx_amd.algo.evaluate import evaluate_from_config
    from gymfx_amd.config import DEFAULT_VALUES

    tr = _make("lstm")
    tr.train_update()
    path = str(tmp_path / "e.pt")
    save_checkpoint(tr, path)
    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed", "synthetic_rows": 500,
           "n_envs": 4, "window_size": 8, "device": "cpu", "seed": 9,
           "checkpoint_file": path, "eval_steps": 12, "hidden_size": 16,
           "position_size": 1000.0, "metrics_plugin": "trading_metrics"}
    out = evaluate_from_config(cfg)
    assert out["policy_model"] == "lstm" and out["n_envs"] == 4
    assert out["eval_steps"] > 0
    # fleet-level digest through the configured metrics plugin
    assert out["metrics"]["metric_schema"] == "trading.metrics.v1"
    assert out["metrics"]["initial_cash"] == cfg["initial_cash"]
    assert "risk_adjusted_return" in out["metrics"] or "rap" in out["metrics"]
