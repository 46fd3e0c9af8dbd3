"""GPU end-to-end deployment chain: train on device -> atomic checkpoint
-> inference-only load into a PolicyServer on the same device -> batched
HTTP acts through the MFMA forward kernels (the README quickstart chain,
on hardware; ADVICE r1 medium item covered on CPU in test_serve.py)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("policy", ["mlp", "lstm"])
def test_train_checkpoint_serve_chain_on_gpu(tmp_path, policy):
    from fastapi.testclient import TestClient

    from gymfx_amd import build_vec_environment
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.serve import create_app
    from gymfx_amd.utils.checkpoint import save_checkpoint

    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed",
           "synthetic_rows": 3000, "n_envs": 256, "window_size": 8,
           "device": "cuda", "seed": 11, "env_start_mode": "spread",
           "autoreset": True, "position_size": 1000.0,
           "policy_model": policy, "hidden_size": 64, "bptt_len": 8}
    env = build_vec_environment(cfg)
    env.reset(seed=11)
    pc = PPOConfig.from_config({**cfg, "rollout_steps": 16,
                                "minibatches": 2, "ppo_epochs": 1})
    tr = PPOTrainer(env, pc)
    for _ in range(2):
        tr.train_update()
    ckpt = str(tmp_path / f"{policy}_gpu.pt")
    save_checkpoint(tr, ckpt)

    # serve from a DIFFERENT (smaller) template env on the same device —
    # the inference-only load must not care about the training n_envs
    app = create_app({**cfg, "n_envs": 8, "checkpoint_file": ckpt})
    client = TestClient(app)
    h = client.get("/health").json()
    assert h["policy"] == policy and h["device"].startswith("cuda")
    D = h["obs_dim"]

    rng = np.random.default_rng(3)
    obs = rng.normal(size=(64, D)).astype(np.float32).tolist()
    r = client.post("/act", json={"obs": obs, "greedy": True,
                                  "session": "s1"}).json()
    assert len(r["actions"]) == 64
    assert all(0 <= a <= 2 for a in r["actions"])
    assert np.isfinite(r["logp"]).all() and np.isfinite(r["value"]).all()
    if policy == "mlp":
        r2 = client.post("/act", json={"obs": obs, "greedy": True}).json()
        assert r2["actions"] == r["actions"]
    else:
        # recurrent session continuity: same obs again through the SAME
        # session advances the state (no crash, valid actions), and a
        # mismatched batch on that session is refused with 409
        r2 = client.post("/act", json={"obs": obs, "greedy": True,
                                       "session": "s1"}).json()
        assert len(r2["actions"]) == 64
        bad = client.post("/act", json={"obs": obs[:8], "greedy": True,
                                        "session": "s1"})
        assert bad.status_code == 409
