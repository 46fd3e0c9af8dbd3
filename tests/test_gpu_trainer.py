"""GPU: PPO trainer hipGraph path vs eager path — bit-equivalence.

The captured graphs replay the exact kernel sequence of the eager bodies
(device counters carry RNG/Adam state; capture warmup is snapshot/restored),
so graph mode must produce bit-identical parameters to eager mode."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _make_trainer(use_graphs):
    from gymfx_amd import build_vec_environment
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(2000, seed=5, vol=4e-4, extra_feature_columns=3)
    cfg = {
        "n_envs": 256,
        "device": "cuda",
        "window_size": 16,
        "preprocessor_plugin": "feature_window_preprocessor",
        "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE",
                            "FEAT_0", "FEAT_1", "FEAT_2"],
        "env_start_mode": "spread",
        "autoreset": True,
        "position_size": 1000.0,
        "seed": 3,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=3)
    pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=3,
                   use_graphs=use_graphs)
    return PPOTrainer(env, pc)


def test_graph_replay_equals_eager():
    tg = _make_trainer(use_graphs=True)
    te = _make_trainer(use_graphs=False)
    for _ in range(3):
        sg = tg.train_update()
        se = te.train_update()
    torch.cuda.synchronize()
    assert tg._graphs_ready
    assert torch.equal(tg.model.params, te.model.params)
    assert torch.equal(tg.env.st.equity, te.env.st.equity)
    for k in sg:
        assert sg[k] == pytest.approx(se[k], rel=1e-5, abs=1e-7)


def test_graph_mode_deterministic_across_trainers():
    t1 = _make_trainer(use_graphs=True)
    t2 = _make_trainer(use_graphs=True)
    for _ in range(2):
        t1.train_update(with_stats=False)
        t2.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert torch.equal(t1.model.params, t2.model.params)


def _make_rec_trainer(use_graphs):
    from gymfx_amd import build_vec_environment
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd.data.feed import synthetic_ohlcv

    md = synthetic_ohlcv(2000, seed=5, vol=4e-4, extra_feature_columns=3)
    cfg = {
        "n_envs": 256,
        "device": "cuda",
        "window_size": 16,
        "preprocessor_plugin": "feature_window_preprocessor",
        "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE",
                            "FEAT_0", "FEAT_1", "FEAT_2"],
        "env_start_mode": "spread",
        "autoreset": True,
        "position_size": 1000.0,
        "seed": 4,
    }
    env = build_vec_environment(cfg, md)
    env.reset(seed=4)
    pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=4,
                   policy="lstm", bptt_len=4, use_graphs=use_graphs)
    return PPOTrainer(env, pc)


def test_recurrent_graph_replay_equals_eager():
    tg = _make_rec_trainer(use_graphs=True)
    te = _make_rec_trainer(use_graphs=False)
    for _ in range(3):
        sg = tg.train_update()
        se = te.train_update()
    torch.cuda.synchronize()
    assert tg._graphs_ready
    assert torch.equal(tg.model.params, te.model.params)
    for k in sg:
        assert sg[k] == pytest.approx(se[k], rel=1e-5, abs=1e-7)


def test_checkpoint_resume_bit_identical_on_gpu():
    from gymfx_amd.utils.checkpoint import load_checkpoint, save_checkpoint
    import tempfile, os

    path = os.path.join(tempfile.mkdtemp(), "ckpt.pt")
    t_full = _make_trainer(use_graphs=True)
    for _ in range(2):
        t_full.train_update(with_stats=False)
    save_checkpoint(t_full, path)
    for _ in range(2):
        t_full.train_update(with_stats=False)
    t_res = _make_trainer(use_graphs=True)
    load_checkpoint(t_res, path)
    for _ in range(2):
        t_res.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert torch.equal(t_full.model.params, t_res.model.params)
    assert torch.equal(t_full.env.st.equity, t_res.env.st.equity)


def test_split_rollout_equals_single_stream():
    """Two-stream split rollout must be bit-identical to one stream (row
    blocks are tile-aligned; the sampler draws by global row index)."""
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    def make(streams):
        md = synthetic_ohlcv(2000, seed=5, vol=4e-4, extra_feature_columns=3)
        cfg = {"n_envs": 256, "device": "cuda", "window_size": 16,
               "preprocessor_plugin": "feature_window_preprocessor",
               "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE",
                                   "FEAT_0", "FEAT_1", "FEAT_2"],
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 9}
        env = build_vec_environment(cfg, md)
        env.reset(seed=9)
        pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=9,
                       rollout_streams=streams)
        return PPOTrainer(env, pc)

    t1 = make(1)
    t2 = make(2)
    for _ in range(2):
        t1.train_update(with_stats=False)
        t2.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert t2._split
    assert torch.equal(t1.act_buf, t2.act_buf)
    assert torch.equal(t1.model.params, t2.model.params)
    assert torch.equal(t1.env.st.equity, t2.env.st.equity)


def test_fused_rollout_equals_unfused():
    """The fused MLP policy kernel (2 MFMA layers + MFMA head + sampler in
    one launch) must be BITWISE identical to the unfused GEMM+sampler
    chain (same BK chunking, MFMA sequence, bias/tanh/rounding, RNG)."""
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    def make(fused):
        md = synthetic_ohlcv(2000, seed=5, vol=4e-4, extra_feature_columns=3)
        cfg = {"n_envs": 256, "device": "cuda", "window_size": 16,
               "preprocessor_plugin": "feature_window_preprocessor",
               "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE",
                                   "FEAT_0", "FEAT_1", "FEAT_2"],
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 13}
        env = build_vec_environment(cfg, md)
        env.reset(seed=13)
        pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=13,
                       fused_rollout=fused)
        return PPOTrainer(env, pc)

    tf = make(True)
    tu = make(False)
    for _ in range(2):
        tf.train_update(with_stats=False)
        tu.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert tf._fused and not tu._fused
    assert torch.equal(tf.act_buf, tu.act_buf)
    assert torch.equal(tf.logp_buf, tu.logp_buf)
    assert torch.equal(tf.val_buf, tu.val_buf)
    assert torch.equal(tf.model.params, tu.model.params)


@pytest.mark.parametrize("policy", ["mlp", "lstm"])
def test_fuse_sample_equals_sample_head(policy):
    """In-env-kernel sampling (fuse_sample) must be BITWISE identical to the
    standalone sample_head kernel chain: same shared helpers (env_common.h),
    same RNG keying (seed, step_base+t, global env row)."""
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    def make(fuse):
        md = synthetic_ohlcv(2000, seed=7, vol=4e-4)
        cfg = {"n_envs": 256, "device": "cuda", "window_size": 16,
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 21}
        env = build_vec_environment(cfg, md)
        env.reset(seed=21)
        pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=21,
                       policy=policy, bptt_len=4, hidden=64, fuse_sample=fuse)
        return PPOTrainer(env, pc)

    tf = make(True)
    tu = make(False)
    for _ in range(2):
        tf.train_update(with_stats=False)
        tu.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert tf._fuse_sample and not tu._fuse_sample
    assert torch.equal(tf.act_buf, tu.act_buf)
    assert torch.equal(tf.logp_buf, tu.logp_buf)
    assert torch.equal(tf.val_buf, tu.val_buf)
    assert torch.equal(tf.rew_buf, tu.rew_buf)
    assert torch.equal(tf.model.params, tu.model.params)
    assert torch.equal(tf.env.st.equity, tu.env.st.equity)


def test_overlap_gather_equals_serial():
    """Side-stream minibatch gather (ping-pong slots) must be bitwise
    identical to the serial single-slot pipeline — same kernels, same
    Feistel sequence, only the stream placement differs."""
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    def make(overlap):
        md = synthetic_ohlcv(2000, seed=9, vol=4e-4)
        cfg = {"n_envs": 256, "device": "cuda", "window_size": 16,
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 17}
        env = build_vec_environment(cfg, md)
        env.reset(seed=17)
        pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=3, seed=17,
                       overlap_gather=overlap)
        return PPOTrainer(env, pc)

    to = make(True)
    ts = make(False)
    for _ in range(3):
        to.train_update(with_stats=False)
        ts.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert to._overlap and not ts._overlap
    assert torch.equal(to.model.params, ts.model.params)
    assert torch.equal(to.model.m, ts.model.m)
    # losses are atomicAdd logging sums: order (not value) may differ
    assert torch.allclose(to.losses, ts.losses, rtol=1e-5, atol=1e-6)
    assert torch.equal(to.env.st.equity, ts.env.st.equity)


def test_trace_file_on_gpu(tmp_path):
    """trace_file phase timing must work when the phases are hipGraph
    replays (HIP events bracket graph launches; lazy drain)."""
    import json

    from gymfx_amd.algo.ppo import train_from_config
    from gymfx_amd.config import DEFAULT_VALUES

    trace = tmp_path / "t.jsonl"
    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed",
           "synthetic_rows": 3000, "n_envs": 256, "window_size": 8,
           "env_start_mode": "spread", "autoreset": True,
           "position_size": 1000.0, "device": "cuda", "seed": 3,
           "rollout_steps": 16, "minibatches": 2, "ppo_epochs": 1,
           "hidden_size": 64, "train_updates": 3, "quiet_mode": True,
           "trace_file": str(trace)}
    out = train_from_config(cfg)
    assert out["updates"] == 3
    lines = [json.loads(l) for l in trace.read_text().splitlines()]
    assert len(lines) == 3
    for rec in lines:
        assert rec["phases_ms"]["rollout"] > 0
        assert rec["phases_ms"]["update"] > 0


def test_fuse_gather_equals_materialized():
    """Gather+first-GEMM fusion (L1 fwd and W1 wgrad read obs rows through
    the epoch permutation) must be BITWISE identical to the materialized
    obs_mb path: identical bf16 rows, identical MFMA K order."""
    from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
    from gymfx_amd import build_vec_environment
    from gymfx_amd.data.feed import synthetic_ohlcv

    def make(fuse):
        md = synthetic_ohlcv(2000, seed=9, vol=4e-4)
        cfg = {"n_envs": 256, "device": "cuda", "window_size": 16,
               "env_start_mode": "spread", "autoreset": True,
               "position_size": 1000.0, "seed": 31}
        env = build_vec_environment(cfg, md)
        env.reset(seed=31)
        pc = PPOConfig(rollout_steps=16, minibatches=4, ppo_epochs=2, seed=31,
                       hidden=64)
        t = PPOTrainer(env, pc)
        t._fuse_gather = fuse
        return t

    tf = make(True)
    tu = make(False)
    for _ in range(3):
        tf.train_update(with_stats=False)
        tu.train_update(with_stats=False)
    torch.cuda.synchronize()
    assert torch.equal(tf.model.params, tu.model.params)
    assert torch.equal(tf.model.grads, tu.model.grads)
    assert torch.equal(tf.act_buf, tu.act_buf)
    assert torch.equal(tf.env.st.equity, tu.env.st.equity)
