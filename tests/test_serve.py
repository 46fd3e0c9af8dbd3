"""Policy serving (mode=serve): checkpoint -> batched actions over HTTP.

The reference has no serving; this is the deployment-side counterpart of
mode=training (serve.py).  Tested in-process with the starlette TestClient
(no network)."""
import numpy as np
import pytest

from gymfx_amd import build_vec_environment
from gymfx_amd.algo.ppo import PPOConfig, PPOTrainer
from gymfx_amd.config import DEFAULT_VALUES
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.utils.checkpoint import save_checkpoint


def _serve_cfg(tmp_path, policy="mlp"):
    """Train 1 update, checkpoint, return a serve config for it."""
    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed",
           "synthetic_rows": 1200, "n_envs": 16, "window_size": 8,
           "device": "cpu", "seed": 4, "env_start_mode": "spread",
           "autoreset": True, "position_size": 1000.0,
           "policy_model": policy, "hidden_size": 16, "bptt_len": 4}
    env = build_vec_environment(cfg)
    env.reset(seed=4)
    pc = PPOConfig.from_config({**cfg, "rollout_steps": 8, "minibatches": 2,
                                "ppo_epochs": 1})
    pc.rollout_steps, pc.minibatches, pc.ppo_epochs = 8, 2, 1
    tr = PPOTrainer(env, pc)
    tr.train_update()
    ckpt = str(tmp_path / f"{policy}.pt")
    save_checkpoint(tr, ckpt)
    return {**cfg, "checkpoint_file": ckpt,
            "rollout_steps": 8, "minibatches": 2, "ppo_epochs": 1}


@pytest.mark.parametrize("policy", ["mlp", "lstm"])
def test_serve_health_and_act(tmp_path, policy):
    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    cfg = _serve_cfg(tmp_path, policy)
    app = create_app(cfg)
    client = TestClient(app)

    h = client.get("/health").json()
    assert h["status"] == "ok" and h["policy"] == policy
    D = h["obs_dim"]

    rng = np.random.default_rng(0)
    obs = rng.normal(size=(5, D)).astype(np.float32).tolist()
    r = client.post("/act", json={"obs": obs, "greedy": True}).json()
    assert len(r["actions"]) == 5
    assert all(0 <= a <= 2 for a in r["actions"])
    assert len(r["logp"]) == 5 and len(r["value"]) == 5
    # greedy is deterministic for the same obs (stateless mlp only)
    if policy == "mlp":
        r2 = client.post("/act", json={"obs": obs, "greedy": True}).json()
        assert r2["actions"] == r["actions"]
    # wrong obs width -> 422, not a crash
    bad = client.post("/act", json={"obs": [[1.0, 2.0]]})
    assert bad.status_code == 422


def test_serve_recurrent_sessions(tmp_path):
    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    app = create_app(_serve_cfg(tmp_path, "lstm"))
    client = TestClient(app)
    D = client.get("/health").json()["obs_dim"]
    rng = np.random.default_rng(1)
    obs = rng.normal(size=(3, D)).astype(np.float32).tolist()
    # two sessions evolve independent recurrent state
    a1 = client.post("/act", json={"obs": obs, "session": "s1"}).json()
    for _ in range(3):  # advance s2's state so it diverges
        client.post("/act", json={"obs": obs, "session": "s2"})
    # resetting s1 then acting reproduces the first response (zero state)
    assert client.post("/session/reset", json={"session": "s1"}).json()["reset"]
    a1b = client.post("/act", json={"obs": obs, "session": "s1"}).json()
    assert a1b["value"] == a1["value"]


def test_serve_concurrent_requests_consistent(tmp_path):
    """Threaded clients hammering /act must produce valid, complete
    responses (the server serializes RNG/session state internally)."""
    from concurrent.futures import ThreadPoolExecutor

    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    app = create_app(_serve_cfg(tmp_path, "lstm"))
    client = TestClient(app)
    D = client.get("/health").json()["obs_dim"]
    obs = np.zeros((4, D), dtype=np.float32).tolist()

    def hit(i):
        r = client.post("/act", json={"obs": obs, "session": f"s{i % 3}"})
        assert r.status_code == 200
        body = r.json()
        assert len(body["actions"]) == 4
        return body

    with ThreadPoolExecutor(8) as ex:
        results = list(ex.map(hit, range(40)))
    assert len(results) == 40


def test_serve_loads_checkpoint_from_different_n_envs(tmp_path):
    """Serving only needs model weights: a policy trained at one n_envs
    must load into a serve config with a different (smaller) template env
    (inference-only load path, utils/checkpoint.load_model_for_inference)."""
    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    cfg = _serve_cfg(tmp_path, "mlp")   # trained with n_envs=16
    cfg = {**cfg, "n_envs": 4}          # serve template is 4 envs
    app = create_app(cfg)
    client = TestClient(app)
    h = client.get("/health").json()
    obs = np.zeros((2, h["obs_dim"]), dtype=np.float32).tolist()
    r = client.post("/act", json={"obs": obs})
    assert r.status_code == 200 and len(r.json()["actions"]) == 2


def test_serve_session_batch_mismatch_409(tmp_path):
    """Continuing a recurrent session with a different batch size must be
    refused (409), never silently reset; after /session/reset the new batch
    size is accepted."""
    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    app = create_app(_serve_cfg(tmp_path, "lstm"))
    client = TestClient(app)
    D = client.get("/health").json()["obs_dim"]
    obs3 = np.zeros((3, D), dtype=np.float32).tolist()
    obs2 = np.zeros((2, D), dtype=np.float32).tolist()
    assert client.post("/act", json={"obs": obs3, "session": "x"}).status_code == 200
    r = client.post("/act", json={"obs": obs2, "session": "x"})
    assert r.status_code == 409
    client.post("/session/reset", json={"session": "x"})
    assert client.post("/act", json={"obs": obs2, "session": "x"}).status_code == 200


def test_serve_session_map_bounded_lru(tmp_path):
    """Arbitrary client session keys must not grow state unboundedly:
    the map is LRU-bounded by serve_max_sessions."""
    from gymfx_amd.serve import PolicyServer

    cfg = {**_serve_cfg(tmp_path, "lstm"), "serve_max_sessions": 4}
    srv = PolicyServer(cfg)
    obs = np.zeros((1, srv.obs_dim), dtype=np.float32)
    for i in range(10):
        srv.act(obs, session=f"k{i}")
    assert len(srv._sessions) == 4
    # most-recent keys survive
    assert set(srv._sessions) == {"k6", "k7", "k8", "k9"}
    # touching k6 then adding a new key evicts k7, not k6
    srv.act(obs, session="k6")
    srv.act(obs, session="new")
    assert "k6" in srv._sessions and "k7" not in srv._sessions


def test_serve_prometheus_metrics(tmp_path):
    """GET /metrics exposes request/action counters, the batch-latency
    histogram and the live-session gauge (production observability —
    SURVEY §5.5's 'structured logging/metrics' requirement on the
    serving side)."""
    from fastapi.testclient import TestClient

    from gymfx_amd.serve import create_app

    cfg = _serve_cfg(tmp_path, "lstm")
    client = TestClient(create_app(cfg))
    D = client.get("/health").json()["obs_dim"]
    obs = np.zeros((3, D), dtype=np.float32).tolist()
    assert client.post("/act", json={"obs": obs, "session": "m1"}).status_code == 200
    # a 409 must count as a conflict, not an ok
    assert client.post("/act", json={"obs": obs[:2], "session": "m1"}).status_code == 409
    text = client.get("/metrics").text
    assert 'gymfx_serve_requests_total{outcome="ok"} 1.0' in text
    assert 'gymfx_serve_requests_total{outcome="conflict"} 1.0' in text
    assert "gymfx_serve_actions_total 3.0" in text
    assert "gymfx_serve_batch_latency_seconds_bucket" in text
    assert "gymfx_serve_live_sessions 1.0" in text
