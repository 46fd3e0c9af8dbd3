"""bench.py driver contract: one JSON line from rank 0 with the BASELINE
metric/config keys — the round driver parses exactly this (BENCH_rNN /
SCALE_rNN)."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]


def _run(args, timeout=420):
    r = subprocess.run([sys.executable, str(REPO / "bench.py"), *args],
                       capture_output=True, text=True, timeout=timeout,
                       cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    return json.loads(lines[0])


def test_bench_json_contract_cpu():
    d = _run(["--steps", "2", "--warmup", "1", "--n-envs", "32",
              "--rollout", "8"])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config", "phases_ms"):
        assert key in d, key
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["dtype"] == "bf16" and d["data"] == "synthetic"
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism",
                "n_envs_per_gpu", "rollout_steps", "obs_dim", "preprocessor",
                "reward", "strategy", "pairs"):
        assert key in cfg, key
    assert cfg["parallelism"] == "dp1"
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_bench_lstm_variant_contract():
    d = _run(["--steps", "1", "--warmup", "0", "--n-envs", "16",
              "--rollout", "16", "--policy", "lstm"])
    assert "LSTM" in d["config"]["model"]
    assert d["config"]["reward"] == "dd_penalized_reward"
    assert d["config"]["strategy"] == "direct_atr_sltp"


def test_bench_torchrun_ws2_contract():
    """The driver's exact SCALE invocation shape (torch.distributed.run,
    nnodes=1, nproc-per-node N, 127.0.0.1 rendezvous): rank 0 prints ONE
    JSON line, parallelism reflects the world size, and the run exits 0
    on the gloo fallback here (nccl=RCCL path on a GPU box)."""
    import socket

    for attempt in range(2):
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        r = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", str(port), str(REPO / "bench.py"),
             "--gpus", "2", "--steps", "2", "--warmup", "1",
             "--n-envs", "32", "--rollout", "8"],
            capture_output=True, text=True, timeout=420, cwd=REPO)
        if r.returncode == 0:
            break
        # the freed port can be re-taken before torchrun binds it — one
        # retry on a fresh port removes that flake from the CI path
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    d = json.loads(lines[0])
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0
