"""CPU smoke for the evidence-producing CLI tools: the soak/serve/engine
benchmarks must run end-to-end at tiny sizes (they produce the tracked
profiles/ artifacts; a silent break would rot the evidence chain)."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]


def _run(args, timeout=420):
    r = subprocess.run([sys.executable, *args], capture_output=True,
                       text=True, timeout=timeout, cwd=REPO)
    assert r.returncode == 0, (r.stdout + r.stderr)[-2000:]
    return r.stdout


def test_train_demo_cli(tmp_path):
    out = tmp_path / "demo.json"
    _run(["tools/train_demo.py", "--updates", "2", "--n-envs", "8",
          "--out", str(out)])
    d = json.loads(out.read_text())
    assert d["updates"] == 2 and d["all_finite"] is True
    assert "reward_curve" in d and "entropy_curve" in d


def test_serve_benchmark_cli():
    out = _run(["tools/serve_benchmark.py", "--batch", "16", "--iters", "3",
                "--warmup", "1"])
    line = [l for l in out.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["schema"] == "serve.benchmark.v1" and d["actions_per_sec"] > 0


def test_engine_benchmark_cli():
    out = _run(["tools/engine_benchmark.py", "--steps", "20",
                "--runs", "1"])
    d = json.loads(out[out.index("{"):])
    assert d["schema"] == "simulation_engine_benchmark.v1"
    assert d["max_rss_mb"] > 0
