"""RCCL-on-silicon evidence.

Round 1 verified the data-parallel path only under gloo on CPU; these tests
put the nccl(=RCCL) backend on a real MI355X: communicator init, collective
correctness, the measured latency of the flat 0.4 MB gradient bucket
(the quantity the overlap design in algo/ppo.py banks on), and — when RCCL
allows two ranks on one device — the actual GradAllReducer pipeline.
"""
import json
import os
import subprocess
import sys
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _init_ws1(port: int):
    import torch.distributed as dist

    if dist.is_initialized():
        dist.destroy_process_group()
    dist.init_process_group(
        backend="nccl", init_method=f"tcp://127.0.0.1:{port}",
        rank=0, world_size=1)
    return dist


def test_rccl_init_and_collectives_ws1():
    """World-size-1 RCCL: communicator init + all_reduce/broadcast kernels
    execute on hardware and preserve values (identity at ws=1)."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    dist = _init_ws1(29611)
    try:
        dev = torch.device("cuda:0")
        t = torch.arange(1024, dtype=torch.float32, device=dev)
        ref = t.clone()
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        torch.cuda.synchronize()
        assert torch.equal(t, ref)
        dist.broadcast(t, src=0)
        dist.barrier()
        torch.cuda.synchronize()
        assert torch.equal(t, ref)
    finally:
        dist.destroy_process_group()


def test_rccl_allreduce_latency_microbench():
    """Latency of the exact gradient-bucket all-reduce shape bench.py
    issues per minibatch (~0.43 MB fp32).  ws=1 measures the RCCL
    launch+kernel floor — the xGMI wire time at ws>1 adds to this.
    Writes gpurun_out/allreduce_lat.json for the profiles record."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    dist = _init_ws1(29612)
    try:
        dev = torch.device("cuda:0")
        sizes = {"mlp_bucket_107k": 107_000, "lstm_bucket_330k": 330_000,
                 "1M": 1_000_000, "16M": 16_000_000}
        out = {}
        for name, n in sizes.items():
            g = torch.randn(n, device=dev)
            for _ in range(10):
                dist.all_reduce(g)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            iters = 100
            for _ in range(iters):
                dist.all_reduce(g)
            torch.cuda.synchronize()
            us = (time.perf_counter() - t0) / iters * 1e6
            out[name] = round(us, 2)
        os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
        with open(os.path.join(REPO, "gpurun_out", "allreduce_lat.json"), "w") as f:
            json.dump({"world_size": 1, "unit": "us/all_reduce", **out}, f)
        # sanity: the bucket collective must be far below one minibatch's
        # fwd/bwd (~1.2 ms) or the overlap design is moot
        assert out["mlp_bucket_107k"] < 1000.0, out
    finally:
        dist.destroy_process_group()


def test_gradallreducer_ws2_shared_device():
    """Two ranks over RCCL exercising GradAllReducer end-to-end.  On a
    1-GPU box both ranks share cuda:0; RCCL may refuse (exit 77 -> skip) —
    on a multi-GPU node this runs for real."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613", os.path.join(REPO, "tests", "_nccl_ws2_probe.py")],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    if r.returncode != 0:
        combined = r.stdout + r.stderr
        if "RCCL refuses shared device" in combined or "exitcode  : 77" in combined.replace("  ", " "):
            pytest.skip("RCCL refuses 2 ranks on one device (expected on 1-GPU box)")
        raise AssertionError(f"ws2 probe failed:\n{combined[-3000:]}")
    assert r.stdout.count("WS2_PROBE_OK") == 2, r.stdout[-2000:]
