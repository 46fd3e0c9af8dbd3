#!/usr/bin/env python3
"""Serving throughput: batched actions/sec through PolicyServer.act()
(in-process — measures the policy forward + sample path without HTTP
overhead; run under gpurun for the MI355X number)."""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--policy", default="mlp", choices=["mlp", "lstm"])
    ap.add_argument("--device", default=None)
    ap.add_argument("--checkpoint", default=None)
    args = ap.parse_args()

    import torch

    from gymfx_amd.config import DEFAULT_VALUES
    from gymfx_amd.serve import PolicyServer

    cfg = {**DEFAULT_VALUES,
           "data_feed_plugin": "synthetic_data_feed",
           "synthetic_rows": 20000, "synthetic_extra_features": 3,
           "preprocessor_plugin": "feature_window_preprocessor",
           "feature_columns": ["OPEN", "HIGH", "LOW", "CLOSE", "VOLUME",
                               "FEAT_0", "FEAT_1", "FEAT_2"],
           "n_envs": 16, "window_size": 32, "seed": 0,
           "policy_model": args.policy,
           "checkpoint_file": args.checkpoint}
    if args.device:
        cfg["device"] = args.device
    srv = PolicyServer(cfg)
    rng = np.random.default_rng(0)
    obs = rng.normal(size=(args.batch, srv.obs_dim)).astype(np.float32)
    for _ in range(args.warmup):
        srv.act(obs, greedy=True, session="bench")
    if srv.device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        srv.act(obs, greedy=True, session="bench")
    if srv.device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "schema": "serve.benchmark.v1",
        "policy": args.policy,
        "device": str(srv.device),
        "batch": args.batch,
        "iters": args.iters,
        "actions_per_sec": args.batch * args.iters / dt,
        "ms_per_batch": dt / args.iters * 1e3,
    }))


if __name__ == "__main__":
    main()
