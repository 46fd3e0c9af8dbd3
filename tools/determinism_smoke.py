#!/usr/bin/env python3
"""Determinism evidence (parity: /root/reference/tools/nautilus_bakeoff.py
repeat-run hash equality + tools/nautilus_parallel_smoke.py cross-process
spawn-pool hash equality).  Emits a JSON verdict."""
import json
import multiprocessing as mp
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def _one_run(_=None):
    import numpy as np
    from gymfx_amd.data.feed import synthetic_ohlcv
    from gymfx_amd.replay import ReplayAdapter

    md = synthetic_ohlcv(400, seed=9, vol=4e-4)
    rng = np.random.default_rng(3)
    actions = rng.integers(0, 3, size=300).tolist()
    cfg = {"window_size": 8, "initial_cash": 10000.0, "position_size": 1000.0,
           "commission": 2e-5, "slippage": 5e-6, "device": "cpu", "seed": 0,
           "strategy_plugin": "direct_fixed_sltp", "sl_pips": 8.0,
           "tp_pips": 16.0}
    return ReplayAdapter().run(cfg, md, actions)


def main():
    r1 = _one_run()
    r2 = _one_run()
    ctx = mp.get_context("spawn")
    with ctx.Pool(2) as pool:
        cross = pool.map(_one_run, [0, 1])
    verdict = {
        "schema": "gymfx.determinism_smoke.v1",
        "in_process_repeat_identical": r1["result_hash"] == r2["result_hash"],
        "cross_process_identical": (
            cross[0]["result_hash"] == cross[1]["result_hash"]
            == r1["result_hash"]
        ),
        "reconciled_vs_oracle": r1["reconciled"],
        "event_hash": r1["event_hash"],
        "result_hash": r1["result_hash"],
        "events": len(r1["events"]),
    }
    verdict["ok"] = all(v for k, v in verdict.items()
                        if k.endswith("identical") or k == "reconciled_vs_oracle")
    print(json.dumps(verdict, indent=2))
    sys.exit(0 if verdict["ok"] else 1)


if __name__ == "__main__":
    main()
