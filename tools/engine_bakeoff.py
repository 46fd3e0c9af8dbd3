#!/usr/bin/env python3
"""Cross-engine bakeoff: the VECTORIZED engine vs the TargetReplay engine
on one deterministic action script.

MI355X-native reincarnation of the reference's nautilus bakeoff
(/root/reference/tools/nautilus_bakeoff.py): two INDEPENDENT execution
engines — the fused-kernel/torch-oracle vectorized env (plus its
ScalarLedger reconciliation) and the Decimal target-position replay —
consume the same OHLC series and the same long/short/flat script under
matched cost assumptions, and their final accounts must agree.

Matching the assumptions: direct fixed-size actions (no brackets),
latency 0, worst_case collisions; TargetReplay's quote adversity equals
the env's slippage (full_spread_rate=0), and margin_init=1 makes the
leveraged margin model identical to the env's notional/leverage.

Exit code 0 iff all three accounts (engine, scalar ledger, target replay)
reconcile within tolerance.
"""
import argparse
import json
import sys
from decimal import Decimal
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np

from gymfx_amd.contracts import (ExecutionCostProfile, InstrumentSpec,
                                 MarketFrame, TargetAction)
from gymfx_amd.data.feed import synthetic_ohlcv
from gymfx_amd.replay import ReplayAdapter
from gymfx_amd.target_replay import TargetReplay

NS_MIN = 60 * 1_000_000_000


def run_bakeoff(rows: int = 400, steps: int = 200, seed: int = 11,
                commission: float = 2e-5, slippage: float = 1e-5,
                leverage: float = 20.0, position_size: float = 1000.0,
                initial_cash: float = 10_000.0) -> dict:
    md = synthetic_ohlcv(rows, seed=seed, vol=4e-4)
    rng = np.random.default_rng(seed)
    actions = [int(a) for a in rng.integers(0, 3, size=steps)]

    # ---- engine A: vectorized env + ScalarLedger reconciliation --------
    cfg = {
        "n_envs": 1, "device": "cpu", "window_size": 8,
        "initial_cash": initial_cash, "position_size": position_size,
        "commission": commission, "slippage": slippage,
        "leverage": leverage, "strategy_plugin": "default_strategy",
    }
    vec = ReplayAdapter().run(cfg, md, actions)

    # ---- engine B: TargetReplay over the same bars + script ------------
    profile = ExecutionCostProfile.from_dict({
        "schema_version": "execution_cost_profile.v1",
        "profile_id": "bakeoff",
        "commission_rate_per_side": str(commission),
        "full_spread_rate": "0",
        # adverse = slippage: identical fill displacement to the env
        "slippage_bps_per_side": str(slippage * 10_000.0),
        "latency_ms": 0, "financing_enabled": False,
        "intrabar_collision_policy": "worst_case",
        "limit_fill_policy": "touch",
        "margin_model": "leveraged", "enforce_margin_preflight": False,
        "random_seed": seed,
    })
    spec = InstrumentSpec(
        symbol="EUR/USD", venue="SIM", base_currency="EUR",
        quote_currency="USD", price_precision=5, size_precision=0,
        margin_init=Decimal("1"), margin_maint=Decimal("1"))
    o = md.columns["OPEN"]; h = md.columns["HIGH"]
    lo = md.columns["LOW"]; c = md.columns["CLOSE"]
    # both engines must value the final account at the SAME close: truncate
    # the frame list at the env's last published bar
    last_bar = vec["last_bar"]
    frames = [MarketFrame("EUR/USD.SIM", 1, (i + 1) * NS_MIN,
                          Decimal(repr(float(o[i]))), Decimal(repr(float(h[i]))),
                          Decimal(repr(float(lo[i]))), Decimal(repr(float(c[i]))),
                          Decimal("1000000"))
              for i in range(last_bar + 1)]
    # drive the targets from the vectorized run's own event log: each
    # target_requested (bar b, action a) becomes an absolute-target action
    # at bar b's frame — delta-to-target netting reproduces the env's
    # close+open flip as ONE equal-notional fill
    targets = []
    for k, e in enumerate(ev for ev in vec["events"]
                          if ev["type"] == "target_requested"):
        want = position_size if e["action"] == 1 else -position_size
        targets.append(TargetAction("EUR/USD.SIM", (e["bar"] + 1) * NS_MIN,
                                    Decimal(repr(want)), f"a{k}"))
    tr = TargetReplay(profile).run(
        instrument_specs=[spec], frames=frames, actions=targets,
        initial_cash=Decimal(repr(initial_cash)),
        default_leverage=Decimal(repr(leverage)))

    eq_vec = vec["engine"]["equity"]
    eq_led = vec["oracle"]["equity"]
    eq_tr = float(Decimal(tr["final_equity"]))
    tol = 1e-6 * initial_cash
    out = {
        "schema": "gymfx.bakeoff.v1",
        "steps": vec["steps"],
        "vector_engine_equity": eq_vec,
        "scalar_ledger_equity": eq_led,
        "target_replay_equity": eq_tr,
        "vec_vs_ledger_ok": bool(vec["reconciled"]),
        "vec_vs_target_replay_ok": bool(abs(eq_vec - eq_tr) <= tol),
        "event_hash_vec": vec["event_hash"],
        "event_hash_target_replay": tr["event_hash"],
    }
    out["ok"] = out["vec_vs_ledger_ok"] and out["vec_vs_target_replay_ok"]
    return out


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("--rows", type=int, default=400)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--seed", type=int, default=11)
    args = ap.parse_args()
    out = run_bakeoff(rows=args.rows, steps=args.steps, seed=args.seed)
    print(json.dumps(out, indent=2))
    return 0 if out["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
