#!/usr/bin/env python3
"""Run a deterministic target-position replay from JSON inputs.

CLI face of gymfx_amd/target_replay.py (the reference exposes its
Nautilus bakeoff through scripts; this is the equivalent entry for the
MI355X-native engine).  Input file schema (see examples/replay/):

{
  "profile": { ... execution_cost_profile.v1 ... } | "path/to/profile.json",
  "instruments": [{"symbol": "EUR/USD", "venue": "SIM",
                   "base_currency": "EUR", "quote_currency": "USD",
                   "price_precision": 5, "size_precision": 0,
                   "margin_init": "0.03", "margin_maint": "0.03"}],
  "frames": [{"instrument_id": "EUR/USD.SIM", "timeframe_minutes": 1,
              "ts_event_ns": 1704153660000000000, "open": "1.10000",
              "high": "...", "low": "...", "close": "...",
              "volume": "1000000",
              "execution_path": ["1.10000", "1.09700", ...]  (optional)}],
  "actions": [{"instrument_id": "EUR/USD.SIM",
               "ts_event_ns": 1704153660000000000,
               "target_units": "1000", "action_id": "a1",
               "stop_loss_price": "1.09800" (optional),
               "take_profit_price": "1.10200" (optional)}],
  "initial_cash": "100000", "base_currency": "USD",
  "default_leverage": "20",
  "financing_rate_data": [{"LOCATION": "EA19", "TIME": "2024-01",
                           "Value": 5.0}]  (when financing_enabled)
}

Outputs the replay result JSON (events, hashes, final balances) to stdout
or --out; --reports additionally writes the canonical execution reports
(gymfx.execution_report.v1).
"""
import argparse
import json
import sys
from decimal import Decimal
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from gymfx_amd.contracts import (ExecutionCostProfile, InstrumentSpec,
                                 MarketFrame, TargetAction)
from gymfx_amd.target_replay import TargetReplay, export_execution_reports


def _dec(v):
    return Decimal(str(v))


def load_inputs(raw):
    prof_raw = raw["profile"]
    if isinstance(prof_raw, str):
        with open(prof_raw, "r", encoding="utf-8") as fh:
            prof_raw = json.load(fh)
    profile = ExecutionCostProfile.from_dict(prof_raw)
    specs = [InstrumentSpec(
        symbol=i["symbol"], venue=i["venue"],
        base_currency=i["base_currency"], quote_currency=i["quote_currency"],
        price_precision=int(i["price_precision"]),
        size_precision=int(i["size_precision"]),
        margin_init=_dec(i["margin_init"]), margin_maint=_dec(i["margin_maint"]),
        min_quantity=_dec(i.get("min_quantity", "1")),
        lot_size=_dec(i["lot_size"]) if i.get("lot_size") else None,
    ) for i in raw["instruments"]]
    frames = [MarketFrame(
        instrument_id=f["instrument_id"],
        timeframe_minutes=int(f["timeframe_minutes"]),
        ts_event_ns=int(f["ts_event_ns"]),
        open=_dec(f["open"]), high=_dec(f["high"]), low=_dec(f["low"]),
        close=_dec(f["close"]), volume=_dec(f["volume"]),
        execution_path=tuple(_dec(p) for p in f["execution_path"])
        if f.get("execution_path") else None,
    ) for f in raw["frames"]]
    actions = [TargetAction(
        instrument_id=a["instrument_id"], ts_event_ns=int(a["ts_event_ns"]),
        target_units=_dec(a["target_units"]), action_id=a["action_id"],
        stop_loss_price=_dec(a["stop_loss_price"])
        if a.get("stop_loss_price") else None,
        take_profit_price=_dec(a["take_profit_price"])
        if a.get("take_profit_price") else None,
    ) for a in raw["actions"]]
    return profile, specs, frames, actions


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("input", help="replay input JSON (see docstring)")
    ap.add_argument("--out", default="-", help="result JSON path (- = stdout)")
    ap.add_argument("--reports", default=None,
                    help="also write canonical execution reports here")
    args = ap.parse_args()
    with open(args.input, "r", encoding="utf-8") as fh:
        raw = json.load(fh)
    profile, specs, frames, actions = load_inputs(raw)
    rates = raw.get("financing_rate_data")
    result = TargetReplay(profile).run(
        instrument_specs=specs, frames=frames, actions=actions,
        initial_cash=_dec(raw.get("initial_cash", "100000")),
        base_currency=str(raw.get("base_currency", "USD")),
        default_leverage=_dec(raw.get("default_leverage", "20")),
        financing_rate_data=rates)
    text = json.dumps(result, indent=2, default=str)
    if args.out == "-":
        print(text)
    else:
        Path(args.out).write_text(text)
    if args.reports:
        reports = export_execution_reports(
            result, specs, profile,
            base_currency=str(raw.get("base_currency", "USD")))
        Path(args.reports).write_text(json.dumps(reports, indent=2))
    return 0


if __name__ == "__main__":
    sys.exit(main())
