"""Target-position replay engine: intrabar execution paths, latency,
seeded fill model, maker/taker fees, margin models, execution reports.

MI355X-native counterpart of the reference's Nautilus-backed replay
(/root/reference/simulation_engines/nautilus_adapter.py:314-449): instead
of embedding a third-party backtest engine, the same observable semantics
are implemented directly over the engine-neutral contracts
(gymfx_amd/contracts.py), so the replay is dependency-free, deterministic
and auditable:

- **Quote synthesis** — each MarketFrame contributes its ``execution_path``
  (or just the close) as ordered intrabar mid points; bid/ask are displaced
  from mid by ``profile.quote_adverse_rate_per_side``
  (ref nautilus_adapter.py:86-133).
- **Target actions** — ``TargetAction.target_units`` is a delta-to-target
  request (ref nautilus_gym.py:118-127): the engine computes the delta
  against the current net position and submits one market order, with
  optional stop-loss / take-profit children.
- **Latency** — an order submitted in reaction to the frame at ``ts``
  becomes executable at ``ts + latency_ms``; it fills at the first path
  point of the first LATER frame past that time
  (LatencyModel semantics, ref nautilus_adapter.py:415-417).
- **Fill model** — seeded, deterministic-per-order probabilistic fills:
  ``prob_fill_on_limit`` / ``prob_fill_on_stop`` / ``prob_slippage``
  (defaults 1/1/0 — byte-identical runs for the reference's default
  ``FillModel(random_seed=...)``; ref nautilus_adapter.py:413).
- **Collision + limit policies** — bracket children are evaluated against
  the intrabar path IN PATH ORDER, which is what resolves same-bar SL/TP
  collisions when a path is given; without a path the
  ``intrabar_collision_policy`` decides the synthetic point order
  (worst_case: entry->SL->TP; ohlc: O,H,L,C; adaptive: O,L,H,C when the
  bar closes down else O,H,L,C).  ``limit_fill_policy`` decides when the
  take-profit limit leg fills: touch (>=, price improvement on gaps),
  cross (strict >, improvement on gaps), conservative (strict >, fills at
  the limit price — no improvement).  The reference validates these
  policies (contracts.py:100-103) but delegates semantics to Nautilus;
  here they are explicit and tested.
- **Fees** — maker rate for limit (take-profit) fills, taker rate for
  market/stop fills (MakerTakerFeeModel parity, ref
  nautilus_adapter.py:414).
- **Margin** — ``standard``: init margin = notional x spec.margin_init;
  ``leveraged``: the same divided by account leverage (Standard- vs
  LeveragedMarginModel, ref nautilus_adapter.py:371-375); optional
  preflight denies orders whose opening margin exceeds free balance
  (order_denied event, ref nautilus_gym.py:129-143).
- **Financing** — FX rollover interest at 22:00 UTC via the OANDA
  calendar's schedule (calendar.compute_rollover_schedule).

The result carries an ordered immutable event log with sha256 event/result
hashes; :func:`export_execution_reports` serializes fills as canonical
engine-neutral reports (field parity with the reference's
trading-contracts export, ref bakeoff.py:306-374).
"""
from __future__ import annotations

import hashlib
import json
from dataclasses import dataclass, field
from decimal import Decimal
from typing import Any, Dict, List, Optional, Sequence, Tuple

from .contracts import (ExecutionCostProfile, InstrumentSpec, MarketFrame,
                        TargetAction)

SCHEMA = "gymfx.target_replay.v1"
REPORT_SCHEMA = "gymfx.execution_report.v1"
ENGINE_VERSION = "2.0"


# ---------------------------------------------------------------------------
# seeded fill model (counter-based: no hidden RNG state, replay-stable)
# ---------------------------------------------------------------------------

def _splitmix64(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    return z ^ (z >> 31)


@dataclass(frozen=True)
class FillModel:
    """Deterministic seeded fill probabilities (Nautilus FillModel parity).

    Defaults replicate ``FillModel(random_seed=seed)``: limit/stop orders
    always fill, no extra slippage — the stream only matters when a
    probability is set below 1 (or above 0 for slippage).
    """

    random_seed: int = 0
    prob_fill_on_limit: float = 1.0
    prob_fill_on_stop: float = 1.0
    prob_slippage: float = 0.0

    def _u01(self, order_seq: int, channel: int) -> float:
        h = _splitmix64((self.random_seed << 20) ^ (order_seq << 2) ^ channel)
        return (h >> 11) / float(1 << 53)

    def limit_fills(self, order_seq: int) -> bool:
        return self._u01(order_seq, 0) < self.prob_fill_on_limit

    def stop_fills(self, order_seq: int) -> bool:
        return self._u01(order_seq, 1) < self.prob_fill_on_stop

    def slips(self, order_seq: int) -> bool:
        return self._u01(order_seq, 2) < self.prob_slippage


# ---------------------------------------------------------------------------
# internal account state (per run; netting OMS, one margin account)
# ---------------------------------------------------------------------------

@dataclass
class _Position:
    units: Decimal = Decimal(0)
    avg_price: Decimal = Decimal(0)
    margin: Decimal = Decimal(0)
    sl: Optional[Decimal] = None
    tp: Optional[Decimal] = None
    action_id: str = ""
    armed_ts: int = -1  # frame ts when brackets were attached (skip that
                        # frame's full-path re-check; in-frame remainder is
                        # checked at arming time)


@dataclass
class _PendingOrder:
    instrument_id: str
    delta: Decimal            # signed units
    action_id: str
    submitted_ns: int
    executable_ns: int
    sl: Optional[Decimal] = None
    tp: Optional[Decimal] = None


def _conversion_rate(spec: InstrumentSpec, mid: Decimal, base_ccy: str) -> Decimal:
    if spec.quote_currency == base_ccy:
        return Decimal(1)
    if spec.base_currency == base_ccy:
        return Decimal(1) / mid
    raise ValueError(
        f"cannot convert {spec.quote_currency} to {base_ccy} via {spec.instrument_id}")


class TargetReplay:
    """Deterministic scripted target-position replay (netting, margin)."""

    def __init__(self, profile: ExecutionCostProfile,
                 fill_model: Optional[FillModel] = None):
        self.profile = profile
        self.fill = fill_model or FillModel(random_seed=profile.random_seed)

    # -- helpers --------------------------------------------------------
    def _points(self, frame: MarketFrame, long: bool) -> Tuple[Decimal, ...]:
        """Ordered intrabar mid points for bracket evaluation: the explicit
        execution_path when given, else a synthetic ordering chosen by the
        collision policy (worst_case puts the position's ADVERSE extreme
        first, so the stop always wins a same-bar SL/TP collision)."""
        if frame.execution_path:
            return tuple(frame.execution_path)
        o, h, l, c = frame.open, frame.high, frame.low, frame.close
        pol = self.profile.intrabar_collision_policy
        if pol == "ohlc":
            return (o, h, l, c)
        if pol == "adaptive":
            # down bar -> assume the high printed before the low
            return (o, h, l, c) if c < o else (o, l, h, c)
        # worst_case
        return (o, l, h, c) if long else (o, h, l, c)

    def run(self, *, instrument_specs: Sequence[InstrumentSpec],
            frames: Sequence[MarketFrame],
            actions: Sequence[TargetAction],
            initial_cash: Decimal = Decimal("100000"),
            base_currency: str = "USD",
            default_leverage: Decimal = Decimal("20"),
            financing_rate_data: Any = None) -> Dict[str, Any]:
        profile = self.profile
        if profile.financing_enabled and financing_rate_data is None:
            raise ValueError(
                "financing_rate_data is required when financing_enabled is true")
        specs = {s.instrument_id: s for s in instrument_specs}
        adverse = profile.quote_adverse_rate_per_side
        lat_ns = profile.latency_ms * 1_000_000
        maker = taker = profile.commission_rate_per_side

        frames_sorted = sorted(frames, key=lambda f: (f.ts_event_ns, f.instrument_id))
        frame_keys = {(f.ts_event_ns, f.instrument_id) for f in frames_sorted}
        by_ts: Dict[int, List[TargetAction]] = {}
        unmatched = 0
        for a in actions:
            if a.instrument_id not in specs:
                raise ValueError(f"action for unknown instrument {a.instrument_id}")
            if (a.ts_event_ns, a.instrument_id) not in frame_keys:
                unmatched += 1  # no frame at that timestamp: never applied
                continue
            by_ts.setdefault(a.ts_event_ns, []).append(a)

        cash = Decimal(initial_cash)
        positions: Dict[str, _Position] = {}
        pending: List[_PendingOrder] = []
        events: List[Dict[str, Any]] = []
        last_mid: Dict[str, Decimal] = {}
        seq = 0
        order_seq = 0

        def emit(ev: Dict[str, Any]) -> None:
            nonlocal seq
            events.append({"sequence": seq, **ev})
            seq += 1

        def margin_for(spec: InstrumentSpec, units: Decimal, mid: Decimal) -> Decimal:
            notional = abs(units) * mid
            m = notional * spec.margin_init
            if profile.margin_model == "leveraged":
                m = m / default_leverage
            return m

        def apply_fill(spec: InstrumentSpec, signed: Decimal, px: Decimal,
                       mid: Decimal, ts: int, action_id: str, liquidity: str,
                       kind: str) -> None:
            """Netting fill: realize pnl on the closing part, re-margin."""
            nonlocal cash, order_seq
            pos = positions.setdefault(spec.instrument_id, _Position())
            conv = _conversion_rate(spec, mid, base_currency)
            qty = abs(signed)
            rate = maker if liquidity == "maker" else taker
            commission = qty * px * rate * conv
            cash -= commission
            cur = pos.units
            if cur != 0 and cur * signed < 0:
                closing = min(abs(cur), qty)
                pnl = closing * (px - pos.avg_price) * (1 if cur > 0 else -1)
                cash += pnl * conv
                new_units = cur + signed
                if new_units == 0:
                    pos.avg_price = Decimal(0)
                    pos.sl = pos.tp = None
                elif cur * new_units < 0:  # flipped through flat
                    pos.avg_price = px
            else:
                new_units = cur + signed
                if cur == 0:
                    pos.avg_price = px
                elif new_units != 0:
                    pos.avg_price = ((abs(cur) * pos.avg_price + qty * px)
                                     / abs(new_units))
            pos.units = new_units
            # margin account: release old requirement, hold the new one
            cash += pos.margin
            pos.margin = margin_for(spec, new_units, mid)
            cash -= pos.margin
            order_seq += 1
            emit({
                "event_type": "order_filled",
                "ts_event_ns": ts,
                "instrument_id": spec.instrument_id,
                "client_order_id": f"O-{order_seq}",
                "side": "BUY" if signed > 0 else "SELL",
                "quantity": str(qty),
                "price": str(px),
                "commission": str(commission),
                "liquidity": liquidity,
                "kind": kind,
                "action_id": action_id,
                "reference_mid": str(mid),
            })

        def check_brackets(spec: InstrumentSpec, mids: Sequence[Decimal],
                           ts: int, gap_open: bool) -> None:
            """Walk the intrabar mid points in order; the first triggered
            child closes the position (path order IS the collision rule).
            The stop is evaluated before the limit AT each point."""
            pos = positions.get(spec.instrument_id)
            if pos is None or pos.units == 0 or (pos.sl is None and pos.tp is None):
                return
            pol = self.profile.limit_fill_policy
            long = pos.units > 0
            for i, mid in enumerate(mids):
                pos = positions.get(spec.instrument_id)
                if pos is None or pos.units == 0:
                    return
                # exit side: long closes by selling at bid, short by buying at ask
                quote = mid * (1 - adverse) if long else mid * (1 + adverse)
                gap = gap_open and i == 0
                if pos.sl is not None:
                    hit = quote <= pos.sl if long else quote >= pos.sl
                    if hit and self.fill.stop_fills(order_seq + 1):
                        # stop-market: fills at the triggering quote
                        apply_fill(spec, -pos.units, quote, mid, ts,
                                   pos.action_id, "taker", "bracket_sl_fill")
                        continue
                if pos.tp is not None:
                    if pol == "touch":
                        hit = quote >= pos.tp if long else quote <= pos.tp
                    else:  # cross / conservative require a strict cross
                        hit = quote > pos.tp if long else quote < pos.tp
                    if hit and self.fill.limit_fills(order_seq + 1):
                        # a gap past the level fills at the (better) opening
                        # quote; conservative never grants price improvement
                        px = quote if (gap and pol != "conservative") else pos.tp
                        apply_fill(spec, -pos.units, px, mid, ts,
                                   pos.action_id, "maker", "bracket_tp_fill")

        # -- financing schedule (per EUR_USD-style instrument) -----------
        roll: Dict[str, Any] = {}
        if profile.financing_enabled:
            from .calendar import compute_rollover_schedule

            ts_secs = [f.ts_event_ns // 1_000_000_000 for f in frames_sorted]
            for iid in specs:
                sym = iid.split(".")[0].replace("/", "_")
                sched = compute_rollover_schedule(ts_secs, sym, financing_rate_data)
                roll[iid] = {frames_sorted[i].ts_event_ns: Decimal(str(float(sched[i])))
                             for i in range(len(frames_sorted))
                             if frames_sorted[i].instrument_id == iid and sched[i] != 0.0}

        # -- main loop: frames in time order ------------------------------
        for frame in frames_sorted:
            spec = specs[frame.instrument_id]
            ts = frame.ts_event_ns
            path0 = (frame.execution_path[0] if frame.execution_path
                     else frame.open)

            # 1. pending orders whose latency has elapsed fill at this
            #    frame's first path point (a market order queued last bar
            #    executes at the first quote, ahead of any bracket trigger
            #    later in the path)
            still: List[_PendingOrder] = []
            for po in pending:
                if po.instrument_id != frame.instrument_id or ts <= po.executable_ns:
                    still.append(po)
                    continue
                mid = path0
                buy = po.delta > 0
                px = mid * (1 + adverse) if buy else mid * (1 - adverse)
                if self.fill.slips(order_seq + 1):
                    slip = mid * profile.slippage_rate_per_side
                    px = px + slip if buy else px - slip
                # margin preflight on the OPENING part of the delta
                if profile.enforce_margin_preflight:
                    pos = positions.get(po.instrument_id, _Position())
                    cur = pos.units
                    opening = Decimal(0)
                    if cur == 0 or cur * po.delta > 0:
                        opening = abs(po.delta)
                    elif abs(po.delta) > abs(cur):
                        opening = abs(po.delta) - abs(cur)
                    if opening > 0 and margin_for(spec, opening, mid) > cash:
                        emit({
                            "event_type": "order_denied",
                            "ts_event_ns": ts,
                            "instrument_id": po.instrument_id,
                            "client_order_id": f"O-{order_seq + 1}",
                            "action_id": po.action_id,
                            "reason": ("insufficient free balance for init "
                                       f"margin of {opening} units"),
                        })
                        continue
                apply_fill(spec, po.delta, px, mid, ts, po.action_id,
                           "taker", "order_filled")
                pos = positions[po.instrument_id]
                if pos.units != 0 and (po.sl is not None or po.tp is not None):
                    pos.sl, pos.tp = po.sl, po.tp
                    pos.action_id = po.action_id
                    pos.armed_ts = ts
                    # children armed AFTER the entry point: remaining path
                    check_brackets(
                        spec, self._points(frame, pos.units > 0)[1:],
                        ts, gap_open=False)
            pending = still

            # 2. bracket children of PREVIOUSLY armed positions over the
            #    full intrabar path (first point = gap check); positions
            #    armed THIS frame had their remaining path checked above
            pos0 = positions.get(frame.instrument_id)
            if pos0 is not None and pos0.units != 0 and pos0.armed_ts < ts:
                check_brackets(spec, self._points(frame, pos0.units > 0),
                               ts, gap_open=True)

            # 3. financing at rollover frames
            if profile.financing_enabled:
                rate = roll.get(frame.instrument_id, {}).get(ts)
                pos = positions.get(frame.instrument_id)
                if rate and pos is not None and pos.units != 0:
                    conv = _conversion_rate(spec, frame.close, base_currency)
                    interest = pos.units * frame.close * rate * conv
                    cash += interest
                    emit({
                        "event_type": "financing",
                        "ts_event_ns": ts,
                        "instrument_id": frame.instrument_id,
                        "amount": str(interest),
                    })

            last_mid[frame.instrument_id] = frame.close

            # 4. actions reacting to this frame -> pending orders
            for act in by_ts.get(ts, []):
                if act.instrument_id != frame.instrument_id:
                    continue
                pos = positions.get(act.instrument_id, _Position())
                delta = act.target_units - pos.units
                emit({
                    "event_type": "target_requested",
                    "ts_event_ns": ts,
                    "instrument_id": act.instrument_id,
                    "action_id": act.action_id,
                    "target_units": str(act.target_units),
                    "delta_units": str(delta),
                })
                if delta != 0:
                    pending.append(_PendingOrder(
                        instrument_id=act.instrument_id, delta=delta,
                        action_id=act.action_id, submitted_ns=ts,
                        executable_ns=ts + lat_ns,
                        sl=act.stop_loss_price, tp=act.take_profit_price))

        # -- result -------------------------------------------------------
        unrealized = Decimal(0)
        margin_held = Decimal(0)
        final_positions = {}
        for iid, pos in positions.items():
            mid = last_mid.get(iid, pos.avg_price)
            conv = _conversion_rate(specs[iid], mid, base_currency)
            unrealized += pos.units * (mid - pos.avg_price) * conv
            margin_held += pos.margin
            final_positions[iid] = {"units": str(pos.units),
                                    "avg_price": str(pos.avg_price)}
        result: Dict[str, Any] = {
            "schema": SCHEMA,
            "engine": "gymfx_target_replay",
            "engine_version": ENGINE_VERSION,
            "profile_id": profile.profile_id,
            "initial_cash": str(initial_cash),
            "base_currency": base_currency,
            "final_balance": str(cash + margin_held),
            "final_equity": str(cash + margin_held + unrealized),
            "positions": final_positions,
            "unmatched_actions": unmatched,
            "events": events,
        }
        result["event_hash"] = hashlib.sha256(
            json.dumps(events, sort_keys=True, default=str).encode()).hexdigest()
        result["result_hash"] = hashlib.sha256(json.dumps(
            {k: result[k] for k in ("final_balance", "final_equity",
                                    "positions", "event_hash")},
            sort_keys=True, default=str).encode()).hexdigest()
        return result


# ---------------------------------------------------------------------------
# canonical execution-report export (ref bakeoff.py:306-374 field parity)
# ---------------------------------------------------------------------------

def export_execution_reports(result: Dict[str, Any],
                             instrument_specs: Sequence[InstrumentSpec],
                             profile: ExecutionCostProfile,
                             *, base_currency: str = "USD") -> List[Dict[str, Any]]:
    """Serialize fill facts as engine-neutral execution reports.

    Same field set as the reference's trading-contracts ExecutionReport
    export; emitted as plain dicts (schema gymfx.execution_report.v1) so
    downstream consumers don't need the trading-contracts package.
    """
    from datetime import datetime, timezone

    specs = {s.instrument_id: s for s in instrument_specs}
    requested = {
        ev["action_id"]: abs(Decimal(ev["delta_units"]))
        for ev in result["events"] if ev["event_type"] == "target_requested"
    }
    reports: List[Dict[str, Any]] = []
    for ev in result["events"]:
        if ev["event_type"] != "order_filled":
            continue
        spec = specs[ev["instrument_id"]]
        mid = Decimal(ev["reference_mid"])
        conv = _conversion_rate(spec, mid, base_currency)
        qty = Decimal(ev["quantity"])
        signed = qty if ev["side"] == "BUY" else -qty
        reports.append({
            "schema": REPORT_SCHEMA,
            "object_id": f"gymfx-fill:{ev['client_order_id']}:{ev['sequence']}",
            "as_of": datetime.fromtimestamp(
                ev["ts_event_ns"] / 1_000_000_000, tz=timezone.utc).isoformat(),
            "producer": {"name": "gymfx-amd-target-replay",
                         "version": ENGINE_VERSION},
            "trace_id": result["result_hash"],
            "order_intent_id": ev["action_id"],
            "state": "filled",
            "requested_units": float(requested.get(ev["action_id"], qty)),
            "filled_units": float(signed),
            "requested_price": float(mid),
            "filled_price": float(Decimal(ev["price"])),
            "spread_cost": float(qty * mid * profile.full_spread_rate
                                 / Decimal(2) * conv),
            "slippage_cost": float(qty * mid * profile.slippage_rate_per_side
                                   * conv),
            "commission": float(Decimal(ev["commission"])),
            "financing": 0.0,
            "conversion_cost": 0.0,
            "broker_ids": {"client_order_id": ev["client_order_id"],
                           "instrument_id": ev["instrument_id"],
                           "cost_currency": base_currency},
            "latency_ms": float(profile.latency_ms),
        })
    return reports
