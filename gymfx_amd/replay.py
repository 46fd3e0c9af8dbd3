"""Deterministic replay adapter + independent reconciliation oracle.

MI355X-native equivalent of the reference's Nautilus replay stack:
- scripted target-action replay with an ordered, immutable event log and
  sha256 event/result hashes
  (/root/reference/simulation_engines/nautilus_adapter.py:136-312, 437-449)
- an INDEPENDENT scalar ledger that recomputes the final account balance
  from fill facts and is reconciled against the engine's native state
  (/root/reference/simulation_engines/bakeoff.py:228-303)

The "engine" here is the vectorized env (torch oracle on CPU, fused HIP
kernels on GPU); the ScalarLedger below is written directly from the broker
spec (fills at next-bar open ± slippage, % commission on notional, margin,
long<->short flip = close+open with two commissions, worst-case intrabar
bracket collision: stop checked before limit, gap-through fills at open) and
shares no code with either engine path.
"""
from __future__ import annotations

import hashlib
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Sequence

import torch

from .data.feed import MarketData
from .envs.vec_env import VecFxEnv


def _hash_events(events: List[Dict[str, Any]]) -> str:
    payload = json.dumps(events, sort_keys=True, default=str).encode()
    return hashlib.sha256(payload).hexdigest()


@dataclass
class ScalarLedger:
    """Independent average-price position ledger (pure Python scalars)."""

    initial_cash: float
    commission: float
    slippage: float
    leverage: float
    position_size: float
    sl_pips: float = 0.0
    tp_pips: float = 0.0
    pip_size: float = 0.0001
    use_brackets: bool = False
    # execution-realism tier (ids == envs/params.py policy maps)
    collision_policy: int = 0     # 0 worst_case, 1 ohlc, 2 adaptive
    limit_fill_policy: int = 0    # 0 touch, 1 cross, 2 conservative
    latency_bars: int = 0
    margin_model: int = 0         # 0 leveraged, 1 standard
    margin_init_rate: float = 0.03

    cash: float = field(init=False)
    pos: float = field(init=False, default=0.0)
    avg_entry: float = field(init=False, default=0.0)
    margin: float = field(init=False, default=0.0)
    commission_paid: float = field(init=False, default=0.0)
    trade_count: int = field(init=False, default=0)
    br_sl: float = field(init=False, default=0.0)
    br_tp: float = field(init=False, default=0.0)
    br_active: bool = field(init=False, default=False)
    br_armed: bool = field(init=False, default=False)
    pend_close: bool = field(init=False, default=False)
    pend_dir: int = field(init=False, default=0)
    pend_sl: float = field(init=False, default=0.0)
    pend_tp: float = field(init=False, default=0.0)
    pend_wait: int = field(init=False, default=0)
    events: List[Dict[str, Any]] = field(init=False, default_factory=list)

    def __post_init__(self):
        self.cash = self.initial_cash

    # -- fills ----------------------------------------------------------
    def _close(self, fill: float, bar: int, kind: str) -> None:
        realized = self.pos * (fill - self.avg_entry)
        comm = abs(self.pos) * fill * self.commission
        self.cash += self.margin + realized - comm
        self.commission_paid += comm
        self.trade_count += 1
        self.events.append({
            "type": kind, "bar": bar, "side": "sell" if self.pos > 0 else "buy",
            "qty": abs(self.pos), "price": round(fill, 12),
            "realized": round(realized, 12), "commission": round(comm, 12),
        })
        self.pos = 0.0
        self.avg_entry = 0.0
        self.margin = 0.0
        self.br_active = self.br_armed = False

    def _open(self, direction: int, size: float, fill: float, bar: int) -> None:
        notional = size * fill
        comm = notional * self.commission
        # leveraged: notional / leverage; standard: init-margin fraction
        self.margin = (notional * self.margin_init_rate
                       if self.margin_model == 1 else notional / self.leverage)
        self.cash -= self.margin + comm
        self.commission_paid += comm
        self.pos = direction * size
        self.avg_entry = fill
        self.events.append({
            "type": "order_filled", "bar": bar,
            "side": "buy" if direction > 0 else "sell", "qty": size,
            "price": round(fill, 12), "commission": round(comm, 12),
        })
        if self.pend_sl > 0 or self.pend_tp > 0:
            self.br_active = self.br_armed = True
            self.br_sl, self.br_tp = self.pend_sl, self.pend_tp

    # -- one bar ---------------------------------------------------------
    def _bracket_trigger(self, o: float, h: float, low: float, c: float):
        """(trigger_price, kind) for the first bracket child hit this bar,
        honoring the collision and limit-fill policies (mirrors
        env_step.hip section 2 and reference_step.py)."""
        is_long = self.pos > 0
        sl, tp = self.br_sl, self.br_tp
        lim = self.limit_fill_policy

        def tp_hit(px: float) -> bool:
            if is_long:
                return px >= tp if lim == 0 else px > tp
            return px <= tp if lim == 0 else px < tp

        tp_gap_px = tp if lim == 2 else o  # conservative: no improvement
        sl_gap = o <= sl if is_long else o >= sl
        if self.collision_policy == 0:  # worst_case: stop absolute priority
            if sl_gap:
                return o, "bracket_sl_fill"
            if (low <= sl) if is_long else (h >= sl):
                return sl, "bracket_sl_fill"
            if tp_hit(o):
                return tp_gap_px, "bracket_tp_fill"
            if tp_hit(h if is_long else low):
                return tp, "bracket_tp_fill"
            return None, None
        # ohlc / adaptive: gap checks at the open, then extremes in order
        if sl_gap:
            return o, "bracket_sl_fill"
        if tp_hit(o):
            return tp_gap_px, "bracket_tp_fill"
        low_first = (c >= o) if self.collision_policy == 2 else False
        for at_low in ((True, False) if low_first else (False, True)):
            if at_low:
                if is_long and low <= sl:
                    return sl, "bracket_sl_fill"
                if not is_long and tp_hit(low):
                    return tp, "bracket_tp_fill"
            else:
                if is_long and tp_hit(h):
                    return tp, "bracket_tp_fill"
                if not is_long and h >= sl:
                    return sl, "bracket_sl_fill"
        return None, None

    def step(self, bar: int, o: float, h: float, low: float, c: float,
             action: int) -> None:
        slip = self.slippage
        buy = lambda px: px * (1.0 + slip)     # noqa: E731
        sell = lambda px: px * (1.0 - slip)    # noqa: E731
        # 1. pending market fills at open (latency holds them in transit)
        if self.pend_wait > 0 and (self.pend_close or self.pend_dir != 0):
            self.pend_wait -= 1
        else:
            if self.pend_close and self.pos != 0:
                fill = buy(o) if self.pos < 0 else sell(o)
                self._close(fill, bar, "order_filled")
            if self.pend_dir != 0 and self.pos == 0:
                fill = buy(o) if self.pend_dir > 0 else sell(o)
                self._open(self.pend_dir, self.position_size, fill, bar)
            self.pend_close = False
            self.pend_dir = 0
            self.pend_sl = self.pend_tp = 0.0
            self.pend_wait = 0
        # 2. bracket children per collision/limit policy
        if self.br_active and not self.br_armed and self.pos != 0:
            trig, kind = self._bracket_trigger(o, h, low, c)
            if trig is not None:
                fill = sell(trig) if self.pos > 0 else buy(trig)
                self._close(fill, bar, kind)
        self.br_armed = False
        # 3. strategy decision (direct / fixed-bracket).  Out-of-range
        # actions coerce to hold, exactly like the engine's decode
        # (env_step.hip: a = (v >= 0 && v <= 2) ? v : 0); action 3
        # (force-flat) exists only INTERNALLY via the event overlay.
        if action < 0 or action > 2:
            action = 0
        held = self.pend_close or self.pend_dir != 0  # in-transit order
        if action in (1, 2):
            self.events.append({"type": "target_requested", "bar": bar,
                                "action": action})
            want = 1 if action == 1 else -1
            if self.pos * want < 0:
                self.pend_close = True
            if self.pos * want <= 0:  # flat or opposite (flip = close+open)
                self.pend_dir = want
                if self.use_brackets:
                    sl_d = self.sl_pips * self.pip_size
                    tp_d = self.tp_pips * self.pip_size
                    self.pend_sl = c - sl_d if want > 0 else c + sl_d
                    self.pend_tp = c + tp_d if want > 0 else c - tp_d
            if not held and (self.pend_close or self.pend_dir != 0):
                self.pend_wait = self.latency_bars  # latency clock starts

    def equity(self, close: float) -> float:
        return self.cash + self.margin + self.pos * (close - self.avg_entry)


class ReplayAdapter:
    """Deterministic scripted replay through the vectorized engine,
    reconciled against the ScalarLedger oracle."""

    def run(self, config: Dict[str, Any], market_data: MarketData,
            actions: Sequence[int]) -> Dict[str, Any]:
        cfg = dict(config)
        cfg["n_envs"] = 1
        cfg.setdefault("env_start_mode", "zero")
        cfg["autoreset"] = False
        env = VecFxEnv(cfg, market_data)
        env.reset(seed=int(cfg.get("seed") or 0))

        p = env.params
        ledger = ScalarLedger(
            initial_cash=p.initial_cash, commission=p.commission,
            slippage=p.slippage, leverage=p.leverage,
            position_size=p.position_size, sl_pips=p.sl_pips,
            tp_pips=p.tp_pips, pip_size=p.pip_size,
            use_brackets=(p.strategy_id == 1),
            collision_policy=p.intrabar_collision_policy,
            limit_fill_policy=p.limit_fill_policy,
            latency_bars=p.latency_bars,
            margin_model=p.margin_model,
            margin_init_rate=p.margin_init_rate,
        )
        o = market_data.columns["OPEN"]
        h = market_data.columns["HIGH"]
        lo = market_data.columns["LOW"]
        c = market_data.columns["CLOSE"]

        device = env.device
        n_steps = 0
        for i, action in enumerate(actions):
            if bool(env.st.terminated[0].item()):
                break
            env.step(torch.tensor([action], dtype=torch.int64, device=device))
            bar = int(env.st.cursor[0].item()) - 1
            ledger.step(bar, float(o[bar]), float(h[bar]), float(lo[bar]),
                        float(c[bar]), int(action))
            n_steps += 1

        bar = int(torch.clamp(env.st.cursor[0] - 1, min=0).item())
        last_bar = bar
        engine = {
            "equity": float(env.st.equity[0].item()),
            "cash": float(env.st.cash[0].item()),
            "commission_paid": float(env.st.commission_paid[0].item()),
            "trade_count": int(env.st.trade_count[0].item()),
            "position": float(env.st.pos[0].item()),
        }
        oracle = {
            "equity": ledger.equity(float(c[bar])),
            "cash": ledger.cash,
            "commission_paid": ledger.commission_paid,
            "trade_count": ledger.trade_count,
            "position": ledger.pos,
        }
        tol = 1e-6 * max(1.0, p.initial_cash)
        recon = {
            k: bool(abs(engine[k] - oracle[k]) <= tol) for k in
            ("equity", "cash", "commission_paid", "position")
        }
        recon["trade_count"] = engine["trade_count"] == oracle["trade_count"]
        result = {
            "schema": "gymfx.replay.v1",
            "steps": n_steps,
            "last_bar": last_bar,
            "engine": engine,
            "oracle": oracle,
            "reconciled": all(recon.values()),
            "reconciliation": recon,
            "events": ledger.events,
            "event_hash": _hash_events(ledger.events),
        }
        result["result_hash"] = hashlib.sha256(
            json.dumps({k: result[k] for k in
                        ("steps", "engine", "event_hash")},
                       sort_keys=True, default=str).encode()).hexdigest()
        return result
