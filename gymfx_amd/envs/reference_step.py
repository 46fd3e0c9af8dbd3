"""Pure-torch vectorized env step — the semantic oracle.

Implements, N-wide and branch-free, the per-bar semantics of the reference
step path (/root/reference/app/env.py:279-328 + app/bt_bridge.py:136-248):

  advance bar -> fill pending market orders at open (slippage, % commission
  on notional, margin ledger) -> resolve bracket SL/TP children (worst-case
  intrabar ordering) -> update ATR -> strategy decision (direct /
  fixed-SLTP / ATR-SLTP) -> publish equity -> reward -> observation.

Known, documented divergences from the reference engine:
  * Data exhaustion terminates on the step that publishes the last bar
    (the reference takes one extra stale step, env.py:169-172 stop()).
  * The ATR true-range buffer updates on every visited bar, including bars
    where the overlay forced action 3 (the reference skips the plugin, and
    therefore the TR append, on those bars — bt_bridge.py:178-188).
  * Trade win/loss classification uses gross realized pnl; avg_trade_pnl is
    net of the closing commission.

This module is also the numerics oracle for the fused HIP env_step kernel
(ops/csrc/env_step.hip): GPU tests assert kernel == this, elementwise.
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from .market import MarketTensors
from .params import (
    EnvParams,
    PREP_FEATURE_WINDOW,
    REWARD_DD,
    REWARD_SHARPE,
    RISK_FIXED_ATR,
    RISK_MARGIN_AWARE,
    SIZE_MODE_NOTIONAL,
    STRATEGY_ATR_SLTP,
    STRATEGY_DIRECT,
    STRATEGY_FIXED_SLTP,
)
from .state import ACTION_COUNTERS, EXEC_COUNTERS, EnvState

_E = {name: i for i, name in enumerate(EXEC_COUNTERS)}
_A = {name: i for i, name in enumerate(ACTION_COUNTERS)}


def coerce_actions(raw: torch.Tensor, params: EnvParams) -> Tuple[torch.Tensor, torch.Tensor]:
    """Map agent actions to {0,1,2} (env.py:343-360). Returns (a, raw_value)."""
    if params.action_space_mode == "continuous":
        vals = raw.reshape(-1).to(torch.float32)
        thr = params.continuous_action_threshold or 0.33
        a = torch.zeros_like(vals, dtype=torch.int64)
        a = torch.where(vals >= thr, torch.ones_like(a), a)
        a = torch.where(vals <= -thr, torch.full_like(a, 2), a)
        return a, vals
    a = raw.reshape(-1).to(torch.int64)
    valid = (a >= 0) & (a <= 2)
    a = torch.where(valid, a, torch.zeros_like(a))
    return a, raw.reshape(-1).to(torch.float32)


def _close_position(
    st: EnvState,
    mask: torch.Tensor,
    fill: torch.Tensor,
    params: EnvParams,
    counter: Optional[str] = None,
) -> None:
    """Close the full position at ``fill`` price for masked envs (ledger from
    default_broker.py:35-53 semantics: % commission on notional, margin
    release, realized pnl credited)."""
    if not bool(mask.any()):
        return
    fill = fill.to(torch.float64)
    pos = st.pos
    realized = pos * (fill - st.avg_entry)
    notional = pos.abs() * fill
    comm = notional * params.commission
    st.cash = torch.where(mask, st.cash + st.margin_used + realized - comm, st.cash)
    st.commission_paid = torch.where(mask, st.commission_paid + comm, st.commission_paid)
    st.last_trade_cost = torch.where(mask, st.last_trade_cost + comm, st.last_trade_cost)
    st.trade_count = torch.where(mask, st.trade_count + 1, st.trade_count)
    won = mask & (realized > 0)
    lost = mask & (realized < 0)
    st.trade_won = torch.where(won, st.trade_won + 1, st.trade_won)
    st.trade_lost = torch.where(lost, st.trade_lost + 1, st.trade_lost)
    net = realized - comm
    st.trade_pnl_sum = torch.where(mask, st.trade_pnl_sum + net, st.trade_pnl_sum)
    st.trade_pnl_sumsq = torch.where(mask, st.trade_pnl_sumsq + net * net, st.trade_pnl_sumsq)
    st.pos = torch.where(mask, torch.zeros_like(pos), pos)
    st.avg_entry = torch.where(mask, torch.zeros_like(fill), st.avg_entry)
    st.margin_used = torch.where(mask, torch.zeros_like(fill), st.margin_used)
    st.br_active = st.br_active & ~mask
    st.br_armed = st.br_armed & ~mask
    if counter is not None:
        st.exec_diag[:, _E[counter]] += mask.to(torch.int32)


def _buy_fill(open_px: torch.Tensor, slip: float) -> torch.Tensor:
    return open_px * (1.0 + slip)


def _sell_fill(open_px: torch.Tensor, slip: float) -> torch.Tensor:
    return open_px * (1.0 - slip)


def step_torch(
    st: EnvState,
    mt: MarketTensors,
    params: EnvParams,
    raw_actions: torch.Tensor,
) -> Dict[str, torch.Tensor]:
    """Advance every env one step. Returns per-env reward/terminated/info."""
    device = st.cursor.device
    N = params.n_envs
    T = mt.T
    slip = params.slippage

    a, raw_vals = coerce_actions(raw_actions, params)

    live = ~st.terminated
    # ---- event-context overlay (env.py:394-440): row = bar_index (pre-
    # advance), position = last published --------------------------------
    eb_i = torch.where(st.end_bar > 0, st.end_bar,
                       torch.full_like(st.end_bar, T))
    row_ov = torch.minimum(st.cursor, eb_i - 1).to(torch.int64)
    pos_sign = torch.sign(st.pos).to(torch.int64)
    blocked_entry = torch.zeros(N, dtype=torch.bool, device=device)
    forced_flat = torch.zeros(N, dtype=torch.bool, device=device)
    if params.event_context_execution_overlay:
        active = mt.ev_no_trade[row_ov] >= params.event_context_no_trade_threshold
        ov = live & active
        st.exec_diag[:, _E["event_context_no_trade_active_steps"]] += ov.to(torch.int32)
        forced_flat = ov & (pos_sign != 0) if params.event_context_force_flat else forced_flat
        if params.event_context_block_new_entries:
            blocked_entry = (
                ov & ~forced_flat & (pos_sign == 0) & ((a == 1) | (a == 2))
            )
        a = torch.where(forced_flat, torch.full_like(a, 3), a)
        a = torch.where(blocked_entry, torch.zeros_like(a), a)
        overridden = forced_flat | blocked_entry
        st.exec_diag[:, _E["event_context_action_overrides"]] += overridden.to(torch.int32)
        st.exec_diag[:, _E["event_context_blocked_entries"]] += blocked_entry.to(torch.int32)
        st.exec_diag[:, _E["event_context_forced_flat_actions"]] += forced_flat.to(torch.int32)

    # ---- action diagnostics (env.py:744-761) ----------------------------
    st.act_diag[:, _A["steps"]] += live.to(torch.int32)
    st.raw_abs_sum += torch.where(live, raw_vals.abs(), torch.zeros_like(raw_vals))
    st.raw_min = torch.where(live, torch.minimum(st.raw_min, raw_vals), st.raw_min)
    st.raw_max = torch.where(live, torch.maximum(st.raw_max, raw_vals), st.raw_max)
    is_long_a = live & (a == 1)
    is_short_a = live & (a == 2)
    is_hold_a = live & ~is_long_a & ~is_short_a
    st.act_diag[:, _A["long_actions"]] += is_long_a.to(torch.int32)
    st.act_diag[:, _A["short_actions"]] += is_short_a.to(torch.int32)
    st.act_diag[:, _A["non_hold_actions"]] += (is_long_a | is_short_a).to(torch.int32)
    st.act_diag[:, _A["hold_actions"]] += is_hold_a.to(torch.int32)
    if params.action_space_mode == "continuous":
        st.act_diag[:, _A["continuous_deadband_actions"]] += is_hold_a.to(torch.int32)

    # ---- advance --------------------------------------------------------
    first = live & ~st.started
    adv = live & st.started
    st.cursor = torch.where(adv, st.cursor + 1, st.cursor)
    exhausted = adv & (st.cursor > eb_i)
    st.cursor = torch.minimum(st.cursor, eb_i)
    valid = adv & ~exhausted
    st.terminated = st.terminated | exhausted

    t = torch.maximum(st.cursor - 1, st.lo_bar).to(torch.int64)  # current bar row
    o_px = mt.open[t]
    h_px = mt.high[t]
    l_px = mt.low[t]
    c_px = mt.close[t]

    st.last_trade_cost = torch.where(
        valid | first, torch.zeros_like(st.last_trade_cost), st.last_trade_cost
    )

    # ---- 1. pending market fills at open(t) -----------------------------
    # latency_bars holds the order in transit (LatencyModel at OHLC
    # granularity): the fill and the pend-clear below are skipped while the
    # wait counter drains
    in_transit = valid & (st.pend_wait > 0) & (st.pend_close | (st.pend_open_dir != 0))
    st.pend_wait = torch.where(in_transit, st.pend_wait - 1, st.pend_wait)
    deliver = ~in_transit

    close_m = valid & deliver & st.pend_close & (st.pos != 0)
    exit_buy = st.pos < 0
    close_fill = torch.where(exit_buy, _buy_fill(o_px, slip), _sell_fill(o_px, slip))
    _close_position(st, close_m, close_fill, params)

    open_m = valid & deliver & (st.pend_open_dir != 0) & (st.pos == 0)
    odir = st.pend_open_dir.to(torch.float32)
    open_fill = torch.where(odir > 0, _buy_fill(o_px, slip), _sell_fill(o_px, slip))
    if bool(open_m.any()):
        size = st.pend_open_size.to(torch.float64)
        fill64 = open_fill.to(torch.float64)
        notional = size * fill64
        comm = notional * params.commission
        # standard: init-margin fraction of notional; leveraged: notional
        # over account leverage (Standard-/LeveragedMarginModel semantics)
        if params.margin_model == 1:
            margin = notional * params.margin_init_rate
        else:
            margin = notional / params.leverage
        if params.enforce_margin_preflight:
            # margin preflight denial (nautilus_gym.py:128-171): drop the
            # order, count it, keep the episode alive
            denied = open_m & (st.cash < margin + comm)
            st.exec_diag[:, _E["margin_preflight_denied"]] += denied.to(torch.int32)
            open_m = open_m & ~denied
        st.cash = torch.where(open_m, st.cash - margin - comm, st.cash)
        st.margin_used = torch.where(open_m, margin, st.margin_used)
        st.commission_paid = torch.where(open_m, st.commission_paid + comm, st.commission_paid)
        st.last_trade_cost = torch.where(open_m, st.last_trade_cost + comm, st.last_trade_cost)
        st.pos = torch.where(open_m, odir.to(torch.float64) * size, st.pos)
        st.avg_entry = torch.where(open_m, fill64, st.avg_entry)
        has_br = (st.pend_sl > 0) | (st.pend_tp > 0)
        arm = open_m & has_br
        st.br_active = st.br_active | arm
        st.br_armed = st.br_armed | arm
        st.br_sl = torch.where(arm, st.pend_sl, st.br_sl)
        st.br_tp = torch.where(arm, st.pend_tp, st.br_tp)
    st.pend_close = st.pend_close & in_transit
    st.pend_open_dir = torch.where(in_transit, st.pend_open_dir,
                                   torch.zeros_like(st.pend_open_dir))
    st.pend_open_size = torch.where(in_transit, st.pend_open_size,
                                    torch.zeros_like(st.pend_open_size))
    st.pend_sl = torch.where(in_transit, st.pend_sl, torch.zeros_like(st.pend_sl))
    st.pend_tp = torch.where(in_transit, st.pend_tp, torch.zeros_like(st.pend_tp))

    # ---- 2. bracket children — collision + limit-fill policies ---------
    # worst_case: stop has absolute priority (open AND adverse extreme
    # before any TP look); ohlc: open, high, low print order; adaptive:
    # up bar assumes the low printed first.  The limit leg honors
    # limit_fill_policy: touch (>=), cross/conservative (strict), and
    # conservative takes no gap price improvement (fills at the limit).
    chk = valid & st.br_active & ~st.br_armed & (st.pos != 0)
    if bool(chk.any()):
        is_long = st.pos > 0
        lim = params.limit_fill_policy

        def tp_ge(px):  # long take-profit trigger at px
            return px >= st.br_tp if lim == 0 else px > st.br_tp

        def tp_le(px):  # short take-profit trigger at px
            return px <= st.br_tp if lim == 0 else px < st.br_tp

        gap_tp_price = st.br_tp if lim == 2 else o_px
        if params.intrabar_collision_policy == 0:  # worst_case
            # long: SL below (sell stop), TP above (sell limit)
            l_sl_gap = chk & is_long & (o_px <= st.br_sl)
            l_sl_hit = chk & is_long & ~l_sl_gap & (l_px <= st.br_sl)
            l_tp_gap = chk & is_long & ~l_sl_gap & ~l_sl_hit & tp_ge(o_px)
            l_tp_hit = (
                chk & is_long & ~l_sl_gap & ~l_sl_hit & ~l_tp_gap & tp_ge(h_px)
            )
            # short: SL above (buy stop), TP below (buy limit)
            s_sl_gap = chk & ~is_long & (o_px >= st.br_sl)
            s_sl_hit = chk & ~is_long & ~s_sl_gap & (h_px >= st.br_sl)
            s_tp_gap = chk & ~is_long & ~s_sl_gap & ~s_sl_hit & tp_le(o_px)
            s_tp_hit = (
                chk & ~is_long & ~s_sl_gap & ~s_sl_hit & ~s_tp_gap & tp_le(l_px)
            )
            sl_gap = l_sl_gap | s_sl_gap
            tp_gap = l_tp_gap | s_tp_gap
            sl_m = sl_gap | l_sl_hit | s_sl_hit
            tp_m = tp_gap | l_tp_hit | s_tp_hit
        else:
            # point-walk: gap checks at the open (stop first), then the
            # two extremes in policy order (close adds no new trigger)
            if params.intrabar_collision_policy == 2:  # adaptive
                low_first = c_px >= o_px  # up bar: dip printed first
            else:  # ohlc: high before low
                low_first = torch.zeros_like(chk)
            sl_gap = chk & torch.where(is_long, o_px <= st.br_sl,
                                       o_px >= st.br_sl)
            tp_gap = chk & ~sl_gap & torch.where(is_long, tp_ge(o_px),
                                                 tp_le(o_px))
            rest = chk & ~sl_gap & ~tp_gap
            # per-extreme triggers (each extreme serves one leg per side)
            low_sl = is_long & (l_px <= st.br_sl)     # long stop at low
            low_tp = ~is_long & tp_le(l_px)           # short limit at low
            high_tp = is_long & tp_ge(h_px)           # long limit at high
            high_sl = ~is_long & (h_px >= st.br_sl)   # short stop at high
            first_sl = torch.where(low_first, low_sl, high_sl)
            first_tp = torch.where(low_first, low_tp, high_tp)
            second_sl = torch.where(low_first, high_sl, low_sl)
            second_tp = torch.where(low_first, high_tp, low_tp)
            sl1 = rest & first_sl
            tp1 = rest & first_tp & ~sl1
            rest2 = rest & ~sl1 & ~tp1
            sl_m = sl_gap | sl1 | (rest2 & second_sl)
            tp_m = tp_gap | tp1 | (rest2 & second_tp & ~(rest2 & second_sl))
        trig = torch.where(
            sl_gap, o_px,
            torch.where(tp_gap, gap_tp_price,
                        torch.where(sl_m, st.br_sl, st.br_tp)),
        )
        # exit side: long exits sell, short exits buy
        fill = torch.where(is_long, _sell_fill(trig, slip), _buy_fill(trig, slip))
        _close_position(st, sl_m, fill, params, counter="bracket_sl_fills")
        _close_position(st, tp_m, fill, params, counter="bracket_tp_fills")
    st.br_armed = st.br_armed & ~valid  # arm active brackets for next bar

    dec = valid | first  # envs making a decision on this bar

    # an order already in transit keeps its remaining latency wait; only a
    # NEW submission (below) starts the clock (matches env_step.hip)
    held_order = st.pend_close | (st.pend_open_dir != 0)

    # ---- 3. ATR true-range update (direct_atr_sltp.py:143-155) ----------
    if params.strategy_id == STRATEGY_ATR_SLTP:
        P = params.atr_period
        pc = st.prev_close_atr
        tr_seed = h_px - l_px
        tr_full = torch.maximum(
            h_px - l_px, torch.maximum((h_px - pc).abs(), (l_px - pc).abs())
        )
        tr = torch.where(torch.isnan(pc), tr_seed, tr_full)
        idx = (st.tr_count % P).to(torch.int64)
        old = st.tr_ring.gather(1, idx.unsqueeze(1)).squeeze(1)
        evict = st.tr_count >= P
        delta = tr - torch.where(evict, old, torch.zeros_like(old))
        st.tr_sum = torch.where(dec, st.tr_sum + delta, st.tr_sum)
        new_ring_val = torch.where(dec, tr, old)
        st.tr_ring.scatter_(1, idx.unsqueeze(1), new_ring_val.unsqueeze(1))
        st.tr_count = torch.where(dec, st.tr_count + 1, st.tr_count)
        st.prev_close_atr = torch.where(dec, c_px, st.prev_close_atr)

    # ---- 4. strategy decision on bar t ----------------------------------
    # action 3 (overlay force-flat) bypasses the strategy (bt_bridge.py:178-188)
    ff3 = dec & (a == 3)
    ff3_close = ff3 & (st.pos != 0)
    st.pend_close = st.pend_close | ff3_close
    st.exec_diag[:, _E["default_orders_submitted"]] += ff3_close.to(torch.int32)
    st.exec_diag[:, _E["event_context_forced_flat_orders"]] += ff3_close.to(torch.int32)

    sdec = dec & ~ff3
    want_long = sdec & (a == 1)
    want_short = sdec & (a == 2)

    if params.strategy_id == STRATEGY_DIRECT:
        entry = want_long | want_short
        st.exec_diag[:, _E["entry_actions_seen"]] += entry.to(torch.int32)
        flip_l = want_long & (st.pos < 0)
        open_l = want_long & (st.pos <= 0)
        flip_s = want_short & (st.pos > 0)
        open_s = want_short & (st.pos >= 0)
        # hold direction already held -> no order
        open_l = open_l & ~(st.pos > 0)
        open_s = open_s & ~(st.pos < 0)
        st.pend_close = st.pend_close | flip_l | flip_s
        dirv = st.pend_open_dir.clone()  # keep in-transit orders (latency)
        dirv = torch.where(open_l, torch.ones_like(dirv), dirv)
        dirv = torch.where(open_s, -torch.ones_like(dirv), dirv)
        st.pend_open_dir = dirv
        st.pend_open_size = torch.where(
            open_l | open_s,
            torch.full_like(st.pend_open_size, params.position_size),
            st.pend_open_size,
        )
        n_orders = (flip_l | flip_s).to(torch.int32) + (open_l | open_s).to(torch.int32)
        st.exec_diag[:, _E["default_orders_submitted"]] += n_orders

    elif params.strategy_id == STRATEGY_FIXED_SLTP:
        sl_d = params.sl_pips * st.pip_env
        tp_d = params.tp_pips * st.pip_env
        flip_l = want_long & (st.pos < 0)
        open_l = want_long & (st.pos <= 0)
        flip_s = want_short & (st.pos > 0)
        open_s = want_short & (st.pos >= 0)
        st.pend_close = st.pend_close | flip_l | flip_s
        dirv = st.pend_open_dir.clone()  # keep in-transit orders (latency)
        dirv = torch.where(open_l, torch.ones_like(dirv), dirv)
        dirv = torch.where(open_s, -torch.ones_like(dirv), dirv)
        st.pend_open_dir = dirv
        opn = open_l | open_s
        st.pend_open_size = torch.where(
            opn, torch.full_like(st.pend_open_size, params.position_size), st.pend_open_size
        )
        st.pend_sl = torch.where(
            open_l, c_px - sl_d, torch.where(open_s, c_px + sl_d, st.pend_sl)
        )
        st.pend_tp = torch.where(
            open_l, c_px + tp_d, torch.where(open_s, c_px - tp_d, st.pend_tp)
        )
        st.exec_diag[:, _E["entry_orders_submitted"]] += opn.to(torch.int32)

    elif params.strategy_id == STRATEGY_ATR_SLTP:
        P = params.atr_period
        in_close_zone = mt.sess_close[t] if params.session_filter else torch.zeros_like(dec)
        in_entry_win = mt.sess_entry[t] if params.session_filter else torch.ones_like(dec)
        sess_ff = sdec & in_close_zone & (st.pos != 0)
        st.pend_close = st.pend_close | sess_ff
        st.exec_diag[:, _E["session_force_closes"]] += sess_ff.to(torch.int32)
        act = sdec & ~sess_ff & (want_long | want_short)
        st.exec_diag[:, _E["entry_actions_seen"]] += act.to(torch.int32)

        blocked_sess = act & ~in_entry_win if params.session_filter else torch.zeros_like(act)
        st.exec_diag[:, _E["blocked_session_filter"]] += blocked_sess.to(torch.int32)
        act = act & ~blocked_sess

        n_tr = torch.clamp(st.tr_count, max=P).to(torch.float32)
        atr = st.tr_sum / torch.clamp(n_tr, min=1.0)
        ready = st.tr_count >= P
        b_warm = act & ~ready
        st.exec_diag[:, _E["blocked_atr_warmup"]] += b_warm.to(torch.int32)
        act = act & ready
        b_atr = act & (atr <= 0)
        st.exec_diag[:, _E["blocked_non_positive_atr"]] += b_atr.to(torch.int32)
        act = act & (atr > 0)

        # sizing (direct_atr_sltp.py:291-311): uses FREE cash
        if params.rel_volume is None:
            size = torch.full_like(st.cash, params.position_size)
        else:
            raw_sz = st.cash * params.rel_volume * params.leverage
            if params.size_mode == SIZE_MODE_NOTIONAL:
                raw_sz = torch.where(c_px > 0, raw_sz / c_px, torch.zeros_like(raw_sz))
            size = torch.clamp(raw_sz, min=params.min_order_volume, max=params.max_order_volume)
        b_sz = act & (size <= 0)
        st.exec_diag[:, _E["blocked_non_positive_size"]] += b_sz.to(torch.int32)
        act = act & (size > 0)
        b_px = act & (c_px <= 0)
        st.exec_diag[:, _E["blocked_non_positive_price"]] += b_px.to(torch.int32)
        act = act & (c_px > 0)

        # effective SL/TP multiples (direct_atr_sltp.py:263-289)
        k_sl, k_tp = params.k_sl, params.k_tp
        if params.sltp_risk_mode != RISK_FIXED_ATR:
            rel = max(0.0, params.rel_volume or 0.0)
            base = params.baseline_rel_volume
            if rel <= base:
                k_sl_eff, k_tp_eff = k_sl, k_tp
            else:
                prog = min(1.0, max(0.0, (rel - base) / (params.max_risk_rel_volume - base)))
                k_sl_eff = max(params.min_k_sl, k_sl * (1.0 - params.rel_volume_sl_shrink_alpha * prog))
                k_tp_eff = k_tp * (1.0 - params.rel_volume_tp_shrink_alpha * prog)
            k_tp_eff = max(k_tp_eff, k_sl_eff * params.min_reward_risk_ratio)
        else:
            k_sl_eff, k_tp_eff = k_sl, k_tp

        sl_dist = k_sl_eff * atr
        tp_dist = k_tp_eff * atr
        if (
            params.sltp_risk_mode == RISK_MARGIN_AWARE
            and params.max_planned_loss_fraction is not None
        ):
            rel_f = max(0.0, params.rel_volume or 0.0)
            mlf = max(0.0, params.max_planned_loss_fraction)
            if rel_f > 0.0 and mlf > 0.0:
                cap = c_px * (mlf / (rel_f * params.leverage))
                sl_dist = torch.minimum(sl_dist, cap)
        if params.min_sltp_frac is not None:
            floor = params.min_sltp_frac * c_px
            sl_dist = torch.maximum(sl_dist, floor)
            tp_dist = torch.maximum(tp_dist, floor)
        if params.max_sltp_frac is not None:
            ceil = params.max_sltp_frac * c_px
            sl_dist = torch.minimum(sl_dist, ceil)
            tp_dist = torch.minimum(tp_dist, ceil)
        tp_dist = torch.where(tp_dist >= c_px, c_px * 0.5, tp_dist)

        go_l = act & (a == 1)
        go_s = act & (a == 2)
        flip_l = go_l & (st.pos < 0)
        open_l = go_l & (st.pos <= 0)
        flip_s = go_s & (st.pos > 0)
        open_s = go_s & (st.pos >= 0)
        st.pend_close = st.pend_close | flip_l | flip_s
        dirv = st.pend_open_dir.clone()  # keep in-transit orders (latency)
        dirv = torch.where(open_l, torch.ones_like(dirv), dirv)
        dirv = torch.where(open_s, -torch.ones_like(dirv), dirv)
        st.pend_open_dir = dirv
        opn = open_l | open_s
        st.pend_open_size = torch.where(opn, size.to(torch.float32), st.pend_open_size)
        st.pend_sl = torch.where(
            open_l, c_px - sl_dist, torch.where(open_s, c_px + sl_dist, st.pend_sl)
        )
        st.pend_tp = torch.where(
            open_l, c_px + tp_dist, torch.where(open_s, c_px - tp_dist, st.pend_tp)
        )
        st.exec_diag[:, _E["entry_orders_submitted"]] += opn.to(torch.int32)

    if params.latency_bars > 0:
        queued_now = (dec & ~held_order
                      & (st.pend_close | (st.pend_open_dir != 0)))
        st.pend_wait = torch.where(
            queued_now,
            torch.full_like(st.pend_wait, params.latency_bars),
            st.pend_wait)

    st.started = st.started | first
    st.episode_step = st.episode_step + dec.to(torch.int32)

    # ---- 4b. financing: FX rollover interest at the scheduled bars ------
    if params.financing_enabled and mt.roll_rate is not None:
        fin = valid & (st.pos != 0)
        interest = st.pos * c_px.to(torch.float64) * mt.roll_rate[t].to(torch.float64)
        st.cash = torch.where(fin, st.cash + interest, st.cash)

    # ---- 5. publish (bt_bridge.py:239-248) ------------------------------
    pub = dec
    st.prev_equity = torch.where(pub, st.equity, st.prev_equity)
    unreal = st.pos * (c_px - st.avg_entry)
    new_eq = st.cash + st.margin_used + unreal
    st.equity = torch.where(pub, new_eq, st.equity)
    busted = pub & (st.equity <= params.min_equity)
    data_done = valid & (st.cursor >= eb_i)
    st.terminated = st.terminated | busted | data_done

    # ---- 6. reward ------------------------------------------------------
    ic = params.initial_cash or 1.0
    r_step = (st.equity - st.prev_equity) / ic
    if params.reward_id == REWARD_SHARPE:
        W = params.sharpe_window
        ridx = (st.rew_count % W).to(torch.int64)
        r32 = r_step.to(torch.float32)
        new_val = torch.where(pub, r32, st.rew_ring.gather(1, ridx.unsqueeze(1)).squeeze(1))
        st.rew_ring.scatter_(1, ridx.unsqueeze(1), new_val.unsqueeze(1))
        st.rew_count = st.rew_count + pub.to(torch.int32)
        n = torch.clamp(st.rew_count, max=W).to(torch.float32)
        s1 = st.rew_ring.sum(dim=1)
        mean = s1 / torch.clamp(n, min=1.0)
        dev = st.rew_ring - mean.unsqueeze(1)
        # zero out unused slots
        slot = torch.arange(W, device=device).unsqueeze(0)
        used = slot < n.unsqueeze(1)
        var = (dev * dev * used).sum(dim=1) / torch.clamp(n - 1.0, min=1.0)
        std = torch.sqrt(var)
        sharpe = torch.where(
            (n >= 2) & (std > 0),
            mean / torch.where(std > 0, std, torch.ones_like(std))
            * (params.annualization_factor ** 0.5),
            torch.zeros_like(std),
        )
        base_reward = sharpe.to(torch.float64)
    elif params.reward_id == REWARD_DD:
        st.peak_equity = torch.where(
            pub,
            torch.maximum(st.peak_equity, torch.maximum(st.equity, st.prev_equity)),
            st.peak_equity,
        )
        dd_norm = torch.where(
            st.peak_equity > 0, (st.peak_equity - st.equity) / ic, torch.zeros_like(r_step)
        )
        base_reward = r_step - params.penalty_lambda * dd_norm
    else:
        base_reward = r_step * params.reward_scale

    penalty = torch.zeros_like(base_reward)
    if (
        params.stage_b_force_close_obs
        and params.stage_b_force_close_reward_penalty
        and params.force_close_exposure_penalty_coef > 0
        and mt.force_close is not None
    ):
        row_fc = torch.clamp(st.cursor, max=T - 1).to(torch.int64)
        hours_to_fc = mt.force_close[row_fc, 1]
        in_zone = mt.force_close[row_fc, 2] > 0
        in_win = (hours_to_fc >= 0) & (
            hours_to_fc <= max(0.0, params.force_close_exposure_penalty_window_hours)
        )
        pos_sign_now = torch.sign(st.pos)
        pen_m = (in_zone | in_win) & (pos_sign_now != 0)
        penalty = torch.where(
            pen_m,
            params.force_close_exposure_penalty_coef * pos_sign_now.abs(),
            penalty,
        )

    reward = torch.where(pub, base_reward - penalty, torch.zeros_like(base_reward))
    st.episode_return = st.episode_return + reward

    # ---- 7. metrics tracking -------------------------------------------
    st.ret_sum = st.ret_sum + torch.where(pub, r_step, torch.zeros_like(r_step))
    st.ret_sumsq = st.ret_sumsq + torch.where(pub, r_step * r_step, torch.zeros_like(r_step))
    st.ret_count = st.ret_count + pub.to(torch.int32)
    st.metric_peak = torch.where(
        pub, torch.maximum(st.metric_peak, st.equity), st.metric_peak
    )
    dd_money = st.metric_peak - st.equity
    st.max_dd_money = torch.where(
        pub, torch.maximum(st.max_dd_money, dd_money), st.max_dd_money
    )
    dd_pct = torch.where(
        st.metric_peak > 0, dd_money / st.metric_peak * 100.0, torch.zeros_like(dd_money)
    )
    st.max_dd_pct = torch.where(pub, torch.maximum(st.max_dd_pct, dd_pct), st.max_dd_pct)

    return {
        "reward": reward,
        "base_reward": torch.where(pub, base_reward, torch.zeros_like(base_reward)),
        "force_close_reward_penalty": torch.where(pub, penalty, torch.zeros_like(penalty)),
        "terminated": st.terminated.clone(),
        "pnl": torch.where(pub, st.equity - st.prev_equity, torch.zeros_like(r_step)),
        "trade_cost": st.last_trade_cost.clone(),
        "coerced_action": a,
    }


# ---------------------------------------------------------------------------
# Observation build
# ---------------------------------------------------------------------------

def build_obs_torch(
    st: EnvState,
    mt: MarketTensors,
    params: EnvParams,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Flat [N, obs_dim] f32 observation (block layout = params.obs_blocks).

    Semantics: default_preprocessor.py:34-77 (price window left-padded with
    the series' first value + first-difference returns + agent state) and
    feature_window_preprocessor.py:99-191 (leakage-safe z-score from
    strictly-past rows via prefix sums, binary passthrough, clip, NaN guard).
    """
    device = st.cursor.device
    N, W = params.n_envs, params.window_size
    T = mt.T
    if out is None:
        out = torch.empty(N, params.obs_dim, dtype=torch.float32, device=device)

    step = st.cursor.to(torch.int64)  # bar_index
    lo = st.lo_bar.to(torch.int64)
    eb = torch.where(st.end_bar > 0, st.end_bar,
                     torch.full_like(st.end_bar, T)).to(torch.int64)
    w_idx = torch.arange(W, device=device, dtype=torch.int64).unsqueeze(0)
    rows = torch.maximum(step.unsqueeze(1) - W + w_idx, lo.unsqueeze(1))
    rows = torch.minimum(rows, (eb - 1).unsqueeze(1))

    off = 0
    slices = params.obs_slices()

    if params.prep_id == PREP_FEATURE_WINDOW:
        F = params.n_features
        S = params.feature_scaling_window
        feat_win = mt.features[rows]  # [N, W, F]
        if params.feature_scaling == "none":
            scaled = feat_win.clone()
        else:
            if params.feature_scaling == "rolling_zscore":
                hist_left = torch.maximum(step - S, lo)
            else:  # expanding
                hist_left = lo.clone()
            m = (step - hist_left).to(torch.float64)
            s1 = mt.feat_prefix1[step] - mt.feat_prefix1[hist_left]  # [N, F] f64
            s2 = mt.feat_prefix2[step] - mt.feat_prefix2[hist_left]
            mean = s1 / torch.clamp(m.unsqueeze(1), min=1.0)
            var = torch.clamp(s2 / torch.clamp(m.unsqueeze(1), min=1.0) - mean * mean, min=0.0)
            std = torch.sqrt(var)
            std = torch.where(std < 1e-8, torch.ones_like(std), std)
            mean32 = mean.to(torch.float32).unsqueeze(1)
            std32 = std.to(torch.float32).unsqueeze(1)
            scaled = (feat_win - mean32) / std32
            # fewer than 2 history rows -> neutral zeros
            scaled = torch.where(
                (m < 2).view(N, 1, 1), torch.zeros_like(scaled), scaled
            )
        if mt.binary_mask is not None and bool(mt.binary_mask.any()):
            scaled = torch.where(
                mt.binary_mask.view(1, 1, F), feat_win, scaled
            )
        clip = params.feature_clip
        if clip and clip > 0:
            scaled = torch.clamp(scaled, -clip, clip)
        scaled = torch.nan_to_num(scaled, nan=0.0, posinf=clip or 0.0, neginf=-(clip or 0.0))
        out[:, slices["features"]] = scaled.reshape(N, W * F)

    if params.include_price_window:
        prices = mt.price[rows]  # [N, W]
        returns = prices - torch.cat([prices[:, :1], prices[:, :-1]], dim=1)
        out[:, slices["prices"]] = prices
        out[:, slices["returns"]] = returns
        window_last = prices[:, -1]
    else:
        window_last = mt.price[
            torch.minimum(torch.maximum(step - 1, lo), eb - 1)]

    if params.include_agent_state:
        ic = params.initial_cash or 1.0
        t_now = torch.minimum(torch.maximum(step - 1, lo), eb - 1)
        price_now = mt.close[t_now]
        pos_sign = torch.sign(st.pos).to(torch.float32)
        unreal = pos_sign * (price_now - window_last) * params.position_size
        agent = torch.stack(
            [
                pos_sign,
                ((st.equity - ic) / ic).to(torch.float32),
                unreal / ic,
                torch.clamp((eb - step).to(torch.float32), min=0.0)
                / torch.clamp((eb - lo).to(torch.float32), min=1.0),
            ],
            dim=1,
        )
        out[:, slices["agent_state"]] = agent

    if params.stage_b_force_close_obs and mt.force_close is not None:
        row_fc = torch.minimum(step, eb - 1)
        out[:, slices["force_close"]] = mt.force_close[row_fc]

    if params.oanda_fx_calendar_obs and mt.calendar is not None:
        row_cal = torch.minimum(step, eb - 1)
        cal = mt.calendar[row_cal][:, :9]  # 9 calendar keys in obs
        ic = params.initial_cash or 1.0
        margin_closeout = torch.zeros(N, 1, dtype=torch.float32, device=device)
        margin_avail = (st.equity / ic).to(torch.float32).unsqueeze(1)
        out[:, slices["calendar"]] = torch.cat([cal, margin_closeout, margin_avail], dim=1)

    del off
    return out
