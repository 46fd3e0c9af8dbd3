"""SoA env state: one tensor per field, N-wide, resident on device.

This is the MI355X-native replacement for the reference's per-env Python
object + worker thread + BTBridge shared state
(/root/reference/app/bt_bridge.py:30-83): all N environments live as
structure-of-arrays tensors and advance in lockstep inside one fused kernel.
"""
from __future__ import annotations

from dataclasses import dataclass, fields
from typing import Dict

import torch

from .params import EnvParams

# execution-diagnostics counter ids (superset of bt_bridge.py:68-83)
EXEC_COUNTERS = (
    "entry_actions_seen",
    "entry_orders_submitted",
    "blocked_session_filter",
    "blocked_atr_warmup",
    "blocked_non_positive_atr",
    "blocked_non_positive_size",
    "blocked_non_positive_price",
    "default_orders_submitted",
    "plugin_apply_errors",
    "event_context_no_trade_active_steps",
    "event_context_action_overrides",
    "event_context_blocked_entries",
    "event_context_forced_flat_actions",
    "event_context_forced_flat_orders",
    "session_force_closes",
    "bracket_sl_fills",
    "bracket_tp_fills",
    "margin_preflight_denied",
)

ACTION_COUNTERS = (
    "steps",
    "hold_actions",
    "long_actions",
    "short_actions",
    "non_hold_actions",
    "continuous_deadband_actions",
)


@dataclass
class EnvState:
    # cursor / lifecycle
    cursor: torch.Tensor          # i32 [N] — bar_index (1-based bars seen)
    started: torch.Tensor         # bool [N] — first action applied yet?
    terminated: torch.Tensor      # bool [N]
    # account
    pos: torch.Tensor             # f64 [N] signed units
    avg_entry: torch.Tensor       # f32 [N]
    cash: torch.Tensor            # f64 [N] free cash (initial - margin - comm + realized)
    margin_used: torch.Tensor     # f32 [N]
    equity: torch.Tensor          # f32 [N]
    prev_equity: torch.Tensor     # f32 [N]
    peak_equity: torch.Tensor     # f32 [N] (dd reward)
    commission_paid: torch.Tensor  # f32 [N]
    last_trade_cost: torch.Tensor  # f32 [N]
    trade_count: torch.Tensor     # i32 [N]
    # pending orders (decided this bar, fill at next bar open)
    pend_close: torch.Tensor      # bool [N]
    pend_open_dir: torch.Tensor   # i8 [N] (0 none, +1 long, -1 short)
    pend_open_size: torch.Tensor  # f32 [N]
    pend_sl: torch.Tensor         # f32 [N] (0 = none) — absolute stop price
    pend_tp: torch.Tensor         # f32 [N] (0 = none) — absolute limit price
    # active bracket on the open position
    br_active: torch.Tensor       # bool [N]
    br_armed: torch.Tensor        # bool [N] (children active from next bar)
    br_sl: torch.Tensor           # f32 [N]
    br_tp: torch.Tensor           # f32 [N]
    # ATR strategy state
    tr_ring: torch.Tensor         # f32 [N, atr_period]
    tr_count: torch.Tensor        # i32 [N]
    tr_sum: torch.Tensor          # f32 [N]
    prev_close_atr: torch.Tensor  # f32 [N] (NaN = unset)
    # sharpe reward ring
    pend_wait: torch.Tensor       # i32 [N] bars the order is in transit
    rew_ring: torch.Tensor        # f32 [N, sharpe_window]
    rew_count: torch.Tensor       # i32 [N]
    rew_s1: torch.Tensor          # f64 [N] sharpe running sum (O(1)/step)
    rew_s2: torch.Tensor          # f64 [N] sharpe running sum of squares
    # per-trade stats (metrics rollup)
    trade_won: torch.Tensor       # i32 [N]
    trade_lost: torch.Tensor      # i32 [N]
    trade_pnl_sum: torch.Tensor   # f32 [N]
    trade_pnl_sumsq: torch.Tensor  # f32 [N]
    # drawdown tracking for metrics (running peak over published equity)
    metric_peak: torch.Tensor     # f32 [N]
    max_dd_money: torch.Tensor    # f32 [N]
    max_dd_pct: torch.Tensor      # f32 [N]
    # step-return accumulators (sharpe analyzer-equivalent)
    ret_sum: torch.Tensor         # f32 [N]
    ret_sumsq: torch.Tensor       # f32 [N]
    ret_count: torch.Tensor       # i32 [N]
    # episode bookkeeping
    episode_step: torch.Tensor    # i32 [N] steps since reset
    episode_return: torch.Tensor  # f32 [N]
    start_offset: torch.Tensor    # i32 [N] — reset cursor base
    # multi-instrument bounds (BASELINE config #5); single-pair: lo=0, end=T
    lo_bar: torch.Tensor          # i32 [N] — first bar of this env's block
    end_bar: torch.Tensor         # i32 [N] — one past last bar of the block
    inst_id: torch.Tensor         # i32 [N] — instrument index (diagnostics)
    pip_env: torch.Tensor         # f32 [N] — per-env pip size
    # diagnostics
    exec_diag: torch.Tensor       # i32 [N, len(EXEC_COUNTERS)]
    act_diag: torch.Tensor        # i32 [N, len(ACTION_COUNTERS)]
    raw_abs_sum: torch.Tensor     # f32 [N]
    raw_min: torch.Tensor         # f32 [N]
    raw_max: torch.Tensor         # f32 [N]

    def to_dict(self) -> Dict[str, torch.Tensor]:
        return {f.name: getattr(self, f.name) for f in fields(self)}


def alloc_state(params: EnvParams, device: torch.device) -> EnvState:
    N = params.n_envs
    f32 = dict(dtype=torch.float32, device=device)
    # Money ledger runs in f64: per-step equity deltas (~1e-4 on a 1e4
    # account at position_size=1) vanish below f32 resolution; market data
    # and observations stay f32.
    f64 = dict(dtype=torch.float64, device=device)
    i32 = dict(dtype=torch.int32, device=device)
    boolk = dict(dtype=torch.bool, device=device)

    st = EnvState(
        cursor=torch.ones(N, **i32),
        started=torch.zeros(N, **boolk),
        terminated=torch.zeros(N, **boolk),
        pos=torch.zeros(N, **f64),
        avg_entry=torch.zeros(N, **f64),
        cash=torch.full((N,), params.initial_cash, **f64),
        margin_used=torch.zeros(N, **f64),
        equity=torch.full((N,), params.initial_cash, **f64),
        prev_equity=torch.full((N,), params.initial_cash, **f64),
        peak_equity=torch.zeros(N, **f64),
        commission_paid=torch.zeros(N, **f64),
        last_trade_cost=torch.zeros(N, **f64),
        trade_count=torch.zeros(N, **i32),
        pend_close=torch.zeros(N, **boolk),
        pend_open_dir=torch.zeros(N, dtype=torch.int8, device=device),
        pend_open_size=torch.zeros(N, **f32),
        pend_sl=torch.zeros(N, **f32),
        pend_tp=torch.zeros(N, **f32),
        pend_wait=torch.zeros(N, **i32),
        br_active=torch.zeros(N, **boolk),
        br_armed=torch.zeros(N, **boolk),
        br_sl=torch.zeros(N, **f32),
        br_tp=torch.zeros(N, **f32),
        tr_ring=torch.zeros(N, max(params.atr_period, 1), **f32),
        tr_count=torch.zeros(N, **i32),
        tr_sum=torch.zeros(N, **f32),
        prev_close_atr=torch.full((N,), float("nan"), **f32),
        rew_ring=torch.zeros(N, max(params.sharpe_window, 2), **f32),
        rew_count=torch.zeros(N, **i32),
        rew_s1=torch.zeros(N, **f64),
        rew_s2=torch.zeros(N, **f64),
        trade_won=torch.zeros(N, **i32),
        trade_lost=torch.zeros(N, **i32),
        trade_pnl_sum=torch.zeros(N, **f64),
        trade_pnl_sumsq=torch.zeros(N, **f64),
        metric_peak=torch.full((N,), params.initial_cash, **f64),
        max_dd_money=torch.zeros(N, **f64),
        max_dd_pct=torch.zeros(N, **f64),
        ret_sum=torch.zeros(N, **f64),
        ret_sumsq=torch.zeros(N, **f64),
        ret_count=torch.zeros(N, **i32),
        episode_step=torch.zeros(N, **i32),
        episode_return=torch.zeros(N, **f64),
        start_offset=torch.zeros(N, **i32),
        lo_bar=torch.zeros(N, **i32),
        end_bar=torch.zeros(N, **i32),
        inst_id=torch.zeros(N, **i32),
        pip_env=torch.full((N,), params.pip_size, **f32),
        exec_diag=torch.zeros(N, len(EXEC_COUNTERS), **i32),
        act_diag=torch.zeros(N, len(ACTION_COUNTERS), **i32),
        raw_abs_sum=torch.zeros(N, **f32),
        raw_min=torch.full((N,), float("inf"), **f32),
        raw_max=torch.full((N,), float("-inf"), **f32),
    )
    return st


def reset_state_(st: EnvState, params: EnvParams, mask: torch.Tensor) -> None:
    """In-place reset of the envs selected by ``mask`` (bool [N])."""
    ic = params.initial_cash
    st.cursor[mask] = 1 + st.start_offset[mask]
    st.started[mask] = False
    st.terminated[mask] = False
    st.pos[mask] = 0.0
    st.avg_entry[mask] = 0.0
    st.cash[mask] = ic
    st.margin_used[mask] = 0.0
    st.equity[mask] = ic
    st.prev_equity[mask] = ic
    st.peak_equity[mask] = 0.0
    st.commission_paid[mask] = 0.0
    st.last_trade_cost[mask] = 0.0
    st.trade_count[mask] = 0
    st.pend_close[mask] = False
    st.pend_open_dir[mask] = 0
    st.pend_open_size[mask] = 0.0
    st.pend_sl[mask] = 0.0
    st.pend_tp[mask] = 0.0
    st.pend_wait[mask] = 0
    st.br_active[mask] = False
    st.br_armed[mask] = False
    st.br_sl[mask] = 0.0
    st.br_tp[mask] = 0.0
    st.tr_ring[mask] = 0.0
    st.tr_count[mask] = 0
    st.tr_sum[mask] = 0.0
    st.prev_close_atr[mask] = float("nan")
    st.rew_ring[mask] = 0.0
    st.rew_count[mask] = 0
    st.rew_s1[mask] = 0.0
    st.rew_s2[mask] = 0.0
    st.trade_won[mask] = 0
    st.trade_lost[mask] = 0
    st.trade_pnl_sum[mask] = 0.0
    st.trade_pnl_sumsq[mask] = 0.0
    st.metric_peak[mask] = ic
    st.max_dd_money[mask] = 0.0
    st.max_dd_pct[mask] = 0.0
    st.ret_sum[mask] = 0.0
    st.ret_sumsq[mask] = 0.0
    st.ret_count[mask] = 0
    st.episode_step[mask] = 0
    st.episode_return[mask] = 0.0
    st.exec_diag[mask] = 0
    st.act_diag[mask] = 0
    st.raw_abs_sum[mask] = 0.0
    st.raw_min[mask] = float("inf")
    st.raw_max[mask] = float("-inf")
