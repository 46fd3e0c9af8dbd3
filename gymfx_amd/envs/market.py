"""Device-resident market tensors + precomputed per-bar tables.

The whole series (OHLC, feature matrix, z-score prefix sums, calendar /
session / event-overlay tables) is uploaded to HBM once; per-step work is
pure gathers.  This removes the reference's per-step pandas row reads
(/root/reference/app/env.py:369-385) and per-step scaler refits
(/root/reference/preprocessor_plugins/feature_window_preprocessor.py:99-133).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from ..data.feed import MarketData
from .params import EnvParams, PREP_FEATURE_WINDOW


@dataclass
class MarketTensors:
    device: torch.device
    T: int
    open: torch.Tensor      # [T] f32
    high: torch.Tensor      # [T] f32
    low: torch.Tensor       # [T] f32
    close: torch.Tensor     # [T] f32
    price: torch.Tensor     # [T] f32 (configured price column)
    # feature-window preprocessor inputs
    features: Optional[torch.Tensor] = None        # [T, F] f32 (raw values)
    feat_prefix1: Optional[torch.Tensor] = None    # [T+1, F] f64 cumsum
    feat_prefix2: Optional[torch.Tensor] = None    # [T+1, F] f64 cumsum of squares
    binary_mask: Optional[torch.Tensor] = None     # [F] bool
    # event-context overlay columns (always materialized; neutral when absent)
    ev_no_trade: Optional[torch.Tensor] = None     # [T] f32 (default 0)
    ev_spread_mult: Optional[torch.Tensor] = None  # [T] f32 (default 1)
    ev_slip_mult: Optional[torch.Tensor] = None    # [T] f32 (default 1)
    # stage-B force-close table [T, 4] (bars_to_fc, hours_to_fc, in_zone, monday_win)
    force_close: Optional[torch.Tensor] = None
    # oanda calendar table [T, 10] (calendar.CALENDAR_FEATURE_KEYS order)
    calendar: Optional[torch.Tensor] = None
    # ATR-session filter masks [T] bool (entry window / close zone)
    sess_entry: Optional[torch.Tensor] = None
    sess_close: Optional[torch.Tensor] = None
    # financing: per-bar rollover multiplier (cash += pos*close*roll_rate[t])
    roll_rate: Optional[torch.Tensor] = None
    timestamps: Optional[torch.Tensor] = None      # [T] i64 epoch seconds


def _col_f32(md: MarketData, name: str, device: torch.device) -> torch.Tensor:
    return torch.from_numpy(np.ascontiguousarray(md.column(name), dtype=np.float32)).to(device)


def build_market_tensors(
    md: MarketData, params: EnvParams, device: torch.device
) -> MarketTensors:
    T = md.n_rows
    mt = MarketTensors(
        device=device,
        T=T,
        open=_col_f32(md, "OPEN", device),
        high=_col_f32(md, "HIGH", device),
        low=_col_f32(md, "LOW", device),
        close=_col_f32(md, "CLOSE", device),
        price=_col_f32(md, params.price_column, device),
    )

    if params.prep_id == PREP_FEATURE_WINDOW:
        missing = [c for c in params.feature_columns if not md.has_column(c)]
        if missing:
            raise ValueError(
                "feature_window_preprocessor: configured feature_columns "
                f"missing from data: {missing[:5]}{'...' if len(missing) > 5 else ''}"
            )
        feat = np.stack(
            [np.asarray(md.column(c), dtype=np.float64) for c in params.feature_columns], axis=1
        )
        # leakage-safe scaling: prefix sums over STRICTLY PAST rows; the
        # per-step mean/std is then O(1) instead of the reference's O(S)
        # rescan (feature_window_preprocessor.py:99-133).
        p1 = np.zeros((T + 1, feat.shape[1]), dtype=np.float64)
        p2 = np.zeros((T + 1, feat.shape[1]), dtype=np.float64)
        np.cumsum(feat, axis=0, out=p1[1:])
        np.cumsum(feat * feat, axis=0, out=p2[1:])
        mt.features = torch.from_numpy(feat.astype(np.float32)).to(device)
        mt.feat_prefix1 = torch.from_numpy(p1).to(device)
        mt.feat_prefix2 = torch.from_numpy(p2).to(device)
        bmask = np.array(
            [c in set(params.feature_binary_columns) for c in params.feature_columns],
            dtype=bool,
        )
        mt.binary_mask = torch.from_numpy(bmask).to(device)

    def opt_col(name: str, default: float) -> torch.Tensor:
        if name and md.has_column(name):
            return _col_f32(md, name, device)
        return torch.full((T,), default, dtype=torch.float32, device=device)

    mt.ev_no_trade = opt_col(params.event_context_no_trade_column, 0.0)
    mt.ev_spread_mult = opt_col(params.event_context_spread_stress_column, 1.0)
    mt.ev_slip_mult = opt_col(params.event_context_slippage_stress_column, 1.0)

    ts = md.timestamps
    if ts is not None:
        mt.timestamps = torch.from_numpy(np.ascontiguousarray(ts)).to(device)

    # stage-B force-close table: pure function of the (UTC) timestamp fields
    # (/root/reference/app/env.py:530-584) — precomputed for the whole series.
    if params.stage_b_force_close_obs or params.stage_b_force_close_reward_penalty:
        fc = np.zeros((T, 4), dtype=np.float32)
        if ts is not None:
            dow = ((ts // 86400 + 3) % 7).astype(np.int64)  # 1970-01-01 = Thursday
            hour = ((ts % 86400) // 3600).astype(np.int64)
            tf_h = params.timeframe_hours or 1.0
            days_ahead = (params.force_close_dow - dow) % 7
            target_h = days_ahead * 24 + (params.force_close_hour - hour)
            target_h = np.where(target_h < 0, target_h + 7 * 24, target_h)
            hours_to_fc = target_h.astype(np.float64)
            fc[:, 0] = hours_to_fc / max(tf_h, 1e-9)  # bars_to_force_close
            fc[:, 1] = hours_to_fc
            fc[:, 2] = (
                (dow == params.force_close_dow)
                & (hour >= params.force_close_hour)
                & (hour < params.force_close_hour + params.force_close_window_hours)
            )
            fc[:, 3] = (dow == 0) & (hour < params.monday_entry_window_hours)
        mt.force_close = torch.from_numpy(fc).to(device)

    if params.oanda_fx_calendar_obs:
        from ..calendar import compute_fx_calendar_table  # noqa: PLC0415

        if ts is not None:
            tf_h = params.timeframe_hours or 1.0
            cal = compute_fx_calendar_table(ts, timeframe_hours=tf_h)
        else:
            cal = np.zeros((T, 10), dtype=np.float32)
        mt.calendar = torch.from_numpy(cal).to(device)

    # ATR-strategy session filter: minute-of-week window from the naive
    # timestamp fields (direct_atr_sltp.py:320-342).
    if params.session_filter:
        if ts is not None:
            dow = ((ts // 86400 + 3) % 7).astype(np.int64)
            minute = ((ts % 86400) // 60).astype(np.int64)
            cur = dow * 24 * 60 + minute
            start = params.entry_dow_start * 24 * 60 + params.entry_hour_start * 60
            end = params.force_close_dow * 24 * 60 + params.force_close_hour * 60
            in_entry = (cur >= start) & (cur < end)
        else:
            in_entry = np.ones(T, dtype=bool)
        mt.sess_entry = torch.from_numpy(in_entry).to(device)
        mt.sess_close = torch.from_numpy(~in_entry).to(device)
    else:
        mt.sess_entry = torch.ones(T, dtype=torch.bool, device=device)
        mt.sess_close = torch.zeros(T, dtype=torch.bool, device=device)

    return mt
