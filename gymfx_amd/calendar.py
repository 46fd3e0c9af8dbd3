"""OANDA FX calendar — DST-aware America/New_York session policy.

Pure functions, plus a vectorized precompute used by the on-device envs:
because every calendar feature is a pure function of the bar timestamp, the
whole (T, n_features) block is computed ONCE at data-load time (host) and
lives on-device as a lookup table; per-step work is a single gather.

Policy table (capability parity with /root/reference/app/oanda_calendar.py:30-48):
  - FX weekly open: Sunday 17:05 New York.
  - FX weekly close: Friday 16:59 New York.
  - Daily FX break: 16:59-17:05 New York.
  - No-trade window: 16:50-17:10 New York.
  - Friday no-new-position cutoff: 14:00 NY; risk-reduction 15:00 NY;
    force-flat 15:45 NY; last-exit 15:55 NY.
All conversions via IANA America/New_York (DST-aware).
"""
from __future__ import annotations

import datetime as _dt
from typing import Any, Dict, Mapping, Optional

import numpy as np

from zoneinfo import ZoneInfo

OANDA_FX_TIMEZONE = "America/New_York"
CALENDAR_POLICY_ID = "oanda_us_fx_ny_v1"

WEEKLY_OPEN_DOW = 6   # Sunday (Mon=0..Sun=6)
WEEKLY_OPEN_HM = (17, 5)
WEEKLY_CLOSE_DOW = 4  # Friday
WEEKLY_CLOSE_HM = (16, 59)
DAILY_BREAK_START_HM = (16, 59)
DAILY_BREAK_END_HM = (17, 5)
NO_TRADE_WINDOW_START_HM = (16, 50)
NO_TRADE_WINDOW_END_HM = (17, 10)
FRIDAY_NO_NEW_POSITION_HM = (14, 0)
FRIDAY_RISK_REDUCTION_HM = (15, 0)
FRIDAY_FORCE_FLAT_HM = (15, 45)
FRIDAY_LAST_EXIT_HM = (15, 55)
BROKER_DAILY_BREAK_NEAR_MINUTES = 30

_NY = ZoneInfo(OANDA_FX_TIMEZONE)

CALENDAR_FEATURE_KEYS = (
    "hours_to_fx_daily_break",
    "bars_to_fx_daily_break",
    "hours_to_friday_close",
    "bars_to_friday_close",
    "is_friday_risk_reduction_window",
    "is_no_new_position_window",
    "is_force_flat_window",
    "is_broker_daily_break_near",
    "broker_market_open",
    "is_no_trade_window",
)


def _hm_minutes(hm) -> int:
    return hm[0] * 60 + hm[1]


def _to_ny(ts: Any) -> Optional[_dt.datetime]:
    """Coerce a timestamp-like value into an aware NY datetime.

    Naive inputs are treated as UTC. Returns None when unparseable.
    """
    if ts is None:
        return None
    if isinstance(ts, _dt.datetime):
        dt = ts
    elif isinstance(ts, (int, float, np.integer, np.floating)):
        try:
            dt = _dt.datetime.fromtimestamp(float(ts), tz=_dt.timezone.utc)
        except (OverflowError, OSError, ValueError):
            return None
    else:
        s = str(ts).strip()
        if not s:
            return None
        if s.endswith("Z"):
            s = s[:-1] + "+00:00"
        try:
            dt = _dt.datetime.fromisoformat(s.replace("T", " "))
        except ValueError:
            for fmt in ("%Y-%m-%d %H:%M:%S", "%Y-%m-%d %H:%M", "%Y-%m-%d"):
                try:
                    dt = _dt.datetime.strptime(s[: len(fmt) + 6], fmt)
                    break
                except ValueError:
                    continue
            else:
                return None
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=_dt.timezone.utc)
    return dt.astimezone(_NY)


def _minute_of_day(dt: _dt.datetime) -> int:
    return dt.hour * 60 + dt.minute


def _next_friday_close(now_ny: _dt.datetime) -> _dt.datetime:
    days_ahead = (WEEKLY_CLOSE_DOW - now_ny.weekday()) % 7
    candidate = now_ny.replace(
        hour=WEEKLY_CLOSE_HM[0], minute=WEEKLY_CLOSE_HM[1], second=0, microsecond=0
    ) + _dt.timedelta(days=days_ahead)
    if candidate < now_ny:
        candidate += _dt.timedelta(days=7)
    return candidate


def _next_daily_break(now_ny: _dt.datetime) -> _dt.datetime:
    today = now_ny.replace(
        hour=DAILY_BREAK_START_HM[0], minute=DAILY_BREAK_START_HM[1], second=0, microsecond=0
    )
    if today <= now_ny:
        today += _dt.timedelta(days=1)
    return today


def is_no_new_position_window(dt_ny: _dt.datetime) -> bool:
    if dt_ny.weekday() != WEEKLY_CLOSE_DOW:
        return False
    mod = _minute_of_day(dt_ny)
    return _hm_minutes(FRIDAY_NO_NEW_POSITION_HM) <= mod < _hm_minutes(WEEKLY_CLOSE_HM)


def is_friday_risk_reduction_window(dt_ny: _dt.datetime) -> bool:
    if dt_ny.weekday() != WEEKLY_CLOSE_DOW:
        return False
    mod = _minute_of_day(dt_ny)
    return _hm_minutes(FRIDAY_RISK_REDUCTION_HM) <= mod < _hm_minutes(WEEKLY_CLOSE_HM)


def is_force_flat_window(dt_ny: _dt.datetime) -> bool:
    if dt_ny.weekday() != WEEKLY_CLOSE_DOW:
        return False
    mod = _minute_of_day(dt_ny)
    return _hm_minutes(FRIDAY_FORCE_FLAT_HM) <= mod < _hm_minutes(WEEKLY_CLOSE_HM)


def is_broker_daily_break_near(
    dt_ny: _dt.datetime, *, near_minutes: int = BROKER_DAILY_BREAK_NEAR_MINUTES
) -> bool:
    mod = _minute_of_day(dt_ny)
    start = _hm_minutes(DAILY_BREAK_START_HM)
    end = _hm_minutes(DAILY_BREAK_END_HM)
    if start <= mod < end:
        return True
    return start - near_minutes < mod < start


def is_no_trade_window(dt_ny: _dt.datetime) -> bool:
    mod = _minute_of_day(dt_ny)
    return _hm_minutes(NO_TRADE_WINDOW_START_HM) <= mod < _hm_minutes(NO_TRADE_WINDOW_END_HM)


def broker_market_open(dt_ny: _dt.datetime) -> bool:
    """True when FX is tradeable: Sun 17:05 NY .. Fri 16:59 NY minus the
    daily 16:59-17:05 NY break."""
    mod = _minute_of_day(dt_ny)
    dow = dt_ny.weekday()
    if dow == 5:  # Saturday
        return False
    if dow == WEEKLY_OPEN_DOW:
        return mod >= _hm_minutes(WEEKLY_OPEN_HM)
    if dow == WEEKLY_CLOSE_DOW and mod >= _hm_minutes(WEEKLY_CLOSE_HM):
        return False
    if _hm_minutes(DAILY_BREAK_START_HM) <= mod < _hm_minutes(DAILY_BREAK_END_HM):
        return False
    return True


def compute_fx_calendar_features(ts: Any, *, timeframe_hours: float = 4.0) -> Dict[str, float]:
    """Scalar-API parity with the reference
    (/root/reference/app/oanda_calendar.py:187-240): neutral zeros on parse
    failure so a rollout never crashes."""
    neutral = {k: 0.0 for k in CALENDAR_FEATURE_KEYS}
    dt_ny = _to_ny(ts)
    if dt_ny is None:
        return neutral
    tf_h = max(float(timeframe_hours or 0.0), 1e-9)
    hours_to_break = (_next_daily_break(dt_ny) - dt_ny).total_seconds() / 3600.0
    hours_to_close = (_next_friday_close(dt_ny) - dt_ny).total_seconds() / 3600.0
    return {
        "hours_to_fx_daily_break": float(max(hours_to_break, 0.0)),
        "bars_to_fx_daily_break": float(max(hours_to_break, 0.0) / tf_h),
        "hours_to_friday_close": float(max(hours_to_close, 0.0)),
        "bars_to_friday_close": float(max(hours_to_close, 0.0) / tf_h),
        "is_friday_risk_reduction_window": 1.0 if is_friday_risk_reduction_window(dt_ny) else 0.0,
        "is_no_new_position_window": 1.0 if is_no_new_position_window(dt_ny) else 0.0,
        "is_force_flat_window": 1.0 if is_force_flat_window(dt_ny) else 0.0,
        "is_broker_daily_break_near": 1.0 if is_broker_daily_break_near(dt_ny) else 0.0,
        "broker_market_open": 1.0 if broker_market_open(dt_ny) else 0.0,
        "is_no_trade_window": 1.0 if is_no_trade_window(dt_ny) else 0.0,
    }


def resolve_broker_metadata(config: Mapping[str, Any]) -> Dict[str, Optional[str]]:
    return {
        "broker_profile": config.get("broker_profile"),
        "market_type": config.get("market_type"),
        "trade_rate_band_id": config.get("trade_rate_band_id"),
        "calendar_policy_id": config.get("calendar_policy_id"),
    }


# ---------------------------------------------------------------------------
# Vectorized precompute (MI355X path): whole-series calendar table
# ---------------------------------------------------------------------------

def ny_local_fields(epoch_seconds: np.ndarray):
    """Vectorized UTC-epoch -> NY-local (weekday, minute_of_day) via pandas
    tz conversion (DST-aware).  Returns (dow[int8], minute_of_day[int16])."""
    import pandas as pd

    idx = pd.to_datetime(np.asarray(epoch_seconds, dtype="int64"), unit="s", utc=True)
    ny = idx.tz_convert(OANDA_FX_TIMEZONE)
    dow = ny.weekday.to_numpy().astype(np.int8)
    mod = (ny.hour.to_numpy() * 60 + ny.minute.to_numpy()).astype(np.int16)
    sec = ny.second.to_numpy().astype(np.int32)
    return dow, mod, sec


def compute_fx_calendar_table(
    epoch_seconds: np.ndarray, *, timeframe_hours: float = 4.0
) -> np.ndarray:
    """Compute the (T, 10) calendar feature table for a full bar series.

    Column order follows CALENDAR_FEATURE_KEYS.  Semantics are identical to
    compute_fx_calendar_features applied per row (verified by tests).
    """
    ts = np.asarray(epoch_seconds, dtype=np.int64)
    n = ts.shape[0]
    out = np.zeros((n, len(CALENDAR_FEATURE_KEYS)), dtype=np.float32)
    if n == 0:
        return out
    dow, mod, sec = ny_local_fields(ts)
    tf_h = max(float(timeframe_hours or 0.0), 1e-9)

    brk = _hm_minutes(DAILY_BREAK_START_HM)
    brk_end = _hm_minutes(DAILY_BREAK_END_HM)
    close_m = _hm_minutes(WEEKLY_CLOSE_HM)
    open_m = _hm_minutes(WEEKLY_OPEN_HM)

    # minutes (NY local wall-clock) to next daily 16:59 — matches the scalar
    # code, which does wall-clock replace()+timedelta arithmetic.
    sec_frac = sec.astype(np.float64) / 60.0
    cur = mod.astype(np.float64) + sec_frac
    min_to_break = (brk - cur) % (24 * 60)
    min_to_break = np.where(min_to_break == 0.0, 24 * 60.0, min_to_break)
    # exactly at 16:59:00 -> next day per scalar (today <= now -> +1 day)
    hours_to_break = min_to_break / 60.0

    cur_week = dow.astype(np.float64) * 24 * 60 + cur
    close_week = WEEKLY_CLOSE_DOW * 24 * 60 + close_m
    # exact Friday-16:59:00 stays 0 (scalar code uses `candidate < now`).
    min_to_close = (close_week - cur_week) % (7 * 24 * 60)
    hours_to_close = min_to_close / 60.0

    is_friday = dow == WEEKLY_CLOSE_DOW
    rr = is_friday & (mod >= _hm_minutes(FRIDAY_RISK_REDUCTION_HM)) & (mod < close_m)
    nnp = is_friday & (mod >= _hm_minutes(FRIDAY_NO_NEW_POSITION_HM)) & (mod < close_m)
    ff = is_friday & (mod >= _hm_minutes(FRIDAY_FORCE_FLAT_HM)) & (mod < close_m)
    near = ((mod >= brk) & (mod < brk_end)) | (
        (mod > brk - BROKER_DAILY_BREAK_NEAR_MINUTES) & (mod < brk)
    )
    open_ = np.ones(n, dtype=bool)
    open_ &= dow != 5
    open_ = np.where(dow == WEEKLY_OPEN_DOW, mod >= open_m, open_)
    open_ &= ~(is_friday & (mod >= close_m))
    open_ &= ~((mod >= brk) & (mod < brk_end))
    ntw = (mod >= _hm_minutes(NO_TRADE_WINDOW_START_HM)) & (
        mod < _hm_minutes(NO_TRADE_WINDOW_END_HM)
    )

    out[:, 0] = np.maximum(hours_to_break, 0.0)
    out[:, 1] = out[:, 0] / tf_h
    out[:, 2] = np.maximum(hours_to_close, 0.0)
    out[:, 3] = out[:, 2] / tf_h
    out[:, 4] = rr
    out[:, 5] = nnp
    out[:, 6] = ff
    out[:, 7] = near
    out[:, 8] = open_
    out[:, 9] = ntw
    return out


# ---------------------------------------------------------------------------
# FX rollover (financing) schedule
# ---------------------------------------------------------------------------

_CCY_LOCATION = {
    "EUR": "EA19", "USD": "USA", "JPY": "JPN", "GBP": "GBR", "AUD": "AUS",
    "NZD": "NZL", "CAD": "CAN", "CHF": "CHE",
}


def rollover_rate_lookup(rate_data, location: str, month: str) -> float:
    """rate_data: iterable of {LOCATION, TIME(YYYY-MM), Value(percent p.a.)}
    — the reference's monthly central-bank rate schema
    (simulation_engines/bakeoff.py:104-113).  Falls back to the latest
    month at or before the requested one."""
    if hasattr(rate_data, "to_dict"):  # pandas DataFrame (reference schema)
        rate_data = rate_data.to_dict("records")
    best = None
    best_time = ""
    for row in rate_data:
        if str(row["LOCATION"]) != location:
            continue
        t = str(row["TIME"])
        if t <= month and t >= best_time:
            best, best_time = float(row["Value"]), t
    if best is None:
        raise ValueError(f"no rollover rate for {location} <= {month}")
    return best


def compute_rollover_schedule(timestamps, instrument: str, rate_data,
                              rollover_hour_utc: int = 22):
    """Per-bar financing multiplier [T] (numpy f32): at the first bar at or
    after each day's rollover time, ``units * (base - quote) / 100 / 365``
    where units is 3 on Wednesday (weekend rollover) else 1; 0 elsewhere.
    Financing applied by the env step as ``cash += pos * close * sched[t]``
    (FXRolloverInterestModule semantics,
    /root/reference/simulation_engines/nautilus_adapter.py:363-368)."""
    import numpy as np

    base_ccy, _, quote_ccy = instrument.partition("_")
    if not quote_ccy and "/" in instrument:
        base_ccy, _, quote_ccy = instrument.partition("/")
    base_loc = _CCY_LOCATION.get(base_ccy[:3].upper())
    quote_loc = _CCY_LOCATION.get(quote_ccy[:3].upper())
    if base_loc is None or quote_loc is None:
        raise ValueError(f"unknown currencies in instrument '{instrument}'")
    ts = np.asarray(timestamps, dtype=np.int64)
    out = np.zeros(len(ts), dtype=np.float32)
    day = ts // 86400
    secs = ts % 86400
    boundary = rollover_hour_utc * 3600
    seen_days = set()
    for i in range(len(ts)):
        if secs[i] >= boundary and int(day[i]) not in seen_days:
            seen_days.add(int(day[i]))
            t = _dt.datetime.fromtimestamp(int(ts[i]), tz=_dt.timezone.utc)
            month = f"{t.year:04d}-{t.month:02d}"
            diff = (rollover_rate_lookup(rate_data, base_loc, month)
                    - rollover_rate_lookup(rate_data, quote_loc, month))
            units = 3.0 if t.weekday() == 2 else 1.0  # Wed: weekend rollover
            out[i] = units * diff / 100.0 / 365.0
    return out
