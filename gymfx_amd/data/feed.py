"""Market data layer: CSV -> columnar host arrays -> device tensors.

Capability parity with the reference data layer
(/root/reference/data_feed_plugins/default_data_feed.py:36-79 and
/root/reference/app/data_handler.py:11-114): CSV with a
``DATE_TIME,OPEN,HIGH,LOW,CLOSE,VOLUME`` header (case preserved), missing
OHLC back-filled from the configured ``price_column``, missing VOLUME = 0,
``max_rows`` truncation, rows with unparseable timestamps dropped.

Unlike the reference (pandas DataFrame consumed row-by-row), the product of
this layer is a columnar ``MarketData`` whose arrays are uploaded ONCE to
device HBM; every per-step read afterwards is an on-device gather.
"""
from __future__ import annotations

import datetime as _dt
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import numpy as np

OHLC_COLUMNS = ("OPEN", "HIGH", "LOW", "CLOSE")


def _parse_timestamps(values: np.ndarray) -> np.ndarray:
    """Parse string timestamps to int64 epoch seconds; NaT -> INT64_MIN."""
    import pandas as pd

    ts = pd.to_datetime(pd.Series(values), errors="coerce", utc=True)
    out = ts.astype("int64").to_numpy()  # ns since epoch; NaT -> int64 min
    nat = ts.isna().to_numpy()
    sec = np.where(nat, np.int64(np.iinfo(np.int64).min), out // 1_000_000_000)
    return sec


@dataclass
class MarketData:
    """Columnar OHLCV (+feature) series for one instrument."""

    columns: Dict[str, np.ndarray]
    timestamps: Optional[np.ndarray] = None  # int64 epoch seconds (UTC)
    date_strings: Optional[np.ndarray] = None
    instrument: str = "EUR_USD"
    timeframe: str = "M1"
    meta: Dict[str, Any] = field(default_factory=dict)

    @property
    def n_rows(self) -> int:
        for v in self.columns.values():
            return int(len(v))
        return 0

    def __len__(self) -> int:
        return self.n_rows

    def column(self, name: str) -> np.ndarray:
        if name not in self.columns:
            raise KeyError(f"column '{name}' not found; have {sorted(self.columns)}")
        return self.columns[name]

    def has_column(self, name: str) -> bool:
        return name in self.columns

    @property
    def open(self) -> np.ndarray:
        return self.column("OPEN")

    @property
    def high(self) -> np.ndarray:
        return self.column("HIGH")

    @property
    def low(self) -> np.ndarray:
        return self.column("LOW")

    @property
    def close(self) -> np.ndarray:
        return self.column("CLOSE")

    @property
    def volume(self) -> np.ndarray:
        return self.column("VOLUME")

    def timeframe_hours(self) -> float:
        """Infer bar duration in hours from the timeframe label
        (semantics of /root/reference/app/env.py:510-528)."""
        raw = str(self.timeframe or "").strip().lower()
        if "_" in raw:
            raw = raw.rsplit("_", 1)[-1]
        try:
            if raw.startswith("m") and raw[1:].isdigit():  # M1, M5 style
                return int(raw[1:]) / 60.0
            if raw.startswith("h") and raw[1:].isdigit():
                return float(int(raw[1:]))
            if raw.endswith("m"):
                return max(0.0, int(raw[:-1]) / 60.0)
            if raw.endswith("h"):
                return float(int(raw[:-1]))
            if raw.endswith("d"):
                return float(int(raw[:-1]) * 24)
        except ValueError:
            return 0.0
        return 0.0


def load_csv(
    file_path: str,
    *,
    date_column: str = "DATE_TIME",
    price_column: str = "CLOSE",
    headers: bool = True,
    max_rows: Optional[int] = None,
    instrument: str = "EUR_USD",
    timeframe: str = "M1",
) -> MarketData:
    """Load an OHLCV CSV into a MarketData (columnar, float64)."""
    import pandas as pd

    df = pd.read_csv(file_path, header=0 if headers else None, nrows=max_rows)
    if not headers:
        df.columns = [f"COL{i}" for i in range(df.shape[1])]

    timestamps = None
    date_strings = None
    # case-insensitive date-column detection; duplicates dropped (the
    # reference matches 'DATE_TIME' with strip().lower() and removes the
    # extras, app/data_handler.py:38-51)
    want = date_column.strip().lower()
    dt_cols = [c for c in df.columns if str(c).strip().lower() == want]
    if dt_cols:
        raw_dates = df[dt_cols[0]].astype(str).to_numpy()
        ts = _parse_timestamps(raw_dates)
        keep = ts != np.iinfo(np.int64).min
        df = df.loc[keep].reset_index(drop=True)
        timestamps = ts[keep]
        date_strings = raw_dates[keep]
        df = df.drop(columns=dt_cols)

    if price_column not in df.columns:
        m = [c for c in df.columns
             if str(c).strip().lower() == price_column.strip().lower()]
        if not m:
            raise ValueError(f"price_column '{price_column}' not found in data")
        price_column = m[0]

    columns: Dict[str, np.ndarray] = {}
    for name in df.columns:
        # numeric coercion with NaN->0 (data_handler.py:63-64); columns
        # with no parseable value at all are treated as auxiliary and
        # skipped rather than becoming all-zero features
        vals = pd.to_numeric(df[name], errors="coerce")
        if vals.notna().any() or len(vals) == 0:
            columns[name] = vals.fillna(0.0).to_numpy(dtype=np.float64)

    for col in OHLC_COLUMNS:
        if col not in columns:
            columns[col] = columns[price_column].copy()
    if "VOLUME" not in columns:
        columns["VOLUME"] = np.zeros(len(columns[price_column]), dtype=np.float64)

    return MarketData(
        columns=columns,
        timestamps=timestamps,
        date_strings=date_strings,
        instrument=instrument,
        timeframe=timeframe,
        meta={"source": str(file_path)},
    )


def write_csv(md: MarketData, file_path: str, *, date_column: str = "DATE_TIME") -> None:
    import pandas as pd

    data: Dict[str, Any] = {}
    if md.date_strings is not None:
        data[date_column] = md.date_strings
    elif md.timestamps is not None:
        data[date_column] = [
            _dt.datetime.fromtimestamp(int(t), tz=_dt.timezone.utc).strftime("%Y-%m-%d %H:%M:%S")
            for t in md.timestamps
        ]
    data.update({k: v for k, v in md.columns.items()})
    pd.DataFrame(data).to_csv(file_path, index=False)


def synthetic_ohlcv(
    n_rows: int,
    *,
    seed: int = 0,
    start_price: float = 1.10,
    vol: float = 1e-4,
    drift: float = 0.0,
    start: str = "2024-01-01 00:00:00",
    bar_minutes: int = 1,
    instrument: str = "EUR_USD",
    extra_feature_columns: int = 0,
) -> MarketData:
    """Deterministic synthetic OHLCV random walk (bench + test fixture).

    The bench harness trains on data of this shape per BASELINE.json
    ("synthetic OHLCV ticks"); there is no network for real datasets.
    """
    rng = np.random.default_rng(seed)
    steps = rng.normal(loc=drift, scale=vol, size=n_rows)
    close = start_price * np.exp(np.cumsum(steps))
    open_ = np.concatenate([[start_price], close[:-1]])
    spread = np.abs(rng.normal(scale=vol * start_price, size=n_rows))
    high = np.maximum(open_, close) + spread
    low = np.minimum(open_, close) - spread
    volume = rng.integers(100, 10_000, size=n_rows).astype(np.float64)

    t0 = int(
        _dt.datetime.strptime(start, "%Y-%m-%d %H:%M:%S")
        .replace(tzinfo=_dt.timezone.utc)
        .timestamp()
    )
    timestamps = t0 + np.arange(n_rows, dtype=np.int64) * (60 * bar_minutes)

    columns = {
        "OPEN": open_,
        "HIGH": high,
        "LOW": low,
        "CLOSE": close,
        "VOLUME": volume,
    }
    for k in range(extra_feature_columns):
        columns[f"FEAT_{k}"] = rng.normal(size=n_rows)

    return MarketData(
        columns=columns,
        timestamps=timestamps,
        instrument=instrument,
        timeframe=f"M{bar_minutes}",
        meta={"source": "synthetic", "seed": seed},
    )


def uptrend_ohlcv(n_rows: int = 500, *, lo: float = 1.10, hi: float = 1.20) -> MarketData:
    """Monotonic uptrend fixture (buy&hold smoke invariant)."""
    close = np.linspace(lo, hi, n_rows)
    t0 = int(
        _dt.datetime(2024, 1, 1, tzinfo=_dt.timezone.utc).timestamp()
    )
    return MarketData(
        columns={
            "OPEN": close.copy(),
            "HIGH": close + 1e-5,
            "LOW": close - 1e-5,
            "CLOSE": close.copy(),
            "VOLUME": np.zeros(n_rows),
        },
        timestamps=t0 + np.arange(n_rows, dtype=np.int64) * 60,
        meta={"source": "uptrend"},
    )


def concat_markets(mds: "list[MarketData]") -> MarketData:
    """Concatenate several instruments' series along the bar axis into one
    MarketData (BASELINE config #5: multi-pair market tensor).

    Per-instrument block boundaries, names and pip sizes land in
    ``meta["instrument_blocks"]``; the vectorized env assigns each env an
    instrument block and bounds its episode/observation windows to it
    (per-env lo_bar/end_bar), so no window or scaling statistic ever crosses
    an instrument boundary.
    """
    if not mds:
        raise ValueError("concat_markets needs at least one MarketData")
    if len(mds) == 1:
        md = mds[0]
        md.meta.setdefault("instrument_blocks", [{
            "instrument": md.instrument, "lo": 0, "end": md.n_rows,
            "pip_size": md.meta.get("pip_size", 0.0001),
        }])
        return md
    keys = set(mds[0].columns)
    for m in mds[1:]:
        if set(m.columns) != keys:
            raise ValueError("all instruments must share the same columns")
    blocks = []
    off = 0
    for m in mds:
        blocks.append({
            "instrument": m.instrument, "lo": off, "end": off + m.n_rows,
            "pip_size": m.meta.get("pip_size", 0.0001),
        })
        off += m.n_rows
    cols = {k: np.concatenate([m.columns[k] for m in mds]) for k in keys}
    ts = None
    if all(m.timestamps is not None for m in mds):
        ts = np.concatenate([m.timestamps for m in mds])
    ds = None
    if all(m.date_strings is not None for m in mds):
        ds = np.concatenate([m.date_strings for m in mds])
    return MarketData(
        columns=cols, timestamps=ts, date_strings=ds,
        instrument="+".join(m.instrument for m in mds),
        timeframe=mds[0].timeframe,
        meta={"instrument_blocks": blocks},
    )
