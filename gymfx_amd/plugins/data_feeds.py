"""Data-feed plugins: CSV and synthetic OHLCV sources -> MarketData.

Contract parity with /root/reference/data_feed_plugins/default_data_feed.py:
``load_data(config)`` returns the columnar series (OHLC back-filled from the
price column, VOLUME->0).  ``build_bt_feed`` has no meaning here — the
MI355X engine consumes MarketData tensors directly; ``build_market(data,
config)`` is the engine-facing equivalent (identity for MarketData).
"""
from __future__ import annotations

from typing import Any, Dict

from ..data.feed import MarketData, load_csv, synthetic_ohlcv
from .base import PluginBase


class DefaultDataFeed(PluginBase):
    plugin_params = {
        "input_data_file": "examples/data/eurusd_sample.csv",
        "date_column": "DATE_TIME",
        "headers": True,
        "max_rows": None,
        "price_column": "CLOSE",
    }

    def load_data(self, config: Dict[str, Any]) -> MarketData:
        p = self._resolve(config)
        return load_csv(
            str(config.get("input_data_file", p["input_data_file"])),
            date_column=str(config.get("date_column", p["date_column"])),
            price_column=str(config.get("price_column", p["price_column"])),
            headers=bool(config.get("headers", p["headers"])),
            max_rows=config.get("max_rows", p["max_rows"]),
            instrument=str(config.get("instrument", "EUR_USD")),
            timeframe=str(config.get("timeframe", "M1")),
        )

    def build_market(self, data: MarketData, config: Dict[str, Any]) -> MarketData:
        return data


class SyntheticDataFeed(PluginBase):
    """Deterministic synthetic OHLCV random walk (bench/training fixture)."""

    plugin_params = {
        "synthetic_rows": 100_000,
        "synthetic_seed": 0,
        "synthetic_start_price": 1.10,
        "synthetic_vol": 1e-4,
        "synthetic_drift": 0.0,
        "synthetic_bar_minutes": 1,
        "synthetic_extra_features": 0,
    }

    def load_data(self, config: Dict[str, Any]) -> MarketData:
        p = self._resolve(config)
        pairs = int(config.get("synthetic_pairs", 1) or 1)
        # BASELINE config #5 names EURUSD+GBPUSD+USDJPY: realistic start
        # prices and per-instrument pip sizes (JPY quotes pip = 0.01)
        majors = [
            (str(config.get("instrument", "EUR_USD")), None, 0.0001),
            ("GBP_USD", 1.27, 0.0001),
            ("USD_JPY", 150.0, 0.01),
        ]
        mds = []
        for i in range(pairs):
            if i < len(majors):
                name, px, pip = majors[i]
            else:
                name, px, pip = f"PAIR_{i}", None, 0.0001
            if px is None:
                px = float(p["synthetic_start_price"]) * (1.0 + 0.1 * i)
            md = synthetic_ohlcv(
                int(p["synthetic_rows"]),
                seed=int(p["synthetic_seed"]) + 7919 * i,
                start_price=px,
                vol=float(p["synthetic_vol"]),
                drift=float(p["synthetic_drift"]),
                bar_minutes=int(p["synthetic_bar_minutes"]),
                instrument=name,
                extra_feature_columns=int(p["synthetic_extra_features"]),
            )
            md.meta["pip_size"] = pip
            mds.append(md)
        if len(mds) == 1:
            return mds[0]
        from ..data.feed import concat_markets
        return concat_markets(mds)

    def build_market(self, data: MarketData, config: Dict[str, Any]) -> MarketData:
        return data
