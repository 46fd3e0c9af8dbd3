"""Preprocessor plugins — single-env numpy implementations.

These are the reference-compatible observation builders
(default_preprocessor.py:34-77 and feature_window_preprocessor.py:99-254
semantics).  They serve three roles:
  1. the single-env Gymnasium wrapper's observation path,
  2. an INDEPENDENT oracle for the vectorized torch/HIP observation
     builders (tests cross-check all three),
  3. the behavioral contract for third-party preprocessor plugins.
"""
from __future__ import annotations

from typing import Any, Dict, List

import numpy as np

from .base import PluginBase

_VALID_SCALINGS = ("none", "rolling_zscore", "expanding_zscore")


def _column(data: Any, name: str) -> np.ndarray:
    """Accept MarketData or a pandas DataFrame (reference contract)."""
    if hasattr(data, "column"):
        return np.asarray(data.column(name), dtype=np.float64)
    return data[name].astype(float).to_numpy()


def _has_column(data: Any, name: str) -> bool:
    if hasattr(data, "has_column"):
        return data.has_column(name)
    return name in data.columns


def _agent_state(
    obs: Dict[str, np.ndarray],
    bridge_state: Dict[str, Any],
    config: Dict[str, Any],
    window_last: float,
    include_price: bool,
) -> None:
    initial_cash = float(bridge_state.get("initial_cash", 1.0) or 1.0)
    equity = float(bridge_state.get("equity", initial_cash))
    price = float(bridge_state.get("price", 0.0) or 0.0)
    position = int(bridge_state.get("position", 0))
    bar_index = int(bridge_state.get("bar_index", 0))
    total_bars = int(bridge_state.get("total_bars", 1) or 1)

    pos_size = float(config.get("position_size", 1.0))
    ref_price = window_last if include_price else price
    unrealized = position * (price - ref_price) * pos_size

    equity_norm = (equity - initial_cash) / initial_cash if initial_cash else 0.0
    pnl_norm = unrealized / initial_cash if initial_cash else 0.0
    remaining = max(0, total_bars - bar_index) / max(1, total_bars)

    obs["position"] = np.array([float(position)], dtype=np.float32)
    obs["equity_norm"] = np.array([float(equity_norm)], dtype=np.float32)
    obs["unrealized_pnl_norm"] = np.array([float(pnl_norm)], dtype=np.float32)
    obs["steps_remaining_norm"] = np.array([float(remaining)], dtype=np.float32)


def _price_window(values: np.ndarray, step: int, window_size: int) -> np.ndarray:
    left = max(0, step - window_size)
    window = values[left:step] if step > 0 else values[:0]
    if len(window) < window_size:
        fill = float(window[0]) if len(window) else float(values[0]) if len(values) else 0.0
        pad = np.full(window_size - len(window), fill, dtype=float)
        window = np.concatenate([pad, window])
    return window


class DefaultPreprocessor(PluginBase):
    plugin_params = {
        "window_size": 32,
        "price_column": "CLOSE",
    }

    def make_observation(
        self,
        *,
        data: Any,
        step: int,
        bridge_state: Dict[str, Any],
        config: Dict[str, Any],
    ) -> Dict[str, np.ndarray]:
        window_size = int(config.get("window_size", self.params["window_size"]))
        price_col = config.get("price_column", self.params["price_column"])
        values = _column(data, price_col)
        window = _price_window(values, step, window_size)
        returns = np.diff(window, prepend=window[0])
        obs: Dict[str, np.ndarray] = {
            "prices": window.astype(np.float32),
            "returns": returns.astype(np.float32),
        }
        _agent_state(obs, bridge_state, config, float(window[-1]), True)
        return obs


class FeatureWindowPreprocessor(PluginBase):
    plugin_params: Dict[str, Any] = {
        "window_size": 32,
        "price_column": "CLOSE",
        "feature_columns": [],
        "feature_binary_columns": [],
        "feature_scaling": "rolling_zscore",
        "feature_scaling_window": 256,
        "include_price_window": True,
        "include_agent_state": True,
        "feature_clip": 10.0,
    }

    plugin_debug_vars: List[str] = [
        "window_size",
        "price_column",
        "feature_scaling",
        "feature_scaling_window",
        "include_price_window",
        "include_agent_state",
    ]

    def __init__(self, config: Dict[str, Any] | None = None):
        super().__init__(config)
        self._cache_key = None
        self._cache_matrix: np.ndarray | None = None

    def get_debug_info(self) -> Dict[str, Any]:
        info = {var: self.params.get(var) for var in self.plugin_debug_vars}
        info["n_features"] = len(self.params.get("feature_columns") or [])
        return info

    def add_debug_info(self, debug_info: Dict[str, Any]) -> None:
        debug_info.update(self.get_debug_info())

    def _matrix(self, data: Any, cols: List[str]) -> np.ndarray:
        key = (id(data), tuple(cols))
        if self._cache_key == key and self._cache_matrix is not None:
            return self._cache_matrix
        mat = np.stack([_column(data, c) for c in cols], axis=1)
        self._cache_key = key
        self._cache_matrix = mat
        return mat

    def make_observation(
        self,
        *,
        data: Any,
        step: int,
        bridge_state: Dict[str, Any],
        config: Dict[str, Any],
    ) -> Dict[str, np.ndarray]:
        cols = list(config.get("feature_columns") or self.params["feature_columns"] or [])
        if not cols:
            raise ValueError(
                "feature_window_preprocessor requires non-empty 'feature_columns'."
            )
        missing = [c for c in cols if not _has_column(data, c)]
        if missing:
            raise ValueError(
                "feature_window_preprocessor: configured feature_columns "
                f"missing from data: {missing[:5]}{'...' if len(missing) > 5 else ''}"
            )
        binary_cols = set(
            config.get("feature_binary_columns")
            or self.params["feature_binary_columns"]
            or []
        )
        binary_mask = np.array([c in binary_cols for c in cols], dtype=bool)
        window_size = int(config.get("window_size", self.params["window_size"]))
        scale_mode = str(
            config.get("feature_scaling", self.params["feature_scaling"])
        ).lower()
        if scale_mode not in _VALID_SCALINGS:
            raise ValueError(
                f"feature_scaling must be one of {_VALID_SCALINGS}; got {scale_mode!r}"
            )
        scale_window = int(
            config.get("feature_scaling_window", self.params["feature_scaling_window"])
        )
        clip = float(config.get("feature_clip", self.params["feature_clip"]))

        values = self._matrix(data, cols)
        n_rows, n_features = values.shape
        left = max(0, step - window_size)
        win = values[left:step] if step > 0 else values[:0]
        if win.shape[0] < window_size:
            pad_row = win[0] if win.shape[0] else (
                values[0] if n_rows else np.zeros(n_features)
            )
            pad = np.tile(pad_row, (window_size - win.shape[0], 1))
            win = np.concatenate([pad, win], axis=0)

        if scale_mode == "rolling_zscore":
            history = values[max(0, step - scale_window):step]
        elif scale_mode == "expanding_zscore":
            history = values[:step]
        else:
            history = np.empty((0, n_features))

        if scale_mode == "none":
            scaled = win.astype(np.float32)
        elif history.shape[0] < 2:
            scaled = np.zeros_like(win, dtype=np.float32)
        else:
            mean = history.mean(axis=0)
            std = history.std(axis=0)
            std = np.where(std < 1e-8, 1.0, std)
            scaled = ((win - mean) / std).astype(np.float32)
        if binary_mask.any():
            scaled[:, binary_mask] = win[:, binary_mask].astype(np.float32)
        if clip and clip > 0:
            np.clip(scaled, -clip, clip, out=scaled)
        scaled = np.nan_to_num(scaled, nan=0.0, posinf=clip, neginf=-clip)

        obs: Dict[str, np.ndarray] = {"features": scaled.astype(np.float32)}
        include_price = bool(
            config.get("include_price_window", self.params["include_price_window"])
        )
        window_last = 0.0
        if include_price:
            price_col = config.get("price_column", self.params["price_column"])
            pw = _price_window(_column(data, price_col), step, window_size)
            obs["prices"] = pw.astype(np.float32)
            obs["returns"] = np.diff(pw, prepend=pw[0]).astype(np.float32)
            window_last = float(pw[-1])
        if bool(config.get("include_agent_state", self.params["include_agent_state"])):
            _agent_state(obs, bridge_state, config, window_last, include_price)
        return obs
