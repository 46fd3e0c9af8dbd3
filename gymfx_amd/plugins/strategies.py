"""Strategy plugins.

The order-placement logic of all three built-in strategies runs INSIDE the
fused step kernel (params.strategy_id dispatch; semantics from
/root/reference/app/bt_bridge.py:203-237, strategy_plugins/direct_fixed_sltp.py
and direct_atr_sltp.py).  The classes here carry the parameter contracts,
the GA hparam schema, and — for the default strategy — the diagnostic
action drivers (random / buy_hold / flat / replay), including a vectorized
``decide_actions_batch`` for the VecFxEnv runner.
"""
from __future__ import annotations

import csv
import random
from typing import Any, Dict, List, Optional

import numpy as np

from .base import PluginBase


class DefaultStrategy(PluginBase):
    plugin_params = {
        "driver_mode": "buy_hold",  # buy_hold | random | flat | replay
        "replay_actions_file": None,
        "seed": None,
    }

    def __init__(self, config: Dict[str, Any] | None = None):
        self._replay_actions: List[int] = []
        self._rng = random.Random()
        self._np_rng = np.random.default_rng()
        super().__init__(config)

    def set_params(self, **kwargs: Any) -> None:
        super().set_params(**kwargs)
        seed = self.params.get("seed")
        if seed is not None:
            self._rng = random.Random(seed)
            self._np_rng = np.random.default_rng(int(seed))
        replay = self.params.get("replay_actions_file")
        if replay:
            with open(replay, "r", encoding="utf-8") as fh:
                self._replay_actions = [
                    int(row.get("action", 0)) for row in csv.DictReader(fh)
                ]

    # single-env diagnostic driver (default_strategy.py:44-54)
    def decide_action(self, obs, info, step: int) -> int:
        mode = self.params.get("driver_mode", "buy_hold")
        if mode == "random":
            return self._rng.choice([0, 1, 2])
        if mode == "flat":
            return 0
        if mode == "replay":
            if step < len(self._replay_actions):
                return self._replay_actions[step]
            return 0
        return 1 if step == 0 else 0

    # vectorized driver (no reference counterpart — N-wide)
    def decide_actions_batch(self, n_envs: int, step: int) -> np.ndarray:
        mode = self.params.get("driver_mode", "buy_hold")
        if mode == "random":
            return self._np_rng.integers(0, 3, size=n_envs)
        if mode == "flat":
            return np.zeros(n_envs, dtype=np.int64)
        if mode == "replay":
            a = self._replay_actions[step] if step < len(self._replay_actions) else 0
            return np.full(n_envs, a, dtype=np.int64)
        return np.full(n_envs, 1 if step == 0 else 0, dtype=np.int64)


class DirectFixedSLTP(PluginBase):
    """Bracket orders with fixed-pip SL/TP (direct_fixed_sltp.py:51-77)."""

    plugin_params = {
        "sl_pips": 20.0,
        "tp_pips": 40.0,
        "pip_size": 0.0001,
        "position_size": 1.0,
    }

    def __init__(self, config: Dict[str, Any] | None = None):
        super().__init__(config)
        # the runner's driver_mode drives actions even when a bracket
        # strategy converts them to orders (reference app/main.py:57-66:
        # the driver loop is independent of the strategy plugin)
        self._driver = DefaultStrategy(config)

    def set_params(self, **kwargs: Any) -> None:
        for k, v in kwargs.items():
            if k in self.plugin_params:
                self.params[k] = v

    def decide_action(self, obs, info, step: int) -> int:
        return self._driver.decide_action(obs, info, step)


class DirectAtrSLTP(PluginBase):
    """ATR-sized brackets with warmup gating, rel_volume/leverage sizing,
    risk-mode SL/TP shrink, margin-aware SL cap, fraction clamps and a
    weekend session filter (direct_atr_sltp.py)."""

    plugin_params = {
        "atr_period": 14,
        "k_sl": 2.0,
        "k_tp": 3.0,
        "position_size": 1.0,
        "rel_volume": None,
        "leverage": 1.0,
        "min_order_volume": 0.0,
        "max_order_volume": 1e12,
        "size_mode": "fx_units",
        "min_sltp_frac": 0.001,
        "max_sltp_frac": 0.20,
        "sltp_risk_mode": "fixed_atr",
        "baseline_rel_volume": 0.05,
        "max_risk_rel_volume": 0.50,
        "rel_volume_sl_shrink_alpha": 0.35,
        "rel_volume_tp_shrink_alpha": 0.20,
        "min_k_sl": 1.0,
        "min_reward_risk_ratio": 1.0,
        "max_planned_loss_fraction": None,
        "session_filter": False,
        "entry_dow_start": 0,
        "entry_hour_start": 12,
        "force_close_dow": 4,
        "force_close_hour": 20,
    }

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        # same driver delegation as DirectFixedSLTP (app/main.py:57-66)
        self._driver = DefaultStrategy(config)

    def set_params(self, **kwargs: Any) -> None:
        for k, v in kwargs.items():
            if k in self.plugin_params:
                self.params[k] = v

    def decide_action(self, obs, info, step: int) -> int:
        return self._driver.decide_action(obs, info, step)

    def effective_sltp_multiples(self, config: Optional[Dict[str, Any]] = None):
        """Scalar oracle for the risk-mode shrink math
        (direct_atr_sltp.py:263-289); the kernel computes the same."""
        p = self._resolve(config or {})
        k_sl = max(0.0, float(p["k_sl"]))
        k_tp = max(0.0, float(p["k_tp"]))
        mode = str(p.get("sltp_risk_mode", "fixed_atr")).strip().lower()
        if mode not in {"rel_volume_aware_atr", "margin_aware_atr"}:
            return k_sl, k_tp
        try:
            rel = max(0.0, float(p.get("rel_volume") or 0.0))
            baseline = max(0.0, float(p.get("baseline_rel_volume", 0.05)))
            max_rel = max(baseline + 1e-12, float(p.get("max_risk_rel_volume", 0.50)))
            sl_alpha = min(max(float(p.get("rel_volume_sl_shrink_alpha", 0.35)), 0.0), 0.95)
            tp_alpha = min(max(float(p.get("rel_volume_tp_shrink_alpha", 0.20)), 0.0), 0.95)
            min_k_sl = max(0.0, float(p.get("min_k_sl", 1.0)))
            min_rr = max(0.0, float(p.get("min_reward_risk_ratio", 1.0)))
        except (TypeError, ValueError):
            return k_sl, max(k_tp, k_sl)
        if rel <= baseline:
            k_sl_eff, k_tp_eff = k_sl, k_tp
        else:
            prog = min(1.0, max(0.0, (rel - baseline) / (max_rel - baseline)))
            k_sl_eff = max(min_k_sl, k_sl * (1.0 - sl_alpha * prog))
            k_tp_eff = k_tp * (1.0 - tp_alpha * prog)
        k_tp_eff = max(k_tp_eff, k_sl_eff * min_rr)
        return k_sl_eff, k_tp_eff

    # GA-tunable hyperparameters (direct_atr_sltp.py:344-350)
    def hparam_schema(self):
        return [
            ("atr_period", 7, 30, "int"),
            ("k_sl", 1.0, 4.0, "float"),
            ("k_tp", 1.5, 6.0, "float"),
        ]
