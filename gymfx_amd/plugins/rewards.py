"""Reward plugins — single-env scalar implementations (reference contract
``compute_reward(prev_equity, new_equity, step, config)``) mirroring
/root/reference/reward_plugins/{pnl,sharpe,dd_penalized}_reward.py.

Inside the vectorized engine the same math runs N-wide in the fused step
kernel (rewards dispatched by params.reward_id); these classes are the
single-env contract + the oracle the kernels are tested against.
"""
from __future__ import annotations

import math
from collections import deque
from typing import Any, Deque, Dict

from .base import PluginBase


class PnlReward(PluginBase):
    plugin_params = {
        "reward_scale": 1.0,
        "initial_cash": 10000.0,
    }

    def compute_reward(self, *, prev_equity, new_equity, step, config) -> float:
        initial_cash = float(config.get("initial_cash", self.params["initial_cash"])) or 1.0
        scale = float(config.get("reward_scale", self.params["reward_scale"]))
        return (float(new_equity) - float(prev_equity)) / initial_cash * scale


class SharpeReward(PluginBase):
    """Annualized rolling Sharpe of per-step normalized returns; warmup -> 0;
    per-env reset detected via non-monotonic step."""

    plugin_params = {
        "window": 64,
        "annualization_factor": 252.0,
        "initial_cash": 10000.0,
    }

    def __init__(self, config: Dict[str, Any] | None = None):
        self._buffer: Deque[float] = deque(maxlen=int(self.plugin_params["window"]))
        self._last_step = -1
        super().__init__(config)

    def set_params(self, **kwargs: Any) -> None:
        super().set_params(**kwargs)
        self._buffer = deque(maxlen=int(self.params["window"]))
        self._last_step = -1

    def compute_reward(self, *, prev_equity, new_equity, step, config) -> float:
        if step <= self._last_step:
            self._buffer.clear()
        self._last_step = int(step)
        initial_cash = float(config.get("initial_cash", self.params["initial_cash"])) or 1.0
        r = (float(new_equity) - float(prev_equity)) / initial_cash
        self._buffer.append(r)
        if len(self._buffer) < 2:
            return 0.0
        n = len(self._buffer)
        mean = sum(self._buffer) / n
        var = sum((x - mean) ** 2 for x in self._buffer) / (n - 1)
        std = math.sqrt(var)
        if std <= 0:
            return 0.0
        ann = float(config.get("annualization_factor", self.params["annualization_factor"]))
        return (mean / std) * math.sqrt(ann)


class DdPenalizedReward(PluginBase):
    """pnl_norm - lambda * drawdown-from-peak, with running-peak tracking."""

    plugin_params = {
        "penalty_lambda": 1.0,
        "initial_cash": 10000.0,
    }

    def __init__(self, config: Dict[str, Any] | None = None):
        self._peak = 0.0
        self._last_step = -1
        super().__init__(config)

    def set_params(self, **kwargs: Any) -> None:
        super().set_params(**kwargs)
        self._peak = 0.0
        self._last_step = -1

    def compute_reward(self, *, prev_equity, new_equity, step, config) -> float:
        if step <= self._last_step:
            self._peak = 0.0
        self._last_step = int(step)
        self._peak = max(self._peak, float(new_equity), float(prev_equity))
        initial_cash = float(config.get("initial_cash", self.params["initial_cash"])) or 1.0
        pnl_norm = (float(new_equity) - float(prev_equity)) / initial_cash
        dd_norm = (self._peak - float(new_equity)) / initial_cash if self._peak > 0 else 0.0
        lam = float(config.get("penalty_lambda", self.params["penalty_lambda"]))
        return pnl_norm - lam * dd_norm
