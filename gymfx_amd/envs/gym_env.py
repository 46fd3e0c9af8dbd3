"""Single-env Gymnasium-compatible wrapper over the vectorized engine.

API parity with the reference GymFxEnv (/root/reference/app/env.py:93-328):
same constructor signature (config + six plugins), same Dict observation
contract, same info keys, same summary() shape.  The difference is under the
hood: instead of a backtrader Cerebro on a worker thread synchronized with
two events (app/bt_bridge.py:30-83), this wraps a VecFxEnv with N=1 whose
step is a single vectorized transition (CPU torch here; the same state
machine runs as one fused HIP kernel at N=4096+ on MI355X).
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import spaces
from ..calendar import compute_fx_calendar_features, resolve_broker_metadata
from .state import EXEC_COUNTERS
from .vec_env import VecFxEnv
from .params import EnvParams


def build_base_observation_space(config: Dict[str, Any], *, window_size: int) -> spaces.Dict:
    """Observation-space contract of /root/reference/app/env.py:31-90."""
    feature_columns = list(config.get("feature_columns") or [])
    include_prices = bool(config.get("include_price_window", not feature_columns))
    include_agent_state = bool(config.get("include_agent_state", True))
    obs: Dict[str, Any] = {}
    if feature_columns:
        obs["features"] = spaces.Box(
            low=-np.inf, high=np.inf, shape=(window_size, len(feature_columns)),
            dtype=np.float32,
        )
    if include_prices:
        obs["prices"] = spaces.Box(-np.inf, np.inf, shape=(window_size,), dtype=np.float32)
        obs["returns"] = spaces.Box(-np.inf, np.inf, shape=(window_size,), dtype=np.float32)
    if include_agent_state:
        obs["position"] = spaces.Box(-1.0, 1.0, shape=(1,), dtype=np.float32)
        obs["equity_norm"] = spaces.Box(-np.inf, np.inf, shape=(1,), dtype=np.float32)
        obs["unrealized_pnl_norm"] = spaces.Box(-np.inf, np.inf, shape=(1,), dtype=np.float32)
        obs["steps_remaining_norm"] = spaces.Box(0.0, 1.0, shape=(1,), dtype=np.float32)
    if not obs:
        raise ValueError("preprocessor observation contract emits no observation blocks")
    return spaces.Dict(obs)


_FC_KEYS = (
    "bars_to_force_close",
    "hours_to_force_close",
    "is_force_close_zone",
    "is_monday_entry_window",
)

_CAL_KEYS = (
    "hours_to_fx_daily_break",
    "bars_to_fx_daily_break",
    "hours_to_friday_close",
    "bars_to_friday_close",
    "is_friday_risk_reduction_window",
    "is_no_new_position_window",
    "is_force_flat_window",
    "is_broker_daily_break_near",
    "broker_market_open",
)


class GymFxEnv(spaces.Env):
    """Gymnasium-style FX trading env backed by the vectorized engine."""

    metadata = {"render_modes": []}

    def __init__(
        self,
        config: Dict[str, Any],
        data_feed_plugin,
        broker_plugin,
        strategy_plugin,
        preprocessor_plugin,
        reward_plugin,
        metrics_plugin,
    ):
        self.config = dict(config)
        self.data_feed_plugin = data_feed_plugin
        self.broker_plugin = broker_plugin
        self.strategy_plugin = strategy_plugin
        self.preprocessor_plugin = preprocessor_plugin
        self.reward_plugin = reward_plugin
        self.metrics_plugin = metrics_plugin

        self.initial_cash = float(self.config.get("initial_cash", 10000.0))
        self.position_size = float(self.config.get("position_size", 1.0))
        self.window_size = int(self.config.get("window_size", 32))
        self.price_column = self.config.get("price_column", "CLOSE")
        self.min_equity = float(
            self.config.get("min_equity") or self.initial_cash * 0.01
        )

        self.market_data = self.data_feed_plugin.load_data(self.config)
        if self.market_data is None or len(self.market_data) < self.window_size + 2:
            raise ValueError("input data is empty or too short for the configured window")
        if not self.market_data.has_column(self.price_column):
            raise ValueError(f"price_column '{self.price_column}' not found in data")
        self.total_bars = len(self.market_data)

        vec_config = dict(self.config)
        vec_config["n_envs"] = 1
        vec_config["autoreset"] = False
        vec_config.setdefault("device", "cpu")
        if hasattr(self.broker_plugin, "broker_params"):
            bp = self.broker_plugin.broker_params(self.config)
            vec_config["initial_cash"] = bp["initial_cash"]
            vec_config["commission"] = bp["commission"]
            vec_config["slippage"] = bp["slippage"]
            vec_config["leverage"] = bp["leverage"]
        self.vec = VecFxEnv(vec_config, self.market_data, use_native=False)
        self.params: EnvParams = self.vec.params

        # --- spaces -------------------------------------------------------
        self.action_space_mode = str(self.config.get("action_space_mode", "discrete")).lower()
        if self.action_space_mode == "continuous":
            self.action_space = spaces.Box(low=-1.0, high=1.0, shape=(1,), dtype=np.float32)
            self.continuous_action_threshold = float(
                self.config.get("continuous_action_threshold", 0.33)
            )
        else:
            self.action_space = spaces.Discrete(3)
            self.continuous_action_threshold = None
        self.observation_space = build_base_observation_space(
            self.config, window_size=self.window_size
        )
        self.stage_b_force_close_obs = self.params.stage_b_force_close_obs
        if self.stage_b_force_close_obs:
            extra = {
                "bars_to_force_close": spaces.Box(0.0, np.inf, shape=(1,), dtype=np.float32),
                "hours_to_force_close": spaces.Box(0.0, np.inf, shape=(1,), dtype=np.float32),
                "is_force_close_zone": spaces.Box(0.0, 1.0, shape=(1,), dtype=np.float32),
                "is_monday_entry_window": spaces.Box(0.0, 1.0, shape=(1,), dtype=np.float32),
            }
            self.observation_space = spaces.Dict({**dict(self.observation_space.spaces), **extra})
        self.oanda_fx_calendar_obs = self.params.oanda_fx_calendar_obs
        if self.oanda_fx_calendar_obs:
            extra = {k: spaces.Box(0.0, np.inf, shape=(1,), dtype=np.float32) for k in _CAL_KEYS}
            extra["margin_closeout_percent"] = spaces.Box(0.0, np.inf, shape=(1,), dtype=np.float32)
            extra["margin_available_norm"] = spaces.Box(0.0, np.inf, shape=(1,), dtype=np.float32)
            self.observation_space = spaces.Dict({**dict(self.observation_space.spaces), **extra})

        self._np_random = np.random.default_rng()
        self._last_info_extras: Dict[str, Any] = {}
        self._was_reset = False
        self._last_raw_action_value = None
        self._last_coerced_action = None
        # bracket/force-close audit trail (parity:
        # /root/reference/strategy_plugins/direct_atr_sltp.py:40-50 —
        # GYMFX_BRACKET_AUDIT=<path> enables append-only JSONL records)
        import os as _os
        self._audit_path = _os.environ.get("GYMFX_BRACKET_AUDIT") or \
            config.get("bracket_audit_file")
        self._prev_audit_state: Dict[str, Any] = {}
        self._ec_sl = EXEC_COUNTERS.index("bracket_sl_fills")
        self._ec_tp = EXEC_COUNTERS.index("bracket_tp_fills")
        self._ec_fc = EXEC_COUNTERS.index("session_force_closes")

    # ------------------------------------------------------------------
    def reset(self, *, seed: Optional[int] = None, options: Optional[Dict[str, Any]] = None):
        if seed is not None:
            self._np_random = np.random.default_rng(seed)
        self.vec.reset(seed=seed)
        self._was_reset = True
        self._last_info_extras = {}
        return self._make_observation(), self._make_info()

    def step(self, action):
        if not self._was_reset:
            raise RuntimeError("Call reset() before step().")
        raw = self._raw_action_value(action)
        if self.action_space_mode == "continuous":
            act_t = torch.tensor([raw], dtype=torch.float32)
        else:
            try:
                act_t = torch.tensor([int(action)], dtype=torch.int64)
            except (TypeError, ValueError):
                act_t = torch.tensor([0], dtype=torch.int64)
        if self.vec.params.event_context_execution_overlay:
            self._last_info_extras = self._event_context_info(act_t)
        if self._audit_path:
            st = self.vec.st
            self._prev_audit_state = {
                "br_active": bool(st.br_active[0].item()),
                "pos": float(st.pos[0].item()),
                "sl_fills": int(st.exec_diag[0, self._ec_sl].item()),
                "tp_fills": int(st.exec_diag[0, self._ec_tp].item()),
                "session_fc": int(st.exec_diag[0, self._ec_fc].item()),
            }
        out = self.vec.step(act_t)
        if self._audit_path:
            self._emit_audit()
        self._last_raw_action_value = raw
        self._last_coerced_action = int(out["coerced_action"][0].item())
        reward = float(out["reward"][0].item())
        base_reward = float(out["base_reward"][0].item())
        penalty = float(out["force_close_reward_penalty"][0].item())
        terminated = bool(out["terminated"][0].item())
        bs = self.vec.bridge_state(0)
        obs = self._make_observation()
        info = self._make_info()
        info.update(
            reward=reward,
            base_reward=base_reward,
            force_close_reward_penalty=penalty,
            pnl=bs["equity"] - bs["prev_equity"],
            trade_cost=bs["last_trade_cost"],
        )
        return obs, reward, terminated, False, info

    def _emit_audit(self) -> None:
        """Append JSONL audit records for bracket arms/fills and session
        force-closes, derived from per-step state deltas."""
        import json as _json

        st = self.vec.st
        prev = self._prev_audit_state
        recs = []
        bar = int(st.cursor[0].item())
        now_active = bool(st.br_active[0].item())
        if now_active and not prev.get("br_active"):
            recs.append({
                "event": "bracket_armed", "bar_index": bar,
                "sl": float(st.br_sl[0].item()),
                "tp": float(st.br_tp[0].item()),
                "position": float(st.pos[0].item()),
            })
        sl_d = int(st.exec_diag[0, self._ec_sl].item()) - prev.get("sl_fills", 0)
        tp_d = int(st.exec_diag[0, self._ec_tp].item()) - prev.get("tp_fills", 0)
        fc_d = int(st.exec_diag[0, self._ec_fc].item()) - prev.get("session_fc", 0)
        if sl_d > 0:
            recs.append({"event": "bracket_sl_fill", "bar_index": bar,
                         "equity": float(st.equity[0].item())})
        if tp_d > 0:
            recs.append({"event": "bracket_tp_fill", "bar_index": bar,
                         "equity": float(st.equity[0].item())})
        if fc_d > 0:
            recs.append({"event": "session_force_close", "bar_index": bar})
        if recs:
            with open(self._audit_path, "a", encoding="utf-8") as fh:
                for r in recs:
                    fh.write(_json.dumps(r) + "\n")

    def close(self):
        pass

    def render(self):  # pragma: no cover
        return None

    # ------------------------------------------------------------------
    def _raw_action_value(self, action) -> float:
        try:
            return float(np.asarray(action).reshape(-1)[0])
        except Exception:
            try:
                return float(action)
            except Exception:
                return 0.0

    def _make_observation(self) -> Dict[str, np.ndarray]:
        bs = self.vec.bridge_state(0)
        step_idx = max(0, min(bs["bar_index"], self.total_bars))
        obs = self.preprocessor_plugin.make_observation(
            data=self.market_data,
            step=step_idx,
            bridge_state={
                "position": bs["position"],
                "equity": bs["equity"],
                "initial_cash": self.initial_cash,
                "price": bs["price"],
                "bar_index": bs["bar_index"],
                "total_bars": self.total_bars,
            },
            config=self.config,
        )
        if self.stage_b_force_close_obs:
            obs = dict(obs)
            fc = self._force_close_features(step_idx)
            for k in _FC_KEYS:
                obs[k] = np.array([fc[k]], dtype=np.float32)
        if self.oanda_fx_calendar_obs:
            obs = dict(obs)
            cal = self._oanda_calendar_features(step_idx)
            for k in _CAL_KEYS:
                obs[k] = np.array([cal[k]], dtype=np.float32)
            obs["margin_closeout_percent"] = np.array([0.0], dtype=np.float32)
            obs["margin_available_norm"] = np.array(
                [bs["equity"] / (self.initial_cash or 1.0)], dtype=np.float32
            )
        return obs

    def _force_close_features(self, step_idx: int) -> Dict[str, float]:
        mt = self.vec.mt
        if mt.force_close is None or mt.timestamps is None:
            return {k: 0.0 for k in _FC_KEYS}
        row = max(0, min(step_idx, self.total_bars - 1))
        fc = mt.force_close[row]
        return {
            "bars_to_force_close": float(fc[0]),
            "hours_to_force_close": float(fc[1]),
            "is_force_close_zone": float(fc[2]),
            "is_monday_entry_window": float(fc[3]),
        }

    def _oanda_calendar_features(self, step_idx: int) -> Dict[str, float]:
        md = self.market_data
        if md.timestamps is None:
            ts = None
        else:
            row = max(0, min(step_idx, self.total_bars - 1))
            ts = int(md.timestamps[row])
        tf_h = float(md.timeframe_hours() or 1.0) or 1.0
        return compute_fx_calendar_features(ts, timeframe_hours=tf_h)

    def _event_context_info(self, act_t) -> Dict[str, Any]:
        """Per-step event-context info block (env.py:383-440 field set);
        the authoritative action rewrite happens inside the fused kernel —
        this mirrors the decision for observability."""
        p = self.vec.params
        mt = self.vec.mt
        st = self.vec.st
        row = int(min(int(st.cursor[0].item()), self.total_bars - 1))
        no_trade = float(mt.ev_no_trade[row].item()) if mt.ev_no_trade is not None else 0.0
        spread_m = float(mt.ev_spread_mult[row].item()) if mt.ev_spread_mult is not None else 1.0
        slip_m = float(mt.ev_slip_mult[row].item()) if mt.ev_slip_mult is not None else 1.0
        active = no_trade >= p.event_context_no_trade_threshold
        position = int(torch.sign(st.pos[0]).item())
        before = int(act_t.reshape(-1)[0].item()) if act_t.dtype == torch.int64 else 0
        after = before
        blocked = forced = False
        if active:
            if p.event_context_force_flat and position != 0:
                after, forced = 3, True
            elif (p.event_context_block_new_entries and position == 0
                  and before in (1, 2)):
                after, blocked = 0, True
        return {
            "event_context_no_trade_value": no_trade,
            "event_context_no_trade_active": 1.0 if active else 0.0,
            "event_context_spread_stress_multiplier": spread_m,
            "event_context_slippage_stress_multiplier": slip_m,
            "event_context_execution_overlay": True,
            "event_context_action_before_overlay": before,
            "event_context_action_after_overlay": after,
            "event_context_action_overridden": bool(after != before),
            "event_context_blocked_entry": blocked,
            "event_context_forced_flat": forced,
            "event_context_position_before_overlay": position,
        }

    def _make_info(self) -> Dict[str, Any]:
        bs = self.vec.bridge_state(0)
        info: Dict[str, Any] = {
            "equity": bs["equity"],
            "position": bs["position"],
            "price": bs["price"],
            "bar_index": bs["bar_index"],
            "total_bars": self.total_bars,
            "trades": bs["trade_count"],
            "commission_paid": bs["commission_paid"],
            "raw_action_value": self._last_raw_action_value,
            "coerced_action": self._last_coerced_action,
            "action_diagnostics": self.vec.action_diagnostics(0),
            "execution_diagnostics": self.vec.execution_diagnostics(0),
        }
        info.update(self._last_info_extras)
        if self.stage_b_force_close_obs:
            step_idx = max(0, min(bs["bar_index"], self.total_bars))
            info.update(self._force_close_features(step_idx))
        if self.oanda_fx_calendar_obs:
            step_idx = max(0, min(bs["bar_index"], self.total_bars))
            info.update(self._oanda_calendar_features(step_idx))
            info["margin_closeout_percent"] = 0.0
            info["margin_available_norm"] = bs["equity"] / (self.initial_cash or 1.0)
            for k, v in resolve_broker_metadata(self.config).items():
                if v is not None:
                    info[k] = v
        return info

    def summary(self) -> Dict[str, Any]:
        bs = self.vec.bridge_state(0)
        summary = self.metrics_plugin.summarize(
            initial_cash=self.initial_cash,
            final_equity=bs["equity"],
            analyzers=self.vec.analyzers(0),
            config=self.config,
        )
        summary["action_diagnostics"] = self.vec.action_diagnostics(0)
        summary["execution_diagnostics"] = self.vec.execution_diagnostics(0)
        summary["event_context_diagnostics"] = dict(self._last_info_extras)
        return summary
