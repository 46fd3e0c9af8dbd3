"""Resolved, validated parameter pack for the vectorized env.

Plugin *names* from the config are resolved here into kernel enum ids +
flat float/int parameter packs: the reference dispatches plugin behavior
through Python objects per step (/root/reference/app/env.py:279-328); the
MI355X design bakes the behavioral choice into the fused step kernel's
switch so per-step cost is zero.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

# --- enums shared with the HIP kernel (keep in sync with ops/csrc/env_common.h)
REWARD_PNL = 0
REWARD_SHARPE = 1
REWARD_DD = 2

STRATEGY_DIRECT = 0       # default target-direction flow (bt_bridge.py:203-237)
STRATEGY_FIXED_SLTP = 1   # direct_fixed_sltp bracket orders
STRATEGY_ATR_SLTP = 2     # direct_atr_sltp ATR-sized brackets

PREP_DEFAULT = 0          # price window + returns + agent state
PREP_FEATURE_WINDOW = 1   # (W, F) z-scored feature tensor

SIZE_MODE_FX_UNITS = 0
SIZE_MODE_NOTIONAL = 1

RISK_FIXED_ATR = 0
RISK_REL_VOLUME_AWARE = 1
RISK_MARGIN_AWARE = 2

REWARD_NAMES = {
    "pnl_reward": REWARD_PNL,
    "sharpe_reward": REWARD_SHARPE,
    "dd_penalized_reward": REWARD_DD,
}
STRATEGY_NAMES = {
    "default_strategy": STRATEGY_DIRECT,
    "direct_fixed_sltp": STRATEGY_FIXED_SLTP,
    "direct_atr_sltp": STRATEGY_ATR_SLTP,
}
PREP_NAMES = {
    "default_preprocessor": PREP_DEFAULT,
    "feature_window_preprocessor": PREP_FEATURE_WINDOW,
}
# execution-realism policies (ids == ops/csrc/env_common.h enums)
_COLLISION_POLICIES = {"worst_case": 0, "ohlc": 1, "adaptive": 2}
_LIMIT_POLICIES = {"touch": 0, "cross": 1, "conservative": 2}
_MARGIN_MODELS = {"leveraged": 0, "standard": 1}
RISK_MODES = {
    "fixed_atr": RISK_FIXED_ATR,
    "rel_volume_aware_atr": RISK_REL_VOLUME_AWARE,
    "margin_aware_atr": RISK_MARGIN_AWARE,
}


def _f(config: Dict[str, Any], key: str, default: float) -> float:
    v = config.get(key, default)
    return default if v is None else float(v)


def _i(config: Dict[str, Any], key: str, default: int) -> int:
    v = config.get(key, default)
    return default if v is None else int(v)


def _b(config: Dict[str, Any], key: str, default: bool) -> bool:
    v = config.get(key, default)
    if v is None:
        return default
    if isinstance(v, str):  # bool("false") is True — route through the
        from ..config.merger import convert_type  # config layer's coercion

        v = convert_type(v)
    return bool(v)


@dataclass
class EnvParams:
    # scale
    n_envs: int = 1
    window_size: int = 32
    # account / broker
    initial_cash: float = 10000.0
    position_size: float = 1.0
    commission: float = 0.0
    slippage: float = 0.0
    leverage: float = 1.0
    min_equity: float = 100.0
    # action space
    action_space_mode: str = "discrete"
    continuous_action_threshold: float = 0.33
    # plugin dispatch
    reward_id: int = REWARD_PNL
    strategy_id: int = STRATEGY_DIRECT
    prep_id: int = PREP_DEFAULT
    # reward params
    reward_scale: float = 1.0
    sharpe_window: int = 64
    annualization_factor: float = 252.0
    penalty_lambda: float = 1.0
    # preprocessor params
    price_column: str = "CLOSE"
    feature_columns: List[str] = field(default_factory=list)
    feature_binary_columns: List[str] = field(default_factory=list)
    feature_scaling: str = "rolling_zscore"   # none|rolling_zscore|expanding_zscore
    feature_scaling_window: int = 256
    feature_clip: float = 10.0
    include_price_window: bool = True
    include_agent_state: bool = True
    # fixed-sltp strategy
    sl_pips: float = 20.0
    tp_pips: float = 40.0
    pip_size: float = 0.0001
    # atr strategy
    atr_period: int = 14
    k_sl: float = 2.0
    k_tp: float = 3.0
    rel_volume: Optional[float] = None
    min_order_volume: float = 0.0
    max_order_volume: float = 1e12
    size_mode: int = SIZE_MODE_FX_UNITS
    min_sltp_frac: Optional[float] = 0.001
    max_sltp_frac: Optional[float] = 0.20
    sltp_risk_mode: int = RISK_FIXED_ATR
    baseline_rel_volume: float = 0.05
    max_risk_rel_volume: float = 0.50
    rel_volume_sl_shrink_alpha: float = 0.35
    rel_volume_tp_shrink_alpha: float = 0.20
    min_k_sl: float = 1.0
    min_reward_risk_ratio: float = 1.0
    max_planned_loss_fraction: Optional[float] = None
    session_filter: bool = False
    entry_dow_start: int = 0
    entry_hour_start: int = 12
    force_close_dow: int = 4
    force_close_hour: int = 20
    # stage-B force close obs / penalty
    stage_b_force_close_obs: bool = False
    force_close_window_hours: int = 4
    monday_entry_window_hours: int = 4
    stage_b_force_close_reward_penalty: bool = False
    force_close_exposure_penalty_coef: float = 0.0
    force_close_exposure_penalty_window_hours: float = 4.0
    # oanda calendar obs
    oanda_fx_calendar_obs: bool = False
    financing_enabled: bool = False
    rollover_hour_utc: int = 22
    enforce_margin_preflight: bool = False
    timeframe_hours: float = 0.0
    # execution-realism tier (execution_cost_profile.v1 semantics)
    intrabar_collision_policy: int = 0   # 0 worst_case, 1 ohlc, 2 adaptive
    limit_fill_policy: int = 0           # 0 touch, 1 cross, 2 conservative
    latency_bars: int = 0                # floor(latency_ms / bar duration)
    margin_model: int = 0                # 0 leveraged, 1 standard
    margin_init_rate: float = 0.03       # used by margin_model=standard
    # event-context overlay
    event_context_execution_overlay: bool = False
    event_context_no_trade_column: str = "event_no_trade_window_active"
    event_context_no_trade_threshold: float = 0.5
    event_context_block_new_entries: bool = True
    event_context_force_flat: bool = False
    event_context_spread_stress_column: str = "event_spread_stress_multiplier"
    event_context_slippage_stress_column: str = "event_slippage_stress_multiplier"
    # runtime
    autoreset: bool = False
    env_start_mode: str = "zero"
    seed: Optional[int] = None

    # -- derived observation layout (filled by finalize) -------------------
    n_features: int = 0
    obs_blocks: List[Tuple[str, int]] = field(default_factory=list)
    obs_dim: int = 0

    @classmethod
    def from_config(cls, config: Dict[str, Any], *, timeframe_hours: float = 0.0) -> "EnvParams":
        p = cls()
        p.n_envs = _i(config, "n_envs", 1)
        p.window_size = _i(config, "window_size", 32)
        p.initial_cash = _f(config, "initial_cash", 10000.0)
        p.position_size = _f(config, "position_size", 1.0)
        p.commission = _f(config, "commission", 0.0)
        slip = config.get("slippage_perc", config.get("slippage", 0.0))
        p.slippage = float(slip or 0.0)
        p.leverage = max(_f(config, "leverage", 1.0), 1e-12)
        p.min_equity = float(
            config.get("min_equity") if config.get("min_equity") is not None
            else p.initial_cash * 0.01
        )
        p.action_space_mode = str(config.get("action_space_mode", "discrete")).lower()
        p.continuous_action_threshold = _f(config, "continuous_action_threshold", 0.33)

        p.reward_id = REWARD_NAMES.get(str(config.get("reward_plugin", "pnl_reward")), REWARD_PNL)
        p.strategy_id = STRATEGY_NAMES.get(
            str(config.get("strategy_plugin", "default_strategy")), STRATEGY_DIRECT
        )
        p.prep_id = PREP_NAMES.get(
            str(config.get("preprocessor_plugin", "default_preprocessor")), PREP_DEFAULT
        )

        p.reward_scale = _f(config, "reward_scale", 1.0)
        p.sharpe_window = _i(config, "window", 64) if p.reward_id == REWARD_SHARPE else 64
        p.annualization_factor = _f(config, "annualization_factor", 252.0)
        p.penalty_lambda = _f(config, "penalty_lambda", 1.0)

        p.price_column = str(config.get("price_column", "CLOSE"))
        p.feature_columns = list(config.get("feature_columns") or [])
        p.feature_binary_columns = list(config.get("feature_binary_columns") or [])
        p.feature_scaling = str(config.get("feature_scaling", "rolling_zscore")).lower()
        if p.feature_scaling not in ("none", "rolling_zscore", "expanding_zscore"):
            raise ValueError(
                "feature_scaling must be one of ('none', 'rolling_zscore', "
                f"'expanding_zscore'); got {p.feature_scaling!r}"
            )
        p.feature_scaling_window = _i(config, "feature_scaling_window", 256)
        p.feature_clip = _f(config, "feature_clip", 10.0)
        p.include_price_window = _b(
            config, "include_price_window", not p.feature_columns
        ) if p.prep_id == PREP_FEATURE_WINDOW else _b(config, "include_price_window", True)
        if p.prep_id == PREP_DEFAULT:
            p.include_price_window = True
        p.include_agent_state = _b(config, "include_agent_state", True)
        if p.prep_id == PREP_DEFAULT:
            p.include_agent_state = True

        p.sl_pips = _f(config, "sl_pips", 20.0)
        p.tp_pips = _f(config, "tp_pips", 40.0)
        p.pip_size = _f(config, "pip_size", 0.0001)

        p.atr_period = _i(config, "atr_period", 14)
        p.k_sl = max(0.0, _f(config, "k_sl", 2.0))
        p.k_tp = max(0.0, _f(config, "k_tp", 3.0))
        rel = config.get("rel_volume")
        p.rel_volume = None if rel is None else float(rel)
        p.min_order_volume = _f(config, "min_order_volume", 0.0)
        p.max_order_volume = _f(config, "max_order_volume", 1e12)
        p.size_mode = (
            SIZE_MODE_NOTIONAL
            if str(config.get("size_mode", "fx_units")).lower() == "notional"
            else SIZE_MODE_FX_UNITS
        )
        msf = config.get("min_sltp_frac", 0.001)
        p.min_sltp_frac = None if msf is None else float(msf)
        xsf = config.get("max_sltp_frac", 0.20)
        p.max_sltp_frac = None if xsf is None else float(xsf)
        p.sltp_risk_mode = RISK_MODES.get(
            str(config.get("sltp_risk_mode", "fixed_atr")).strip().lower(), RISK_FIXED_ATR
        )
        p.baseline_rel_volume = max(0.0, _f(config, "baseline_rel_volume", 0.05))
        p.max_risk_rel_volume = max(
            p.baseline_rel_volume + 1e-12, _f(config, "max_risk_rel_volume", 0.50)
        )
        p.rel_volume_sl_shrink_alpha = min(
            max(_f(config, "rel_volume_sl_shrink_alpha", 0.35), 0.0), 0.95
        )
        p.rel_volume_tp_shrink_alpha = min(
            max(_f(config, "rel_volume_tp_shrink_alpha", 0.20), 0.0), 0.95
        )
        p.min_k_sl = max(0.0, _f(config, "min_k_sl", 1.0))
        p.min_reward_risk_ratio = max(0.0, _f(config, "min_reward_risk_ratio", 1.0))
        mplf = config.get("max_planned_loss_fraction")
        p.max_planned_loss_fraction = None if mplf is None else float(mplf)
        p.session_filter = _b(config, "session_filter", False)
        p.entry_dow_start = _i(config, "entry_dow_start", 0)
        p.entry_hour_start = _i(config, "entry_hour_start", 12)
        p.force_close_dow = _i(config, "force_close_dow", 4)
        p.force_close_hour = _i(config, "force_close_hour", 20)

        p.financing_enabled = _b(config, "financing_enabled", False)
        p.enforce_margin_preflight = _b(config, "enforce_margin_preflight", False)
        p.intrabar_collision_policy = _COLLISION_POLICIES[
            str(config.get("intrabar_collision_policy", "worst_case"))]
        p.limit_fill_policy = _LIMIT_POLICIES[
            str(config.get("limit_fill_policy", "touch"))]
        p.margin_model = _MARGIN_MODELS[
            str(config.get("margin_model", "leveraged"))]
        p.margin_init_rate = _f(config, "margin_init_rate", 0.03)
        latency_ms = _f(config, "latency_ms", 0.0)
        p.rollover_hour_utc = int(config.get("rollover_hour_utc", 22) or 22)
        p.stage_b_force_close_obs = _b(config, "stage_b_force_close_obs", False)
        p.force_close_window_hours = _i(config, "force_close_window_hours", 4)
        p.monday_entry_window_hours = _i(config, "monday_entry_window_hours", 4)
        p.stage_b_force_close_reward_penalty = _b(
            config, "stage_b_force_close_reward_penalty", False
        )
        p.force_close_exposure_penalty_coef = _f(
            config, "force_close_exposure_penalty_coef", 0.0
        )
        p.force_close_exposure_penalty_window_hours = _f(
            config, "force_close_exposure_penalty_window_hours", float(p.force_close_window_hours)
        )

        p.oanda_fx_calendar_obs = bool(
            _b(config, "oanda_fx_calendar_obs", False)
            or str(config.get("broker_profile") or "").lower() == "oanda_us_fx"
        )
        p.timeframe_hours = float(timeframe_hours)

        p.event_context_execution_overlay = _b(config, "event_context_execution_overlay", False)
        p.event_context_no_trade_column = str(
            config.get("event_context_no_trade_column", "event_no_trade_window_active")
        )
        p.event_context_no_trade_threshold = _f(config, "event_context_no_trade_threshold", 0.5)
        p.event_context_block_new_entries = _b(config, "event_context_block_new_entries", True)
        p.event_context_force_flat = _b(config, "event_context_force_flat", False)
        p.event_context_spread_stress_column = str(
            config.get("event_context_spread_stress_column", "event_spread_stress_multiplier")
        )
        p.event_context_slippage_stress_column = str(
            config.get("event_context_slippage_stress_column", "event_slippage_stress_multiplier")
        )

        p.autoreset = _b(config, "autoreset", False)
        p.env_start_mode = str(config.get("env_start_mode", "zero"))
        seed = config.get("seed")
        p.seed = None if seed is None else int(seed)

        # execution-cost profile overrides commission/slippage when provided
        profile_path = config.get("execution_cost_profile")
        if profile_path:
            from ..contracts import load_execution_cost_profile  # noqa: PLC0415

            prof = load_execution_cost_profile(profile_path)
            f = prof.as_floats()
            p.commission = f["commission_rate_per_side"]
            p.slippage = f["quote_adverse_rate_per_side"]
            p.financing_enabled = p.financing_enabled or prof.financing_enabled
            p.enforce_margin_preflight = (p.enforce_margin_preflight
                                          or prof.enforce_margin_preflight)
            # execution-realism tier: the profile is authoritative for the
            # policy fields (round 1 parsed-then-dropped them — VERDICT #2)
            p.intrabar_collision_policy = _COLLISION_POLICIES[
                prof.intrabar_collision_policy]
            p.limit_fill_policy = _LIMIT_POLICIES[prof.limit_fill_policy]
            p.margin_model = _MARGIN_MODELS[prof.margin_model]
            latency_ms = float(prof.latency_ms)

        # sub-bar latency collapses to next-open at OHLC granularity;
        # each full bar duration of latency delays the fill one more bar
        bar_ms = p.timeframe_hours * 3_600_000.0
        p.latency_bars = int(latency_ms // bar_ms) if bar_ms > 0 else 0

        p.finalize()
        return p

    def finalize(self) -> None:
        """Compute the flat observation layout (block name, width)."""
        self.n_features = len(self.feature_columns)
        blocks: List[Tuple[str, int]] = []
        if self.prep_id == PREP_FEATURE_WINDOW:
            if not self.feature_columns:
                raise ValueError(
                    "feature_window_preprocessor requires non-empty 'feature_columns'."
                )
            blocks.append(("features", self.window_size * self.n_features))
        if self.include_price_window:
            blocks.append(("prices", self.window_size))
            blocks.append(("returns", self.window_size))
        if self.include_agent_state:
            blocks.append(("agent_state", 4))
        if self.stage_b_force_close_obs:
            blocks.append(("force_close", 4))
        if self.oanda_fx_calendar_obs:
            blocks.append(("calendar", 11))
        if not blocks:
            raise ValueError("preprocessor observation contract emits no observation blocks")
        self.obs_blocks = blocks
        self.obs_dim = sum(w for _, w in blocks)

    def obs_slices(self) -> Dict[str, slice]:
        out: Dict[str, slice] = {}
        off = 0
        for name, width in self.obs_blocks:
            out[name] = slice(off, off + width)
            off += width
        return out
