"""Python glue for the HIP engine: packs EnvParams into the kernel param
dicts and owns a GymFxEngine instance per VecFxEnv (pointers cached once —
the per-step path is two kernel launches, no re-validation)."""
from __future__ import annotations

from typing import Dict

import torch

from ..envs.market import MarketTensors
from ..envs.params import EnvParams
from ..envs.state import EnvState
from . import native

# flag bits — keep in sync with csrc/env_common.h
F_CONTINUOUS = 1 << 0
F_INCLUDE_PRICE = 1 << 1
F_INCLUDE_AGENT = 1 << 2
F_STAGEB_OBS = 1 << 3
F_CALENDAR_OBS = 1 << 4
F_OVERLAY = 1 << 5
F_OVERLAY_BLOCK = 1 << 6
F_OVERLAY_FF = 1 << 7
F_SESSION_FILTER = 1 << 8
F_HAS_MINFRAC = 1 << 9
F_HAS_MAXFRAC = 1 << 10
F_HAS_RELVOL = 1 << 11
F_HAS_MPLF = 1 << 12
F_STAGEB_PENALTY = 1 << 13
F_AUTORESET = 1 << 14
F_FINANCING = 1 << 15
F_PREFLIGHT = 1 << 16

_SCALING = {"none": 0, "rolling_zscore": 1, "expanding_zscore": 2}


def _flags(p: EnvParams) -> int:
    f = 0
    if p.action_space_mode == "continuous":
        f |= F_CONTINUOUS
    if p.include_price_window:
        f |= F_INCLUDE_PRICE
    if p.include_agent_state:
        f |= F_INCLUDE_AGENT
    if p.stage_b_force_close_obs:
        f |= F_STAGEB_OBS
    if p.oanda_fx_calendar_obs:
        f |= F_CALENDAR_OBS
    if p.event_context_execution_overlay:
        f |= F_OVERLAY
    if p.event_context_block_new_entries:
        f |= F_OVERLAY_BLOCK
    if p.event_context_force_flat:
        f |= F_OVERLAY_FF
    if p.session_filter:
        f |= F_SESSION_FILTER
    if p.min_sltp_frac is not None:
        f |= F_HAS_MINFRAC
    if p.max_sltp_frac is not None:
        f |= F_HAS_MAXFRAC
    if p.rel_volume is not None:
        f |= F_HAS_RELVOL
    if p.max_planned_loss_fraction is not None:
        f |= F_HAS_MPLF
    if p.stage_b_force_close_obs and p.stage_b_force_close_reward_penalty:
        f |= F_STAGEB_PENALTY
    if p.autoreset:
        f |= F_AUTORESET
    if p.financing_enabled:
        f |= F_FINANCING
    if p.enforce_margin_preflight:
        f |= F_PREFLIGHT
    return f


def pack_params(p: EnvParams, T: int):
    sl = p.obs_slices()

    def off(name: str) -> int:
        return sl[name].start if name in sl else -1

    iparams = {
        "n_envs": p.n_envs,
        "T": T,
        "window": p.window_size,
        "n_features": p.n_features,
        "reward_id": p.reward_id,
        "strategy_id": p.strategy_id,
        "prep_id": p.prep_id,
        "scaling_mode": _SCALING[p.feature_scaling],
        "scale_window": p.feature_scaling_window,
        "sharpe_window": p.sharpe_window,
        "atr_period": p.atr_period,
        "size_mode": p.size_mode,
        "risk_mode": p.sltp_risk_mode,
        "collision_policy": p.intrabar_collision_policy,
        "limit_policy": p.limit_fill_policy,
        "latency_bars": p.latency_bars,
        "margin_model": p.margin_model,
        "flags": _flags(p),
        "obs_dim": p.obs_dim,
        "off_features": off("features"),
        "off_prices": off("prices"),
        "off_returns": off("returns"),
        "off_agent": off("agent_state"),
        "off_fc": off("force_close"),
        "off_cal": off("calendar"),
    }
    fparams = {
        "initial_cash": p.initial_cash,
        "position_size": p.position_size,
        "commission": p.commission,
        "slippage": p.slippage,
        "leverage": p.leverage,
        "min_equity": p.min_equity,
        "cont_threshold": p.continuous_action_threshold,
        "reward_scale": p.reward_scale,
        "annualization": p.annualization_factor,
        "penalty_lambda": p.penalty_lambda,
        "sl_pips": p.sl_pips,
        "tp_pips": p.tp_pips,
        "pip_size": p.pip_size,
        "k_sl": p.k_sl,
        "k_tp": p.k_tp,
        "rel_volume": p.rel_volume if p.rel_volume is not None else 0.0,
        "min_order_volume": p.min_order_volume,
        "max_order_volume": p.max_order_volume,
        "min_sltp_frac": p.min_sltp_frac if p.min_sltp_frac is not None else 0.0,
        "max_sltp_frac": p.max_sltp_frac if p.max_sltp_frac is not None else 0.0,
        "baseline_rel_volume": p.baseline_rel_volume,
        "max_risk_rel_volume": p.max_risk_rel_volume,
        "sl_shrink_alpha": p.rel_volume_sl_shrink_alpha,
        "tp_shrink_alpha": p.rel_volume_tp_shrink_alpha,
        "min_k_sl": p.min_k_sl,
        "min_rr": p.min_reward_risk_ratio,
        "mplf": (
            p.max_planned_loss_fraction if p.max_planned_loss_fraction is not None else 0.0
        ),
        "fc_pen_coef": p.force_close_exposure_penalty_coef,
        "fc_pen_window_hours": p.force_close_exposure_penalty_window_hours,
        "feature_clip": p.feature_clip,
        "overlay_threshold": p.event_context_no_trade_threshold,
        "margin_init_rate": p.margin_init_rate,
    }
    return iparams, fparams


def _market_dict(mt: MarketTensors) -> Dict[str, torch.Tensor]:
    d = {
        "open": mt.open,
        "high": mt.high,
        "low": mt.low,
        "close": mt.close,
        "price": mt.price,
        "ev_no_trade": mt.ev_no_trade,
        "sess_entry": mt.sess_entry,
        "sess_close": mt.sess_close,
    }
    for k in ("features", "feat_prefix1", "feat_prefix2", "binary_mask",
              "force_close", "calendar", "roll_rate"):
        v = getattr(mt, k)
        if v is not None:
            d[k] = v
    return d


class NativeEngine:
    """One fused-kernel engine bound to a VecFxEnv's state tensors."""

    def __init__(self, st: EnvState, mt: MarketTensors, params: EnvParams):
        ext = native.load()
        if ext is None:
            raise RuntimeError("_gymfx_hip extension missing (see ops/native.py)")
        iparams, fparams = pack_params(params, mt.T)
        self._params = params
        self._engine = ext.GymFxEngine(iparams, fparams, st.to_dict(), _market_dict(mt))

    def step(self, actions: torch.Tensor, obs_out: torch.Tensor,
             reward_out: torch.Tensor = None, terminated_out: torch.Tensor = None,
             obs_bf16_out: torch.Tensor = None, env_lo: int = 0,
             env_hi: int = 0, head: torch.Tensor = None,
             logp_out: torch.Tensor = None, value_out: torch.Tensor = None,
             step_base: torch.Tensor = None, sample_seed: int = 0,
             sample_step: int = 0, fuse_obs: bool = False,
             h2: torch.Tensor = None, w3t: torch.Tensor = None,
             b3: torch.Tensor = None, rnn_h: torch.Tensor = None,
             rnn_c: torch.Tensor = None) -> Dict[str, torch.Tensor]:
        if self._params.action_space_mode == "continuous":
            actions = actions.to(torch.float32).contiguous()
        else:
            actions = actions.to(torch.int64).contiguous()
        return dict(self._engine.step(actions, obs_out, reward_out,
                                      terminated_out, obs_bf16_out,
                                      env_lo, env_hi, head, logp_out,
                                      value_out, step_base, sample_seed,
                                      sample_step, fuse_obs, h2, w3t, b3,
                                      rnn_h, rnn_c))

    def build_obs(self, obs_out: torch.Tensor,
                  obs_bf16_out: torch.Tensor = None) -> None:
        self._engine.build_obs(obs_out, obs_bf16_out)
