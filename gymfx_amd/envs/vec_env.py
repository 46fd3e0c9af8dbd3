"""VecFxEnv — N trading environments as SoA device tensors.

The MI355X-native replacement for the reference's one-env-per-thread design
(/root/reference/app/env.py:93 + app/bt_bridge.py): all N envs advance in
lockstep; on GPU the whole step (action decode -> overlay -> broker fills ->
brackets -> ATR -> strategy -> equity -> reward -> observation) is ONE fused
HIP kernel (ops/csrc/env_step.hip); on CPU the same semantics run as
vectorized torch ops (envs/reference_step.py — also the kernel's oracle).
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from ..data.feed import concat_markets
from .market import MarketTensors, build_market_tensors
from .params import EnvParams
from .reference_step import build_obs_torch, step_torch
from .state import ACTION_COUNTERS, EXEC_COUNTERS, EnvState, alloc_state, reset_state_


def resolve_device(spec: Any = "auto") -> torch.device:
    if isinstance(spec, torch.device):
        return spec
    s = str(spec or "auto")
    if s == "auto":
        return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    return torch.device(s)


class VecFxEnv:
    """Vectorized FX trading environment (N envs, SoA state, device-resident)."""

    def __init__(
        self,
        config: Dict[str, Any],
        market_data,
        *,
        device: Any = None,
        use_native: Optional[bool] = None,
    ):
        self.config = dict(config)
        if isinstance(market_data, (list, tuple)):
            market_data = concat_markets(list(market_data))
        self.market_data = market_data
        self.device = resolve_device(device if device is not None else config.get("device", "auto"))
        self.params = EnvParams.from_config(
            self.config, timeframe_hours=market_data.timeframe_hours()
        )
        # per-instrument blocks (multi-pair: BASELINE config #5)
        self.instrument_blocks = market_data.meta.get(
            "instrument_blocks",
            [{"instrument": market_data.instrument, "lo": 0,
              "end": market_data.n_rows, "pip_size": self.params.pip_size}],
        )
        min_rows = min(b["end"] - b["lo"] for b in self.instrument_blocks)
        if min_rows < self.params.window_size + 2:
            raise ValueError("input data is empty or too short for the configured window")
        self.mt: MarketTensors = build_market_tensors(market_data, self.params, self.device)
        if self.params.financing_enabled:
            self._build_rollover_schedule()
        self.st: EnvState = alloc_state(self.params, self.device)
        self.total_bars = self.mt.T
        self._rng = np.random.default_rng(self.params.seed)
        self._native = None
        self._obs = torch.empty(
            self.params.n_envs, self.params.obs_dim, dtype=torch.float32, device=self.device
        )
        self._assign_instrument_blocks()
        self._assign_start_offsets()
        if use_native is None:
            use_native = self.device.type == "cuda"
        if use_native:
            # On a GPU the HIP extension is REQUIRED — never silently fall
            # back to the eager torch path.
            from ..ops import native  # noqa: PLC0415

            native.require()
            from ..ops.wrappers import NativeEngine  # noqa: PLC0415

            self._native = NativeEngine(self.st, self.mt, self.params)

    # ------------------------------------------------------------------
    @property
    def n_envs(self) -> int:
        return self.params.n_envs

    @property
    def obs_dim(self) -> int:
        return self.params.obs_dim

    def _build_rollover_schedule(self) -> None:
        """Per-bar financing multipliers, per instrument block
        (calendar.compute_rollover_schedule; FXRolloverInterestModule
        parity, nautilus_adapter.py:363-368)."""
        from ..calendar import compute_rollover_schedule

        md = self.market_data
        if md.timestamps is None:
            raise ValueError("financing_enabled requires bar timestamps")
        rate_data = self.config.get("rollover_rate_data")
        if not rate_data:
            # reference-style CSV path (LOCATION,TIME,Value monthly rows —
            # the config key of tests/test_nautilus_gym_bridge.py:24-26)
            path = self.config.get("financing_rate_data_file")
            if path:
                import csv

                with open(path, newline="", encoding="utf-8") as fh:
                    rate_data = list(csv.DictReader(fh))
        if not rate_data:
            raise ValueError(
                "financing_enabled requires rollover_rate_data (inline) or "
                "financing_rate_data_file (CSV path)")
        sched = np.zeros(md.n_rows, dtype=np.float32)
        for b in self.instrument_blocks:
            sched[b["lo"]:b["end"]] = compute_rollover_schedule(
                md.timestamps[b["lo"]:b["end"]], b["instrument"], rate_data,
                self.params.rollover_hour_utc)
        self.mt.roll_rate = torch.from_numpy(sched).to(self.device)

    def _assign_instrument_blocks(self) -> None:
        """Round-robin envs over instrument blocks; set per-env episode
        bounds (lo_bar/end_bar) and pip size."""
        N = self.params.n_envs
        B = len(self.instrument_blocks)
        inst = np.arange(N, dtype=np.int32) % B
        lo = np.array([self.instrument_blocks[i]["lo"] for i in inst], dtype=np.int32)
        end = np.array([self.instrument_blocks[i]["end"] for i in inst], dtype=np.int32)
        pip = np.array([self.instrument_blocks[i].get("pip_size", self.params.pip_size)
                        for i in inst], dtype=np.float32)
        self.st.inst_id.copy_(torch.from_numpy(inst).to(self.device))
        self.st.lo_bar.copy_(torch.from_numpy(lo).to(self.device))
        self.st.end_bar.copy_(torch.from_numpy(end).to(self.device))
        self.st.pip_env.copy_(torch.from_numpy(pip).to(self.device))

    def _assign_start_offsets(self) -> None:
        N, W = self.params.n_envs, self.params.window_size
        mode = self.params.env_start_mode
        lo = self.st.lo_bar.cpu().numpy().astype(np.int64)
        end = self.st.end_bar.cpu().numpy().astype(np.int64)
        max_off = np.maximum(end - W - 2, lo)
        if mode == "spread" and N > 1:
            # spread each env within its own instrument block
            frac = np.zeros(N)
            B = len(self.instrument_blocks)
            for b in range(B):
                idx = np.where((np.arange(N) % B) == b)[0]
                if len(idx) > 1:
                    frac[idx] = np.linspace(0.0, 1.0, len(idx))
            off = np.floor(lo + frac * (max_off - lo)).astype(np.int32)
        elif mode == "random":
            off = (lo + self._rng.integers(0, 1 << 30, size=N)
                   % np.maximum(max_off - lo + 1, 1)).astype(np.int32)
        else:
            off = lo.astype(np.int32)
        self.st.start_offset = torch.from_numpy(off).to(self.device)

    # ------------------------------------------------------------------
    def reset(self, *, seed: Optional[int] = None) -> torch.Tensor:
        if seed is not None:
            self._rng = np.random.default_rng(seed)
            if self.params.env_start_mode == "random":
                self._assign_start_offsets()
        mask = torch.ones(self.params.n_envs, dtype=torch.bool, device=self.device)
        reset_state_(self.st, self.params, mask)
        self._build_obs()
        return self._obs

    def _build_obs(self) -> None:
        if self._native is not None:
            self._native.build_obs(self._obs)
        else:
            build_obs_torch(self.st, self.mt, self.params, out=self._obs)

    def step(self, actions: torch.Tensor, *,
             reward_out: Optional[torch.Tensor] = None,
             terminated_out: Optional[torch.Tensor] = None,
             obs_bf16_out: Optional[torch.Tensor] = None,
             env_lo: int = 0, env_hi: int = 0,
             head: Optional[torch.Tensor] = None,
             logp_out: Optional[torch.Tensor] = None,
             value_out: Optional[torch.Tensor] = None,
             step_base: Optional[torch.Tensor] = None,
             sample_seed: int = 0,
             sample_step: int = 0,
             fuse_obs: bool = False,
             h2: Optional[torch.Tensor] = None,
             w3t: Optional[torch.Tensor] = None,
             b3: Optional[torch.Tensor] = None,
             rnn_h: Optional[torch.Tensor] = None,
             rnn_c: Optional[torch.Tensor] = None) -> Dict[str, torch.Tensor]:
        """Advance all envs. Returns dict with obs/reward/terminated tensors.

        reward_out / terminated_out / obs_bf16_out: optional preallocated
        device tensors the fused kernels write directly (rollout slabs — no
        copy kernels; used by the PPO trainer's hipGraph bodies).

        head / logp_out / value_out / step_base / sample_seed / sample_step:
        fused policy sampling (native engine only) — the step kernel samples
        the action from `head` itself, writing into `actions`, saving one
        kernel launch per rollout step.

        rnn_h / rnn_c: recurrent-state autoreset fused into the step — the
        kernel zeros the terminated env's LSTM state rows [env_lo:env_hi)
        itself (replaces the trainer's per-step mask_reset launch)."""
        if not isinstance(actions, torch.Tensor):
            actions = torch.as_tensor(actions, device=self.device)
        actions = actions.to(self.device)
        if self._native is not None:
            # fused HIP path: step + autoreset + obs in two kernel launches
            info = self._native.step(actions, self._obs, reward_out,
                                     terminated_out, obs_bf16_out,
                                     env_lo, env_hi, head, logp_out,
                                     value_out, step_base, sample_seed,
                                     sample_step, fuse_obs, h2, w3t, b3,
                                     rnn_h, rnn_c)
            info["obs"] = self._obs
            return info
        if head is not None or h2 is not None:
            raise ValueError("fused sampling requires the native engine")
        if env_lo != 0 or env_hi not in (0, self.params.n_envs):
            raise ValueError("env range stepping requires the native engine")
        info = step_torch(self.st, self.mt, self.params, actions)
        terminated = info["terminated"]
        if self.params.autoreset:
            done = terminated.clone()
            if bool(done.any()):
                reset_state_(self.st, self.params, done)
            info["terminated"] = done
        if rnn_h is not None:
            # same semantics as the native kernel's fused state reset
            done_rows = info["terminated"]
            rnn_h[done_rows] = 0
            rnn_c[done_rows] = 0
        self._build_obs()
        info["obs"] = self._obs
        if reward_out is not None:
            reward_out.copy_(info["reward"])
        if terminated_out is not None:
            terminated_out.copy_(info["terminated"])
        if obs_bf16_out is not None:
            obs_bf16_out.view(-1).copy_(self._obs.reshape(-1).to(torch.bfloat16))
        return info

    # ------------------------------------------------------------------
    # Observability / summary
    # ------------------------------------------------------------------
    def bridge_state(self, i: int = 0) -> Dict[str, Any]:
        """Reference-shaped per-env state view (BTBridge fields)."""
        st = self.st
        t = int(torch.clamp(st.cursor[i] - 1, min=0).item())
        return {
            "position": int(torch.sign(st.pos[i]).item()),
            "equity": float(st.equity[i].item()),
            "prev_equity": float(st.prev_equity[i].item()),
            "initial_cash": self.params.initial_cash,
            "price": float(self.mt.close[t].item()),
            "bar_index": int(st.cursor[i].item()),
            "total_bars": self.total_bars,
            "trade_count": int(st.trade_count[i].item()),
            "commission_paid": float(st.commission_paid[i].item()),
            "last_trade_cost": float(st.last_trade_cost[i].item()),
            "terminated": bool(st.terminated[i].item()),
        }

    def action_diagnostics(self, i: int = 0) -> Dict[str, Any]:
        d = {
            name: int(self.st.act_diag[i, k].item())
            for k, name in enumerate(ACTION_COUNTERS)
        }
        d["raw_abs_sum"] = float(self.st.raw_abs_sum[i].item())
        rmin = float(self.st.raw_min[i].item())
        rmax = float(self.st.raw_max[i].item())
        d["raw_min"] = None if rmin == float("inf") else rmin
        d["raw_max"] = None if rmax == float("-inf") else rmax
        d["continuous_action_threshold"] = (
            self.params.continuous_action_threshold
            if self.params.action_space_mode == "continuous"
            else None
        )
        return d

    def execution_diagnostics(self, i: int = 0) -> Dict[str, int]:
        return {
            name: int(self.st.exec_diag[i, k].item())
            for k, name in enumerate(EXEC_COUNTERS)
        }

    def analyzers(self, i: int = 0) -> Dict[str, Any]:
        """Analyzer-equivalents for the metrics plugin contract
        (trades/sharpe/drawdown/sqn digest, bt_bridge.py:277-281)."""
        st = self.st
        n_tr = int(st.trade_count[i].item())
        pnl_sum = float(st.trade_pnl_sum[i].item())
        pnl_sumsq = float(st.trade_pnl_sumsq[i].item())
        sqn = None
        if n_tr >= 2:
            mean = pnl_sum / n_tr
            var = max(pnl_sumsq / n_tr - mean * mean, 0.0)
            std = var ** 0.5
            if std > 0:
                sqn = (n_tr ** 0.5) * mean / std
        n_ret = int(st.ret_count[i].item())
        sharpe = None
        if n_ret >= 2:
            rs = float(st.ret_sum[i].item())
            rss = float(st.ret_sumsq[i].item())
            mean = rs / n_ret
            var = max((rss - n_ret * mean * mean) / (n_ret - 1), 0.0)
            std = var ** 0.5
            if std > 0:
                ann = float(self.config.get("annualization_factor", 252.0))
                sharpe = mean / std * (ann ** 0.5)
        return {
            "trades": {
                "total": {"total": n_tr},
                "won": {"total": int(st.trade_won[i].item())},
                "lost": {"total": int(st.trade_lost[i].item())},
                "pnl": {"net": {"average": (pnl_sum / n_tr) if n_tr else None}},
            },
            "sharpe": {"sharperatio": sharpe},
            "drawdown": {
                "max": {
                    "drawdown": float(st.max_dd_pct[i].item()),
                    "moneydown": float(st.max_dd_money[i].item()),
                }
            },
            "sqn": {"sqn": sqn},
            "time_return": {},
        }

    def fleet_analyzers(self) -> Dict[str, Any]:
        """Fleet-pooled analyzer digest (same shape as :meth:`analyzers`)
        so the metrics plugins can summarize a vectorized evaluation run:
        trade counters and pnl moments sum over envs, the Sharpe pools
        every env's step returns, drawdown reports the fleet mean (pct)
        and the worst env (money)."""
        st = self.st
        n_tr = int(st.trade_count.sum().item())
        pnl_sum = float(st.trade_pnl_sum.sum().item())
        pnl_sumsq = float(st.trade_pnl_sumsq.sum().item())
        sqn = None
        if n_tr >= 2:
            mean = pnl_sum / n_tr
            var = max(pnl_sumsq / n_tr - mean * mean, 0.0)
            if var > 0:
                sqn = (n_tr ** 0.5) * mean / (var ** 0.5)
        n_ret = int(st.ret_count.sum().item())
        sharpe = None
        if n_ret >= 2:
            rs = float(st.ret_sum.sum().item())
            rss = float(st.ret_sumsq.sum().item())
            mean = rs / n_ret
            var = max((rss - n_ret * mean * mean) / (n_ret - 1), 0.0)
            if var > 0:
                ann = float(self.config.get("annualization_factor", 252.0))
                sharpe = mean / (var ** 0.5) * (ann ** 0.5)
        return {
            "trades": {
                "total": {"total": n_tr},
                "won": {"total": int(st.trade_won.sum().item())},
                "lost": {"total": int(st.trade_lost.sum().item())},
                "pnl": {"net": {"average": (pnl_sum / n_tr) if n_tr else None}},
            },
            "sharpe": {"sharperatio": sharpe},
            "drawdown": {
                "max": {
                    "drawdown": float(st.max_dd_pct.mean().item()),
                    "moneydown": float(st.max_dd_money.max().item()),
                }
            },
            "sqn": {"sqn": sqn},
            "time_return": {},
        }

    def vec_summary(self) -> Dict[str, Any]:
        """Whole-fleet aggregates (no reference counterpart)."""
        st = self.st
        return {
            "n_envs": self.params.n_envs,
            "mean_equity": float(st.equity.mean().item()),
            "mean_episode_return": float(st.episode_return.mean().item()),
            "mean_episode_steps": float(st.episode_step.float().mean().item()),
            "terminated_envs": int(st.terminated.sum().item()),
            "total_trades": int(st.trade_count.sum().item()),
        }
