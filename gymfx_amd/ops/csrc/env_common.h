// Shared structs for the fused env-step/obs kernels (gfx950 / CDNA4 only).
// Keep enums in sync with gymfx_amd/envs/params.py.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define GFX_DEV __device__ __forceinline__

namespace gymfx {

// plugin dispatch enums (== envs/params.py)
enum RewardId { REWARD_PNL = 0, REWARD_SHARPE = 1, REWARD_DD = 2 };
enum StrategyId { STRAT_DIRECT = 0, STRAT_FIXED = 1, STRAT_ATR = 2 };
enum PrepId { PREP_DEFAULT = 0, PREP_FEATURE_WINDOW = 1 };
enum ScalingMode { SCALE_NONE = 0, SCALE_ROLLING = 1, SCALE_EXPANDING = 2 };
enum SizeMode { SIZE_FX_UNITS = 0, SIZE_NOTIONAL = 1 };
enum RiskMode { RISK_FIXED = 0, RISK_RELVOL = 1, RISK_MARGIN = 2 };
// execution-realism policies (contracts.py execution_cost_profile.v1)
enum CollisionPolicy { COLL_WORST = 0, COLL_OHLC = 1, COLL_ADAPTIVE = 2 };
enum LimitPolicy { LIM_TOUCH = 0, LIM_CROSS = 1, LIM_CONSERVATIVE = 2 };
enum MarginModel { MARGIN_LEVERAGED = 0, MARGIN_STANDARD = 1 };

// flag bits (== ops/wrappers.py FLAG_*)
enum Flags {
  F_CONTINUOUS = 1 << 0,
  F_INCLUDE_PRICE = 1 << 1,
  F_INCLUDE_AGENT = 1 << 2,
  F_STAGEB_OBS = 1 << 3,
  F_CALENDAR_OBS = 1 << 4,
  F_OVERLAY = 1 << 5,
  F_OVERLAY_BLOCK = 1 << 6,
  F_OVERLAY_FF = 1 << 7,
  F_SESSION_FILTER = 1 << 8,
  F_HAS_MINFRAC = 1 << 9,
  F_HAS_MAXFRAC = 1 << 10,
  F_HAS_RELVOL = 1 << 11,
  F_HAS_MPLF = 1 << 12,
  F_STAGEB_PENALTY = 1 << 13,
  F_AUTORESET = 1 << 14,
  F_FINANCING = 1 << 15,
  F_PREFLIGHT = 1 << 16,
};

// execution-diagnostics counter indices (== envs/state.py EXEC_COUNTERS)
enum ExecCounter {
  E_ENTRY_ACTIONS_SEEN = 0,
  E_ENTRY_ORDERS_SUBMITTED,
  E_BLOCKED_SESSION_FILTER,
  E_BLOCKED_ATR_WARMUP,
  E_BLOCKED_NON_POSITIVE_ATR,
  E_BLOCKED_NON_POSITIVE_SIZE,
  E_BLOCKED_NON_POSITIVE_PRICE,
  E_DEFAULT_ORDERS_SUBMITTED,
  E_PLUGIN_APPLY_ERRORS,
  E_EV_NO_TRADE_ACTIVE_STEPS,
  E_EV_ACTION_OVERRIDES,
  E_EV_BLOCKED_ENTRIES,
  E_EV_FORCED_FLAT_ACTIONS,
  E_EV_FORCED_FLAT_ORDERS,
  E_SESSION_FORCE_CLOSES,
  E_BRACKET_SL_FILLS,
  E_BRACKET_TP_FILLS,
  E_MARGIN_PREFLIGHT_DENIED,
  EXEC_COUNTER_N,
};

enum ActCounter {
  A_STEPS = 0,
  A_HOLD,
  A_LONG,
  A_SHORT,
  A_NON_HOLD,
  A_DEADBAND,
  ACT_COUNTER_N,
};

// ---------------------------------------------------------------------------
// Shared categorical-sampler math (used by sample_head_kernel AND the fused
// env_step sampling path so both are bitwise-identical by construction).
// libm expf/logf are CALLS on gfx950; these are single v_exp_f32/v_log_f32.
// hipcc-only: bindings.cpp (plain host c++) includes this header for the
// structs and must not see the amdgcn builtins.
// ---------------------------------------------------------------------------
#ifdef __HIPCC__
GFX_DEV float fast_exp(float x) {
  return __builtin_amdgcn_exp2f(x * 1.4426950408889634f);
}
GFX_DEV float fast_log(float x) {
  return __builtin_amdgcn_logf(x) * 0.6931471805599453f;
}
// Counter-based RNG: deterministic in (seed, step, env).
GFX_DEV uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
GFX_DEV float head_logz(const float* row, int n_actions) {
  float mx = row[0];
  for (int j = 1; j < n_actions; ++j) mx = fmaxf(mx, row[j]);
  float z = 0.f;
  for (int j = 0; j < n_actions; ++j) z += fast_exp(row[j] - mx);
  return fast_log(z) + mx;
}
GFX_DEV int sample_categorical(const float* row, int n_actions, float logz,
                               uint64_t seed, uint64_t step, int64_t mg) {
  const uint64_t r =
      splitmix64(seed ^ (step * 0x51E1F5ull + (uint64_t)mg * 0x9E37ull));
  float u = (float)((r >> 11) * (1.0 / 9007199254740992.0));  // [0,1)
  u = fminf(u, 0.999999f);
  float c = 0.f;
  int a = n_actions - 1;
  for (int j = 0; j < n_actions; ++j) {
    c += fast_exp(row[j] - logz);
    if (u < c) { a = j; break; }
  }
  return a;
}
#endif  // __HIPCC__

// Row-permutation descriptor for the gather+first-GEMM fusion (the L1
// forward GEMM / W1 wgrad read rollout rows through the epoch Feistel
// permutation instead of a materialized minibatch copy).
struct FeistelMap {
  uint32_t n = 0;   // permutation domain (total rollout rows)
  int half = 0;
  uint64_t seed = 0;
  int minibatches = 1;
  int M_mb = 0;     // rows per minibatch
  long long ctr_off = 0;
  const unsigned long long* step_base = nullptr;
  const unsigned long long* mb_ctr = nullptr;
};

struct EnvParamsK {
  int n_envs, T, window, n_features;
  int reward_id, strategy_id, prep_id;
  int scaling_mode, scale_window, sharpe_window, atr_period;
  int size_mode, risk_mode;
  int collision_policy, limit_policy, latency_bars, margin_model;
  int flags;
  int obs_dim;
  int off_features, off_prices, off_returns, off_agent, off_fc, off_cal;

  double initial_cash, position_size, commission, slippage, leverage, min_equity;
  double cont_threshold, reward_scale, annualization, penalty_lambda;
  double sl_pips, tp_pips, pip_size;
  double k_sl, k_tp, rel_volume, min_order_volume, max_order_volume;
  double min_sltp_frac, max_sltp_frac;
  double baseline_rel_volume, max_risk_rel_volume;
  double sl_shrink_alpha, tp_shrink_alpha, min_k_sl, min_rr, mplf;
  double fc_pen_coef, fc_pen_window_hours;
  double feature_clip, overlay_threshold;
  double margin_init_rate;
  // fused policy sampling (set per step() call when head != null)
  int head_hidden;           // h2 width for the head-in-step fusion
  int rnn_hidden;            // LSTM state width for the fused state reset
  unsigned long long sample_seed;
  long long sample_step;
  int sample_nact;
};

struct EnvPtrs {
  // market (device, read-only)
  const float *open_px, *high_px, *low_px, *close_px, *price_px;
  const float *features;      // [T, F]
  const double *p1, *p2;      // [T+1, F] prefix sums
  const bool *binary_mask;    // [F]
  const float *ev_no_trade;   // [T]
  const float *force_close;   // [T, 4] or null
  const float *calendar;      // [T, 10] or null
  const bool *sess_entry, *sess_close;  // [T]
  const float *roll_rate;    // [T] financing multiplier or null
  // state (device, mutable)
  int *cursor;
  bool *started, *terminated;
  double *pos, *avg_entry, *cash, *margin_used, *equity, *prev_equity, *peak_equity;
  double *commission_paid, *last_trade_cost;
  int *trade_count;
  bool *pend_close;
  int8_t *pend_open_dir;
  float *pend_open_size, *pend_sl, *pend_tp;
  int *pend_wait;            // bars the pending order is still in transit
  bool *br_active, *br_armed;
  float *br_sl, *br_tp;
  float *tr_ring;            // [N, P]
  int *tr_count;
  float *tr_sum;
  float *prev_close_atr;
  float *rew_ring;           // [N, W_sharpe]
  int *rew_count;
  double *rew_s1, *rew_s2;   // O(1) sharpe running sum / sum-of-squares
  int *trade_won, *trade_lost;
  double *trade_pnl_sum, *trade_pnl_sumsq;
  double *metric_peak, *max_dd_money, *max_dd_pct, *ret_sum, *ret_sumsq;
  int *ret_count;
  int *episode_step;
  double *episode_return;
  int *start_offset;
  const int *lo_bar;         // [N] first bar of env's instrument block
  const int *end_bar;        // [N] one past last bar
  const int *inst_id;        // [N]
  const float *pip_env;      // [N] per-env pip size
  int *exec_diag;            // [N, EXEC_COUNTER_N]
  int *act_diag;             // [N, ACT_COUNTER_N]
  float *raw_abs_sum, *raw_min, *raw_max;
  // fused policy sampling (optional; saves one kernel launch per rollout
  // step): when `head` is non-null the step kernel samples the action
  // itself — identical math to sample_head_kernel — and writes it to
  // actions_out before decoding it.
  const float *head;         // [env_cnt, sample_nact+1] f32 (rows local to env_lo)
  // head-in-step fusion: when h2/w3t/b3 are set (and head is null) the
  // step kernel computes head = h2 @ W3 + b3 per env itself — removes the
  // tiny head GEMM launch from the latency-bound rollout chain
  const void *h2;            // [env_cnt, head_hidden] bf16
  const void *w3t;           // [sample_nact+1, head_hidden] bf16
  const float *b3;           // [sample_nact+1]
  // recurrent-state autoreset: when set, the step kernel zeros the
  // terminated env's LSTM state rows itself — one fewer launch per
  // rollout step than the separate mask_reset kernel (done rows are
  // rare, so the per-lane zero loop is almost always skipped)
  void *rnn_h;               // [env_cnt, rnn_hidden] bf16 (rows local to env_lo)
  float *rnn_c;              // [env_cnt, rnn_hidden] f32
  int64_t *actions_out;      // sampled action (same buffer `actions` reads)
  float *logp_out;           // [N] f32
  float *value_out;          // [N] f32
  const unsigned long long *step_base;  // device counter (graph-replayable)
  // step inputs/outputs
  const void *actions;       // int64 (discrete) or float32 (continuous)
  float *reward_out, *base_reward_out, *penalty_out;
  bool *terminated_out;
  int64_t *coerced_out;
  float *obs_out;            // [N, obs_dim]
  void *obs_bf16_out;        // optional bf16 mirror (written by build_obs)
};

}  // namespace gymfx
