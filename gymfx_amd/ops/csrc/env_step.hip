// Fused vectorized env step + observation build (gfx950).
//
// One lane per environment: the whole per-bar transition of the reference
// engine (action decode -> event overlay -> pending market fills at open ->
// bracket SL/TP children (worst-case ordering) -> ATR update -> strategy
// decision -> equity publish -> reward -> metrics -> optional autoreset)
// runs as straight-line scalar code per lane; N environments = N lanes.
//
// Semantics oracle: gymfx_amd/envs/reference_step.py (torch CPU) — GPU tests
// assert elementwise equality.  Reference provenance for the behavior:
// /root/reference/app/bt_bridge.py:136-248, app/env.py:279-440,
// broker_plugins/default_broker.py, strategy_plugins/direct_{fixed,atr}_sltp.py,
// reward_plugins/*.py.
#include "env_common.h"

namespace gymfx {

// Fill prices are computed in f32 to match the torch oracle exactly
// (f32 market tensor * f32-cast scalar); the money ledger then runs in f64.
GFX_DEV float buy_fill(float o, float slip) { return o * (1.0f + slip); }
GFX_DEV float sell_fill(float o, float slip) { return o * (1.0f - slip); }

// Close the full position at `fill`; mirrors reference_step._close_position.
GFX_DEV void close_position(const EnvPtrs& P, const EnvParamsK& K, int n,
                            double fill, int counter) {
  double pos = P.pos[n];
  double realized = pos * (fill - P.avg_entry[n]);
  double comm = fabs(pos) * fill * K.commission;
  P.cash[n] += P.margin_used[n] + realized - comm;
  P.commission_paid[n] += comm;
  P.last_trade_cost[n] += comm;
  P.trade_count[n] += 1;
  if (realized > 0) P.trade_won[n] += 1;
  if (realized < 0) P.trade_lost[n] += 1;
  double net = realized - comm;
  P.trade_pnl_sum[n] += net;
  P.trade_pnl_sumsq[n] += net * net;
  P.pos[n] = 0.0;
  P.avg_entry[n] = 0.0;
  P.margin_used[n] = 0.0;
  P.br_active[n] = false;
  P.br_armed[n] = false;
  if (counter >= 0) P.exec_diag[n * EXEC_COUNTER_N + counter] += 1;
}

GFX_DEV void reset_env(const EnvPtrs& P, const EnvParamsK& K, int n) {
  const double ic = K.initial_cash;
  P.cursor[n] = 1 + P.start_offset[n];
  P.started[n] = false;
  P.terminated[n] = false;
  P.pos[n] = 0.0;
  P.avg_entry[n] = 0.0;
  P.cash[n] = ic;
  P.margin_used[n] = 0.0;
  P.equity[n] = ic;
  P.prev_equity[n] = ic;
  P.peak_equity[n] = 0.0;
  P.commission_paid[n] = 0.0;
  P.last_trade_cost[n] = 0.0;
  P.trade_count[n] = 0;
  P.pend_close[n] = false;
  P.pend_open_dir[n] = 0;
  P.pend_open_size[n] = 0.f;
  P.pend_sl[n] = 0.f;
  P.pend_tp[n] = 0.f;
  P.pend_wait[n] = 0;
  P.br_active[n] = false;
  P.br_armed[n] = false;
  P.br_sl[n] = 0.f;
  P.br_tp[n] = 0.f;
  for (int i = 0; i < K.atr_period; ++i) P.tr_ring[(int64_t)n * K.atr_period + i] = 0.f;
  P.tr_count[n] = 0;
  P.tr_sum[n] = 0.f;
  P.prev_close_atr[n] = __builtin_nanf("");
  for (int i = 0; i < K.sharpe_window; ++i) P.rew_ring[(int64_t)n * K.sharpe_window + i] = 0.f;
  P.rew_count[n] = 0;
  P.rew_s1[n] = 0.0;
  P.rew_s2[n] = 0.0;
  P.trade_won[n] = 0;
  P.trade_lost[n] = 0;
  P.trade_pnl_sum[n] = 0.0;
  P.trade_pnl_sumsq[n] = 0.0;
  P.metric_peak[n] = ic;
  P.max_dd_money[n] = 0.0;
  P.max_dd_pct[n] = 0.0;
  P.ret_sum[n] = 0.0;
  P.ret_sumsq[n] = 0.0;
  P.ret_count[n] = 0;
  P.episode_step[n] = 0;
  P.episode_return[n] = 0.0;
  for (int i = 0; i < EXEC_COUNTER_N; ++i) P.exec_diag[n * EXEC_COUNTER_N + i] = 0;
  for (int i = 0; i < ACT_COUNTER_N; ++i) P.act_diag[n * ACT_COUNTER_N + i] = 0;
  P.raw_abs_sum[n] = 0.f;
  P.raw_min[n] = INFINITY;
  P.raw_max[n] = -INFINITY;
}

GFX_DEV void env_step_one(const EnvPtrs& P, const EnvParamsK& K, const int n,
                          const int env_lo) {
  const int T = K.T;
  // per-env instrument block bounds (multi-pair market; single-pair: 0..T)
  const int LO = P.lo_bar ? P.lo_bar[n] : 0;
  const int EB = (P.end_bar && P.end_bar[n] > 0) ? P.end_bar[n] : T;
  const float slip = (float)K.slippage;
  int* ediag = P.exec_diag + (int64_t)n * EXEC_COUNTER_N;
  int* adiag = P.act_diag + (int64_t)n * ACT_COUNTER_N;

  // ---- fused policy sampling (optional) ------------------------------
  // One thread already owns env row n; sampling here (identical math to
  // sample_head_kernel, shared helpers in env_common.h) removes one kernel
  // launch per rollout step from the latency-bound per-step chain.
  if (P.head || P.h2) {
    const int A = K.sample_nact;
    float row_l[8];
    const float* row;
    if (P.h2) {
      // head = h2 @ W3 + b3 computed inline (W3t is 2 KB, L2-resident;
      // the h2 row streams as bf16x8 chunks) — one fewer rollout launch
      const __bf16* h = reinterpret_cast<const __bf16*>(P.h2) +
                        (int64_t)(n - env_lo) * K.head_hidden;
      const __bf16* w = reinterpret_cast<const __bf16*>(P.w3t);
      float s[8];
      for (int j = 0; j <= A; ++j) s[j] = 0.f;
      typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
      for (int k8 = 0; k8 < K.head_hidden; k8 += 8) {
        const bf16x8v hv = *reinterpret_cast<const bf16x8v*>(&h[k8]);
        for (int j = 0; j <= A; ++j) {
          const bf16x8v wv = *reinterpret_cast<const bf16x8v*>(
              &w[(int64_t)j * K.head_hidden + k8]);
          for (int i = 0; i < 8; ++i) s[j] += (float)hv[i] * (float)wv[i];
        }
      }
      for (int j = 0; j <= A; ++j) row_l[j] = s[j] + P.b3[j];
      row = row_l;
    } else {
      row = P.head + (int64_t)(n - env_lo) * (A + 1);
    }
    const float logz = head_logz(row, A);
    uint64_t stp = (uint64_t)K.sample_step;
    if (P.step_base) stp += *P.step_base;
    const int sa = sample_categorical(row, A, logz, K.sample_seed, stp, n);
    P.actions_out[n] = sa;
    P.logp_out[n] = row[sa] - logz;
    P.value_out[n] = row[A];
  }

  // ---- action decode (env.py:343-360) --------------------------------
  float raw;
  int64_t a;
  if (K.flags & F_CONTINUOUS) {
    raw = reinterpret_cast<const float*>(P.actions)[n];
    const double thr = K.cont_threshold != 0.0 ? K.cont_threshold : 0.33;
    a = (raw >= thr) ? 1 : (raw <= -thr) ? 2 : 0;
  } else {
    int64_t v = reinterpret_cast<const int64_t*>(P.actions)[n];
    raw = (float)v;
    a = (v >= 0 && v <= 2) ? v : 0;
  }

  const bool live = !P.terminated[n];

  // ---- event-context overlay (env.py:394-440) ------------------------
  // row = pre-advance bar_index (the reference reads the upcoming row).
  if (K.flags & F_OVERLAY) {
    int row_ov = min(P.cursor[n], EB - 1);
    bool active = P.ev_no_trade[row_ov] >= (float)K.overlay_threshold;
    int psign = P.pos[n] > 0 ? 1 : (P.pos[n] < 0 ? -1 : 0);
    if (live && active) {
      ediag[E_EV_NO_TRADE_ACTIVE_STEPS] += 1;
      bool forced = (K.flags & F_OVERLAY_FF) && psign != 0;
      bool blocked = !forced && (K.flags & F_OVERLAY_BLOCK) && psign == 0 &&
                     (a == 1 || a == 2);
      if (forced) a = 3;
      if (blocked) a = 0;
      if (forced || blocked) ediag[E_EV_ACTION_OVERRIDES] += 1;
      if (blocked) ediag[E_EV_BLOCKED_ENTRIES] += 1;
      if (forced) ediag[E_EV_FORCED_FLAT_ACTIONS] += 1;
    }
  }

  // ---- action diagnostics (env.py:744-761) ---------------------------
  if (live) {
    adiag[A_STEPS] += 1;
    P.raw_abs_sum[n] += fabsf(raw);
    P.raw_min[n] = fminf(P.raw_min[n], raw);
    P.raw_max[n] = fmaxf(P.raw_max[n], raw);
    if (a == 1) { adiag[A_LONG] += 1; adiag[A_NON_HOLD] += 1; }
    else if (a == 2) { adiag[A_SHORT] += 1; adiag[A_NON_HOLD] += 1; }
    else {
      adiag[A_HOLD] += 1;
      if (K.flags & F_CONTINUOUS) adiag[A_DEADBAND] += 1;
    }
  }

  // ---- advance --------------------------------------------------------
  const bool first = live && !P.started[n];
  const bool adv = live && P.started[n];
  bool exhausted = false;
  if (adv) {
    P.cursor[n] += 1;
    if (P.cursor[n] > EB) { exhausted = true; P.cursor[n] = EB; P.terminated[n] = true; }
  }
  const bool valid = adv && !exhausted;
  const int t = max(P.cursor[n] - 1, LO);
  const float o_px = P.open_px[t];
  const float h_px = P.high_px[t];
  const float l_px = P.low_px[t];
  const float c_px = P.close_px[t];

  if (valid || first) P.last_trade_cost[n] = 0.0;

  // ---- 1. pending market fills at open(t) ----------------------------
  // latency_ms >= bar duration holds the order in transit for
  // K.latency_bars extra bars before it reaches the book (LatencyModel
  // semantics at OHLC granularity, nautilus_adapter.py:415-417)
  const bool in_transit = valid && P.pend_wait[n] > 0 &&
                          (P.pend_close[n] || P.pend_open_dir[n] != 0);
  if (in_transit) {
    P.pend_wait[n] -= 1;
  } else {
    if (valid && P.pend_close[n] && P.pos[n] != 0.0) {
      float fill = (P.pos[n] < 0) ? buy_fill(o_px, slip) : sell_fill(o_px, slip);
      close_position(P, K, n, (double)fill, -1);
    }
    if (valid && P.pend_open_dir[n] != 0 && P.pos[n] == 0.0) {
      double dir = (double)P.pend_open_dir[n];
      double fill = (double)(dir > 0 ? buy_fill(o_px, slip) : sell_fill(o_px, slip));
      double size = (double)P.pend_open_size[n];
      double notional = size * fill;
      double comm = notional * K.commission;
      // standard: init-margin fraction of full notional; leveraged:
      // notional / account leverage (Standard-/LeveragedMarginModel,
      // nautilus_adapter.py:371-375)
      double margin = (K.margin_model == MARGIN_STANDARD)
                          ? notional * K.margin_init_rate
                          : notional / K.leverage;
      if ((K.flags & F_PREFLIGHT) && P.cash[n] < margin + comm) {
        // margin preflight denial (nautilus_gym.py:128-171 semantics):
        // the order is dropped, diagnostics incremented, episode continues
        ediag[E_MARGIN_PREFLIGHT_DENIED] += 1;
        goto preflight_denied;
      }
      P.cash[n] -= margin + comm;
      P.margin_used[n] = margin;
      P.commission_paid[n] += comm;
      P.last_trade_cost[n] += comm;
      P.pos[n] = dir * size;
      P.avg_entry[n] = fill;
      if (P.pend_sl[n] > 0.f || P.pend_tp[n] > 0.f) {
        P.br_active[n] = true;
        P.br_armed[n] = true;  // children active from NEXT bar
        P.br_sl[n] = P.pend_sl[n];
        P.br_tp[n] = P.pend_tp[n];
      }
    preflight_denied:;
    }
    P.pend_close[n] = false;
    P.pend_open_dir[n] = 0;
    P.pend_open_size[n] = 0.f;
    P.pend_sl[n] = 0.f;
    P.pend_tp[n] = 0.f;
  }

  // ---- 2. bracket children -------------------------------------------
  // Collision policy decides the intrabar point order when both children
  // lie inside one bar (contracts.py intrabar_collision_policy):
  //   worst_case — the stop has absolute priority (checked on the open
  //                AND the adverse extreme before any take-profit look);
  //   ohlc      — open, high, low, close print order;
  //   adaptive  — up bar assumes the low printed first, down bar the high.
  // The limit_fill_policy governs the take-profit leg: touch fills at >=,
  // cross/conservative need a strict cross, and conservative never takes
  // gap price improvement (fills at the limit price).
  if (valid && P.br_active[n] && !P.br_armed[n] && P.pos[n] != 0.0) {
    const bool is_long = P.pos[n] > 0;
    const float sl = P.br_sl[n], tp = P.br_tp[n];
    const int lim = K.limit_policy;
    float trig = 0.f;
    int hit = 0;  // 0 none, 1 sl, 2 tp
    const auto tp_ge = [&](float px) {  // long take-profit trigger
      return lim == LIM_TOUCH ? px >= tp : px > tp;
    };
    const auto tp_le = [&](float px) {  // short take-profit trigger
      return lim == LIM_TOUCH ? px <= tp : px < tp;
    };
    const float tp_gap_trig = (lim == LIM_CONSERVATIVE) ? tp : o_px;
    if (K.collision_policy == COLL_WORST) {
      if (is_long) {
        if (o_px <= sl) { hit = 1; trig = o_px; }
        else if (l_px <= sl) { hit = 1; trig = sl; }
        else if (tp_ge(o_px)) { hit = 2; trig = tp_gap_trig; }
        else if (tp_ge(h_px)) { hit = 2; trig = tp; }
      } else {
        if (o_px >= sl) { hit = 1; trig = o_px; }
        else if (h_px >= sl) { hit = 1; trig = sl; }
        else if (tp_le(o_px)) { hit = 2; trig = tp_gap_trig; }
        else if (tp_le(l_px)) { hit = 2; trig = tp; }
      }
    } else {
      // point-walk: gap checks at the open (stop first), then the two
      // extremes in policy order; the close is within [low, high] so it
      // can add no new trigger.
      const bool low_first = (K.collision_policy == COLL_ADAPTIVE)
                                 ? (c_px >= o_px)  // up bar: dip printed first
                                 : false;          // ohlc: high before low
      if (is_long && o_px <= sl) { hit = 1; trig = o_px; }
      else if (!is_long && o_px >= sl) { hit = 1; trig = o_px; }
      else if (is_long && tp_ge(o_px)) { hit = 2; trig = tp_gap_trig; }
      else if (!is_long && tp_le(o_px)) { hit = 2; trig = tp_gap_trig; }
      else {
        for (int k = 0; k < 2 && !hit; ++k) {
          if ((k == 0) == low_first) {  // at the bar low
            if (is_long) { if (l_px <= sl) { hit = 1; trig = sl; } }
            else if (tp_le(l_px)) { hit = 2; trig = tp; }
          } else {                      // at the bar high
            if (is_long) { if (tp_ge(h_px)) { hit = 2; trig = tp; } }
            else if (h_px >= sl) { hit = 1; trig = sl; }
          }
        }
      }
    }
    if (hit) {
      float fill = is_long ? sell_fill(trig, slip) : buy_fill(trig, slip);
      close_position(P, K, n, (double)fill,
                     hit == 1 ? E_BRACKET_SL_FILLS : E_BRACKET_TP_FILLS);
    }
  }
  if (valid) P.br_armed[n] = false;

  const bool dec = valid || first;

  // ---- 3. ATR true-range update (direct_atr_sltp.py:143-155) ---------
  float atr = 0.f;
  bool atr_ready = false;
  if (K.strategy_id == STRAT_ATR && dec) {
    const int Pn = K.atr_period;
    float pc = P.prev_close_atr[n];
    float tr = isnan(pc)
                   ? (h_px - l_px)
                   : fmaxf(h_px - l_px, fmaxf(fabsf(h_px - pc), fabsf(l_px - pc)));
    int cnt = P.tr_count[n];
    int idx = cnt % Pn;
    float old = P.tr_ring[(int64_t)n * Pn + idx];
    P.tr_sum[n] += tr - (cnt >= Pn ? old : 0.f);
    P.tr_ring[(int64_t)n * Pn + idx] = tr;
    P.tr_count[n] = cnt + 1;
    P.prev_close_atr[n] = c_px;
    int n_tr = min(P.tr_count[n], Pn);
    atr = P.tr_sum[n] / (float)max(n_tr, 1);
    atr_ready = P.tr_count[n] >= Pn;
  }

  // ---- 4. strategy decision on bar t ---------------------------------
  // `held` = an order already in transit (latency); a NEW decision this
  // bar starts the latency clock, a held order keeps its remaining wait
  // (the strategy may refresh the order fields, the arrival time is set
  // by the first submission).
  const bool held_order = P.pend_close[n] || P.pend_open_dir[n] != 0;
  if (dec && a == 3) {  // overlay force-flat bypasses the strategy
    if (P.pos[n] != 0.0) {
      P.pend_close[n] = true;
      ediag[E_DEFAULT_ORDERS_SUBMITTED] += 1;
      ediag[E_EV_FORCED_FLAT_ORDERS] += 1;
    }
  } else if (dec && K.strategy_id == STRAT_DIRECT) {
    if (a == 1 || a == 2) {
      ediag[E_ENTRY_ACTIONS_SEEN] += 1;
      const double pos = P.pos[n];
      int orders = 0;
      if (a == 1) {
        if (pos < 0) { P.pend_close[n] = true; orders += 1; }
        if (pos <= 0) {
          P.pend_open_dir[n] = 1;
          P.pend_open_size[n] = (float)K.position_size;
          orders += 1;
        }
      } else {
        if (pos > 0) { P.pend_close[n] = true; orders += 1; }
        if (pos >= 0) {
          P.pend_open_dir[n] = -1;
          P.pend_open_size[n] = (float)K.position_size;
          orders += 1;
        }
      }
      ediag[E_DEFAULT_ORDERS_SUBMITTED] += orders;
    }
  } else if (dec && K.strategy_id == STRAT_FIXED) {
    if (a == 1 || a == 2) {
      const double pos = P.pos[n];
      const float pip = P.pip_env ? P.pip_env[n] : (float)K.pip_size;
      const float sl_d = (float)K.sl_pips * pip;
      const float tp_d = (float)K.tp_pips * pip;
      if (a == 1) {
        if (pos < 0) P.pend_close[n] = true;
        if (pos <= 0) {
          P.pend_open_dir[n] = 1;
          P.pend_open_size[n] = (float)K.position_size;
          P.pend_sl[n] = c_px - sl_d;
          P.pend_tp[n] = c_px + tp_d;
          ediag[E_ENTRY_ORDERS_SUBMITTED] += 1;
        }
      } else {
        if (pos > 0) P.pend_close[n] = true;
        if (pos >= 0) {
          P.pend_open_dir[n] = -1;
          P.pend_open_size[n] = (float)K.position_size;
          P.pend_sl[n] = c_px + sl_d;
          P.pend_tp[n] = c_px - tp_d;
          ediag[E_ENTRY_ORDERS_SUBMITTED] += 1;
        }
      }
    }
  } else if (dec && K.strategy_id == STRAT_ATR) {
    const bool sess_on = (K.flags & F_SESSION_FILTER) != 0;
    const bool in_close_zone = sess_on ? P.sess_close[t] : false;
    const bool in_entry_win = sess_on ? P.sess_entry[t] : true;
    if (in_close_zone && P.pos[n] != 0.0) {
      P.pend_close[n] = true;
      ediag[E_SESSION_FORCE_CLOSES] += 1;
    } else if (a == 1 || a == 2) {
      ediag[E_ENTRY_ACTIONS_SEEN] += 1;
      bool ok = true;
      if (sess_on && !in_entry_win) { ediag[E_BLOCKED_SESSION_FILTER] += 1; ok = false; }
      if (ok && !atr_ready) { ediag[E_BLOCKED_ATR_WARMUP] += 1; ok = false; }
      if (ok && atr <= 0.f) { ediag[E_BLOCKED_NON_POSITIVE_ATR] += 1; ok = false; }
      double size = K.position_size;
      if (K.flags & F_HAS_RELVOL) {
        double raw_sz = P.cash[n] * K.rel_volume * K.leverage;
        if (K.size_mode == SIZE_NOTIONAL) raw_sz = c_px > 0.f ? raw_sz / (double)c_px : 0.0;
        size = fmin(fmax(raw_sz, K.min_order_volume), K.max_order_volume);
      }
      if (ok && size <= 0.0) { ediag[E_BLOCKED_NON_POSITIVE_SIZE] += 1; ok = false; }
      if (ok && c_px <= 0.f) { ediag[E_BLOCKED_NON_POSITIVE_PRICE] += 1; ok = false; }
      if (ok) {
        // risk-mode k shrink (direct_atr_sltp.py:263-289)
        double k_sl_eff = K.k_sl, k_tp_eff = K.k_tp;
        if (K.risk_mode != RISK_FIXED) {
          double rel = fmax(0.0, (K.flags & F_HAS_RELVOL) ? K.rel_volume : 0.0);
          if (rel > K.baseline_rel_volume) {
            double prog = fmin(1.0, fmax(0.0, (rel - K.baseline_rel_volume) /
                                                  (K.max_risk_rel_volume -
                                                   K.baseline_rel_volume)));
            k_sl_eff = fmax(K.min_k_sl, K.k_sl * (1.0 - K.sl_shrink_alpha * prog));
            k_tp_eff = K.k_tp * (1.0 - K.tp_shrink_alpha * prog);
          }
          k_tp_eff = fmax(k_tp_eff, k_sl_eff * K.min_rr);
        }
        // f32 arithmetic with f32-cast scalars — op-for-op the torch oracle
        float sl_dist = (float)k_sl_eff * atr;
        float tp_dist = (float)k_tp_eff * atr;
        if (K.risk_mode == RISK_MARGIN && (K.flags & F_HAS_MPLF)) {
          double rel = fmax(0.0, (K.flags & F_HAS_RELVOL) ? K.rel_volume : 0.0);
          double mlf = fmax(0.0, K.mplf);
          if (rel > 0.0 && mlf > 0.0)
            sl_dist = fminf(sl_dist, c_px * (float)(mlf / (rel * K.leverage)));
        }
        if (K.flags & F_HAS_MINFRAC) {
          float floor_d = (float)K.min_sltp_frac * c_px;
          sl_dist = fmaxf(sl_dist, floor_d);
          tp_dist = fmaxf(tp_dist, floor_d);
        }
        if (K.flags & F_HAS_MAXFRAC) {
          float ceil_d = (float)K.max_sltp_frac * c_px;
          sl_dist = fminf(sl_dist, ceil_d);
          tp_dist = fminf(tp_dist, ceil_d);
        }
        if (tp_dist >= c_px) tp_dist = c_px * 0.5f;
        const double pos = P.pos[n];
        if (a == 1) {
          if (pos < 0) P.pend_close[n] = true;
          if (pos <= 0) {
            P.pend_open_dir[n] = 1;
            P.pend_open_size[n] = (float)size;
            P.pend_sl[n] = c_px - sl_dist;
            P.pend_tp[n] = c_px + tp_dist;
            ediag[E_ENTRY_ORDERS_SUBMITTED] += 1;
          }
        } else {
          if (pos > 0) P.pend_close[n] = true;
          if (pos >= 0) {
            P.pend_open_dir[n] = -1;
            P.pend_open_size[n] = (float)size;
            P.pend_sl[n] = c_px + sl_dist;
            P.pend_tp[n] = c_px - tp_dist;
            ediag[E_ENTRY_ORDERS_SUBMITTED] += 1;
          }
        }
      }
    }
  }

  if (dec && !held_order && K.latency_bars > 0 &&
      (P.pend_close[n] || P.pend_open_dir[n] != 0))
    P.pend_wait[n] = K.latency_bars;

  if (first) P.started[n] = true;
  if (dec) P.episode_step[n] += 1;
  // ---- 4b. financing: FX rollover interest at the scheduled bars ------
  if ((K.flags & F_FINANCING) && P.roll_rate && valid && P.pos[n] != 0.0)
    P.cash[n] += P.pos[n] * (double)c_px * (double)P.roll_rate[t];


  // ---- 5. publish (bt_bridge.py:239-248) ------------------------------
  const bool pub = dec;
  double reward = 0.0, base_reward = 0.0, penalty = 0.0;
  if (pub) {
    P.prev_equity[n] = P.equity[n];
    double unreal = P.pos[n] * ((double)c_px - P.avg_entry[n]);
    P.equity[n] = P.cash[n] + P.margin_used[n] + unreal;
    if (P.equity[n] <= K.min_equity) P.terminated[n] = true;
    if (valid && P.cursor[n] >= EB) P.terminated[n] = true;

    // ---- 6. reward ---------------------------------------------------
    const double ic = K.initial_cash != 0.0 ? K.initial_cash : 1.0;
    const double r_step = (P.equity[n] - P.prev_equity[n]) / ic;
    if (K.reward_id == REWARD_SHARPE) {
      // O(1)/step rolling Sharpe: the ring holds the
      // window values only so the evicted term can be subtracted from the
      // f64 running sums; the f32 rescans of round 1 cost ~8% of the env
      // kernel at W=64 (profiles/bench_mlp_sharpe.json).  f64 accumulators
      // make the cancellation in s2 - m*mean^2 benign at FX return scale.
      const int W = K.sharpe_window;
      const int cnt = P.rew_count[n];
      const int slot = cnt % W;
      const float evicted = P.rew_ring[(int64_t)n * W + slot];
      const float nv = (float)r_step;
      P.rew_ring[(int64_t)n * W + slot] = nv;
      P.rew_count[n] = cnt + 1;
      const double s1 = P.rew_s1[n] + ((double)nv - (double)evicted);
      const double s2 = P.rew_s2[n] + ((double)nv * nv - (double)evicted * evicted);
      P.rew_s1[n] = s1;
      P.rew_s2[n] = s2;
      const int m = min(cnt + 1, W);
      if (m >= 2) {
        const double mean = s1 / (double)m;
        const double var = fmax((s2 - (double)m * mean * mean) / (double)(m - 1), 0.0);
        const double stdv = sqrt(var);
        if (stdv > 0.0) base_reward = (mean / stdv) * sqrt(K.annualization);
      }
    } else if (K.reward_id == REWARD_DD) {
      P.peak_equity[n] = fmax(P.peak_equity[n], fmax(P.equity[n], P.prev_equity[n]));
      double dd = P.peak_equity[n] > 0 ? (P.peak_equity[n] - P.equity[n]) / ic : 0.0;
      base_reward = r_step - K.penalty_lambda * dd;
    } else {
      base_reward = r_step * K.reward_scale;
    }

    if ((K.flags & F_STAGEB_PENALTY) && K.fc_pen_coef > 0 && P.force_close) {
      int row_fc = min(P.cursor[n], EB - 1);
      float hours = P.force_close[row_fc * 4 + 1];
      bool in_zone = P.force_close[row_fc * 4 + 2] > 0.f;
      bool in_win = hours >= 0.f && hours <= fmaxf(0.f, (float)K.fc_pen_window_hours);
      int psign = P.pos[n] > 0 ? 1 : (P.pos[n] < 0 ? -1 : 0);
      if ((in_zone || in_win) && psign != 0) penalty = K.fc_pen_coef;
    }
    reward = base_reward - penalty;
    P.episode_return[n] += reward;

    // ---- 7. metrics ---------------------------------------------------
    P.ret_sum[n] += r_step;
    P.ret_sumsq[n] += r_step * r_step;
    P.ret_count[n] += 1;
    P.metric_peak[n] = fmax(P.metric_peak[n], P.equity[n]);
    double dd_money = P.metric_peak[n] - P.equity[n];
    P.max_dd_money[n] = fmax(P.max_dd_money[n], dd_money);
    if (P.metric_peak[n] > 0)
      P.max_dd_pct[n] = fmax(P.max_dd_pct[n], dd_money / P.metric_peak[n] * 100.0);
  }

  P.reward_out[n] = (float)reward;
  P.base_reward_out[n] = (float)base_reward;
  P.penalty_out[n] = (float)penalty;
  P.terminated_out[n] = P.terminated[n];
  P.coerced_out[n] = a;

  // ---- 7b. recurrent-state autoreset (mask_reset fused in) -----------
  // zero the terminated env's LSTM h/c rows here instead of launching a
  // separate mask_reset kernel after every rollout step (the separate
  // launch is pure launch latency: done rows are rare).
  if (P.rnn_h != nullptr && P.terminated[n]) {
    __bf16 *hrow = reinterpret_cast<__bf16 *>(P.rnn_h) +
                   (int64_t)(n - env_lo) * K.rnn_hidden;
    float *crow = P.rnn_c + (int64_t)(n - env_lo) * K.rnn_hidden;
    for (int j = 0; j < K.rnn_hidden; ++j) hrow[j] = (__bf16)0.f;
    for (int j = 0; j < K.rnn_hidden; ++j) crow[j] = 0.f;
  }

  // ---- 8. autoreset (training path; no reference counterpart) --------
  if ((K.flags & F_AUTORESET) && P.terminated[n]) reset_env(P, K, n);
}

__global__ void env_step_kernel(const EnvPtrs P, const EnvParamsK K,
                                const int env_lo, const int env_cnt) {
  const int n = env_lo + blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= env_lo + env_cnt) return;
  env_step_one(P, K, n, env_lo);
}

// ---------------------------------------------------------------------------
// Observation build: elementwise over [N, obs_dim].
// Semantics: envs/reference_step.py build_obs_torch (itself mirroring
// default_preprocessor.py:34-77 / feature_window_preprocessor.py:99-191).
// ---------------------------------------------------------------------------
// Per-(env, feature) z-score stats: mean/std depend only on (step, f), NOT
// on the window row w — computing them once per feature instead of once per
// element drops the p1/p2 prefix-sum traffic by the window factor (32x).
// Encoding in (sm, sd): sd > 0 -> z-score with mean sm / std sd;
// sd == 0 -> raw passthrough (binary or scaling off); sd < 0 -> emit 0
// (fewer than 2 history rows).  Cast points match build_obs_torch exactly.
GFX_DEV void obs_feature_stats(const EnvPtrs& P, const EnvParamsK& K,
                               const int step, const int LO, const int f,
                               float* sm, float* sd) {
  const int F = K.n_features;
  const bool is_binary = P.binary_mask && P.binary_mask[f];
  if (K.scaling_mode == SCALE_NONE || is_binary) {
    *sm = 0.f;
    *sd = 0.f;
    return;
  }
  int hist_left = (K.scaling_mode == SCALE_ROLLING)
                      ? max(step - K.scale_window, LO) : LO;
  int m = step - hist_left;
  if (m < 2) {
    *sm = 0.f;
    *sd = -1.f;
    return;
  }
  double s1 = P.p1[(int64_t)step * F + f] - P.p1[(int64_t)hist_left * F + f];
  double s2 = P.p2[(int64_t)step * F + f] - P.p2[(int64_t)hist_left * F + f];
  double mean = s1 / (double)m;
  double var = s2 / (double)m - mean * mean;
  var = fmax(var, 0.0);
  double stdv = sqrt(var);
  if (stdv < 1e-8) stdv = 1.0;
  // f64 stats cast to f32 BEFORE the subtract/divide — the oracle's
  // exact op order (reference_step.build_obs_torch).
  *sm = (float)mean;
  *sd = (float)stdv;
}

GFX_DEV void build_obs_one(const EnvPtrs& P, const EnvParamsK& K, const int n,
                           const int j, const float* stat_m = nullptr,
                           const float* stat_d = nullptr) {
  const int T = K.T, W = K.window, F = K.n_features;
  {
    const int step = P.cursor[n];
    const int LO = P.lo_bar ? P.lo_bar[n] : 0;
    const int EB = (P.end_bar && P.end_bar[n] > 0) ? P.end_bar[n] : T;
    float val = 0.f;

    if (K.off_features >= 0 && j < K.off_features + W * F && j >= K.off_features) {
      const int q = j - K.off_features;
      const int w = q / F, f = q % F;
      int row = step - W + w;
      row = max(row, LO);
      row = min(row, EB - 1);
      float x = P.features[(int64_t)row * F + f];
      float sm, sd;
      if (stat_m) {
        sm = stat_m[f];
        sd = stat_d[f];
      } else {
        obs_feature_stats(P, K, step, LO, f, &sm, &sd);
      }
      if (sd == 0.f) {
        val = x;
      } else if (sd < 0.f) {
        val = 0.f;
      } else {
        val = (x - sm) / sd;
      }
      // NaN guard BEFORE the clamp (fminf/fmaxf silently drop NaNs on CDNA):
      // torch maps nan->0, +/-inf->+/-clip (reference_step.build_obs_torch).
      if (isnan(val)) val = 0.f;
      if (K.feature_clip > 0) {
        float c = (float)K.feature_clip;
        if (isinf(val)) val = val > 0 ? c : -c;
        val = fminf(fmaxf(val, -c), c);
      } else if (isinf(val)) {
        val = 0.f;
      }
    } else if (K.off_prices >= 0 && j >= K.off_prices && j < K.off_prices + W) {
      const int w = j - K.off_prices;
      int row = min(max(step - W + w, LO), EB - 1);
      val = P.price_px[row];
    } else if (K.off_returns >= 0 && j >= K.off_returns && j < K.off_returns + W) {
      const int w = j - K.off_returns;
      int row = min(max(step - W + w, LO), EB - 1);
      int row_prev = min(max(step - W + w - 1, LO), EB - 1);
      if (w == 0) row_prev = row;
      val = P.price_px[row] - P.price_px[row_prev];
    } else if (K.off_agent >= 0 && j >= K.off_agent && j < K.off_agent + 4) {
      const int q = j - K.off_agent;
      const double ic = K.initial_cash != 0.0 ? K.initial_cash : 1.0;
      if (q == 0) {
        val = P.pos[n] > 0 ? 1.f : (P.pos[n] < 0 ? -1.f : 0.f);
      } else if (q == 1) {
        val = (float)((P.equity[n] - ic) / ic);
      } else if (q == 2) {
        int t_now = min(max(step - 1, LO), EB - 1);
        int row_last = min(max(step - 1, LO), EB - 1);
        float psign = P.pos[n] > 0 ? 1.f : (P.pos[n] < 0 ? -1.f : 0.f);
        float unreal =
            psign * (P.close_px[t_now] - P.price_px[row_last]) * (float)K.position_size;
        val = unreal / (float)ic;
      } else {
        val = fmaxf((float)(EB - step), 0.f) / (float)max(EB - LO, 1);
      }
    } else if (K.off_fc >= 0 && j >= K.off_fc && j < K.off_fc + 4) {
      const int q = j - K.off_fc;
      int row = min(step, EB - 1);
      val = P.force_close ? P.force_close[row * 4 + q] : 0.f;
    } else if (K.off_cal >= 0 && j >= K.off_cal && j < K.off_cal + 11) {
      const int q = j - K.off_cal;
      int row = min(step, EB - 1);
      if (q < 9) val = P.calendar ? P.calendar[row * 10 + q] : 0.f;
      else if (q == 9) val = 0.f;  // margin_closeout_percent
      else {
        const double ic = K.initial_cash != 0.0 ? K.initial_cash : 1.0;
        val = (float)(P.equity[n] / ic);
      }
    }
    P.obs_out[(int64_t)n * K.obs_dim + j] = val;
    if (P.obs_bf16_out)  // bf16 mirror: feeds the policy GEMM directly
      reinterpret_cast<__bf16*>(P.obs_bf16_out)[(int64_t)n * K.obs_dim + j] =
          (__bf16)val;
  }
}

__global__ void build_obs_kernel(const EnvPtrs P, const EnvParamsK K,
                                 const int env_lo, const int env_cnt) {
  const int64_t total = (int64_t)env_cnt * K.obs_dim;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int n = env_lo + (int)(idx / K.obs_dim);
    const int j = (int)(idx % K.obs_dim);
    build_obs_one(P, K, n, j);
  }
}

// One env per block: the per-feature z-score stats are computed once into
// LDS and shared by all window rows (the elementwise kernel re-derived them
// per element — W x more p1/p2 prefix-sum loads, the kernel's dominant HBM
// traffic at 14 us/step).
constexpr int OBS_MAX_F = 64;
__global__ void build_obs_env_kernel(const EnvPtrs P, const EnvParamsK K,
                                     const int env_lo, const int env_cnt) {
  __shared__ float sm[OBS_MAX_F], sd[OBS_MAX_F];
  const int n = env_lo + blockIdx.x;
  if (n >= env_lo + env_cnt) return;  // uniform per block
  const int T = K.T, F = K.n_features;
  const int step = P.cursor[n];
  const int LO = P.lo_bar ? P.lo_bar[n] : 0;
  (void)T;
  for (int f = threadIdx.x; f < F; f += blockDim.x)
    obs_feature_stats(P, K, step, LO, f, &sm[f], &sd[f]);
  __syncthreads();
  for (int j = threadIdx.x; j < K.obs_dim; j += blockDim.x)
    build_obs_one(P, K, n, j, sm, sd);
}

// ---------------------------------------------------------------------------
// Fused step + observation: one env per 64-lane wave.  Lane 0 runs the
// serial per-env step chain, then the whole wave builds that env's
// observation row.  MEASURED SLOWER than the two-launch split at N=4096
// (22.55 vs 22.0 ms/update; a block-per-env variant was worse still at
// 24.4): the obs build is work-bound, and no intra-kernel arrangement
// matches the standalone kernel's full-width parallelism — the saved
// launch (~9 us) does not cover the lost width.  Kept behind fuse_obs
// (default off) with a bitwise-equality GPU test, same as fused_rollout.
// ---------------------------------------------------------------------------
__global__ void env_step_obs_kernel(const EnvPtrs P, const EnvParamsK K,
                                    const int env_lo, const int env_cnt) {
  const int wpb = blockDim.x >> 6;  // waves per block
  const int n = env_lo + blockIdx.x * wpb + ((int)threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const bool valid = n < env_lo + env_cnt;
  if (valid && lane == 0) env_step_one(P, K, n, env_lo);
  __syncthreads();  // step-phase global writes visible before the obs reads
  if (valid)
    for (int j = lane; j < K.obs_dim; j += 64) build_obs_one(P, K, n, j);
}

void launch_env_step(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                     int env_cnt, hipStream_t stream) {
  // 64-thread blocks: the kernel is a per-env dependent-load latency chain;
  // smaller blocks spread the waves over 4x more CUs.
  const int block = 64;
  const int grid = (env_cnt + block - 1) / block;
  hipLaunchKernelGGL(env_step_kernel, dim3(grid), dim3(block), 0, stream, P, K,
                     env_lo, env_cnt);
}

void launch_build_obs(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                      int env_cnt, hipStream_t stream) {
  const int block = 256;
  // per-env LDS-stats kernel whenever the z-score path is active and the
  // feature count fits the LDS stats arrays
  if (K.off_features >= 0 && K.scaling_mode != SCALE_NONE &&
      K.n_features <= OBS_MAX_F) {
    hipLaunchKernelGGL(build_obs_env_kernel, dim3(env_cnt), dim3(block), 0,
                       stream, P, K, env_lo, env_cnt);
    return;
  }
  int64_t total = (int64_t)env_cnt * K.obs_dim;
  int64_t blocks = (total + block - 1) / block;
  int grid = (int)(blocks < 16384 ? blocks : 16384);
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(build_obs_kernel, dim3(grid), dim3(block), 0, stream, P, K,
                     env_lo, env_cnt);
}

void launch_env_step_obs(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                         int env_cnt, hipStream_t stream) {
  const int block = 256;  // 4 envs (waves) per block
  const int wpb = block / 64;
  const int grid = (env_cnt + wpb - 1) / wpb;
  hipLaunchKernelGGL(env_step_obs_kernel, dim3(grid), dim3(block), 0, stream,
                     P, K, env_lo, env_cnt);
}

}  // namespace gymfx
