// pybind11 bindings: GymFxEngine holds raw device pointers + the param pack
// so the per-step hot path does zero dict lookups / tensor re-validation.
// Built in-tree via torch.utils.cpp_extension (hipcc, gfx950).
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <unordered_map>
#include <string>
#include <vector>

#include "env_common.h"

namespace gymfx {

void launch_env_step(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                     int env_cnt, hipStream_t stream);
void launch_build_obs(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                      int env_cnt, hipStream_t stream);
void launch_env_step_obs(const EnvPtrs& P, const EnvParamsK& K, int env_lo,
                         int env_cnt, hipStream_t stream);
void launch_gemm(const void* A, const void* B, const float* bias, void* C,
                 const void* Yact, int M, int N, int K, bool trans_b, int act,
                 bool dact_tanh, bool add_bias, bool accum, hipStream_t stream,
                 const FeistelMap* fmp = nullptr);
void launch_lstm_cell_fwd(const void* gates_pre, const void* gates_h,
                          const float* c_prev, float* c_new, void* h_new,
                          const bool* done, void* h_masked, float* c_masked,
                          void* acts_out, int64_t M, int H,
                          hipStream_t stream);
void launch_lstm_cell_bwd(const void* acts, const float* c_prev,
                          const float* c_new, const void* dh_head,
                          const float* dh_next, const float* dc_next,
                          const bool* done, void* dgates, float* dc_prev,
                          int64_t M, int H, hipStream_t stream);
bool launch_lstm_gemm_cell_fwd(const void* A, const void* B,
                               const void* gates_pre, void* acts_out,
                               const float* c_prev, float* c_new, void* h_new,
                               const bool* done, void* h_masked,
                               float* c_masked, int M, int N, int K,
                               hipStream_t stream);
bool launch_lstm_bwd_fused(const void* acts, const float* c_prev,
                           const float* c_new, const void* dh_head,
                           const float* dh_next, const float* dc_next,
                           const bool* done, const void* B, void* dgates,
                           float* dc_prev, float* dh_prev, int M, int H,
                           hipStream_t stream);
void launch_mask_reset(void* h, float* c, const bool* done, int64_t M, int H,
                       hipStream_t stream);
void launch_masked_state(const void* h_raw, const float* c_raw,
                         const bool* done, void* h_in, float* c_in, int64_t M,
                         int H, hipStream_t stream);
void launch_mb_gather_seq(const void* obs_src, const int64_t* act_src,
                          const float* logp_src, const float* adv_src,
                          const float* ret_src, const bool* done_src,
                          const float* h0_src,
                          const float* c0_src, void* obs_mb, int64_t* act_mb,
                          float* logp_mb, float* adv_mb, float* ret_mb,
                          bool* done_mb,
                          void* h0_mb, float* c0_mb, int Mseq, int L, int D,
                          int H, int N, uint32_t n_seq, int half,
                          uint64_t seed, int minibatches,
                          const unsigned long long* step_base,
                          const unsigned long long* mb_ctr,
                          hipStream_t stream);
void launch_wgrad(const void* X, const void* dY, float* dW_part, float* db_part,
                  float* dW, float* db, int M, int N, int K, int slabs,
                  hipStream_t stream, const FeistelMap* fmp = nullptr);
void launch_gae(const float* rewards, const float* values, const bool* dones,
                float* adv, float* ret, int T, int N, float gamma, float lam,
                hipStream_t stream);
void launch_adam(float* p, const float* g, float* m, float* v, void* p_bf16,
                 int64_t n, float lr, float beta1, float beta2, float eps,
                 float bc1, float bc2, const float* gscale, const int* step_ctr,
                 hipStream_t stream);
void launch_grad_clip(const float* g, int64_t n, float max_norm, float* part,
                      float* scale, int nparts, hipStream_t stream);
void launch_grad_sumsq(const float* g, int64_t n, float* part, int nparts,
                       hipStream_t stream);
void launch_adam_clip(float* p, const float* g, float* m, float* v,
                      void* p_bf16, int64_t n, float lr, float beta1,
                      float beta2, float eps, float bc1, float bc2,
                      const int* step_ctr, const float* clip_part, int nparts,
                      float max_norm, hipStream_t stream);
void launch_sample_head(const float* head, int M, int n_actions, uint64_t seed,
                        uint64_t step, int64_t* actions, float* logp,
                        float* value, float* entropy, int greedy,
                        const unsigned long long* step_base, int row_offset,
                        hipStream_t stream);
void launch_increment_u64(unsigned long long* ctr, unsigned long long delta,
                          hipStream_t stream);
void launch_mlp_policy_rollout(const void* obs, const void* W1t,
                               const float* b1, const void* W2t,
                               const float* b2, const void* W3t,
                               const float* b3, int64_t* actions, float* logp,
                               float* value, int N, int D, int n_actions,
                               uint64_t seed, uint64_t step,
                               const unsigned long long* step_base,
                               int row_offset, int greedy, hipStream_t stream);
void launch_increment_i32(int* ctr, int delta, hipStream_t stream);
void launch_mb_gather(const void* obs_src, const int64_t* act_src,
                      const float* logp_src, const float* adv_src,
                      const float* ret_src, void* obs_mb, int64_t* act_mb,
                      float* logp_mb, float* adv_mb, float* ret_mb, int M,
                      int D, uint32_t n, int half, uint64_t seed,
                      int minibatches, const unsigned long long* step_base,
                      const unsigned long long* mb_ctr, hipStream_t stream,
                      int skip_obs = 0);
void launch_ppo_loss_bwd(const float* head, const int64_t* actions,
                         const float* old_logp, const float* adv,
                         const float* ret, void* dhead, int M, int n_actions,
                         float clip_eps, float ent_coef, float vf_coef,
                         float inv_count, float* losses, hipStream_t stream);
void launch_adv_normalize(float* adv, int64_t n, float* part, int nparts,
                          hipStream_t stream);
void launch_f32_to_bf16(const float* in, void* out, int64_t n,
                        hipStream_t stream);
void launch_transpose_bf16(const void* src, void* dst, int K, int N,
                           hipStream_t stream);

namespace {

using TensorMap = std::unordered_map<std::string, torch::Tensor>;

template <typename T>
T* ptr(TensorMap& m, const char* key, bool required = true) {
  auto it = m.find(key);
  if (it == m.end() || !it->second.defined()) {
    TORCH_CHECK(!required, "GymFxEngine: missing tensor '", key, "'");
    return nullptr;
  }
  return it->second.data_ptr<T>();
}

}  // namespace

struct GymFxEngine {
  EnvPtrs P{};
  EnvParamsK K{};
  TensorMap state, market, outputs;
  bool continuous = false;

  GymFxEngine(const py::dict& iparams, const py::dict& fparams,
              const py::dict& state_d, const py::dict& market_d) {
    for (auto item : state_d)
      state[py::cast<std::string>(item.first)] = py::cast<torch::Tensor>(item.second);
    for (auto item : market_d)
      market[py::cast<std::string>(item.first)] = py::cast<torch::Tensor>(item.second);

    auto gi = [&](const char* k) { return py::cast<int64_t>(iparams[k]); };
    auto gf = [&](const char* k) { return py::cast<double>(fparams[k]); };

    K.n_envs = (int)gi("n_envs");
    K.T = (int)gi("T");
    K.window = (int)gi("window");
    K.n_features = (int)gi("n_features");
    K.reward_id = (int)gi("reward_id");
    K.strategy_id = (int)gi("strategy_id");
    K.prep_id = (int)gi("prep_id");
    K.scaling_mode = (int)gi("scaling_mode");
    K.scale_window = (int)gi("scale_window");
    K.sharpe_window = (int)gi("sharpe_window");
    K.atr_period = (int)gi("atr_period");
    K.size_mode = (int)gi("size_mode");
    K.risk_mode = (int)gi("risk_mode");
    K.collision_policy = (int)gi("collision_policy");
    K.limit_policy = (int)gi("limit_policy");
    K.latency_bars = (int)gi("latency_bars");
    K.margin_model = (int)gi("margin_model");
    K.flags = (int)gi("flags");
    K.obs_dim = (int)gi("obs_dim");
    K.off_features = (int)gi("off_features");
    K.off_prices = (int)gi("off_prices");
    K.off_returns = (int)gi("off_returns");
    K.off_agent = (int)gi("off_agent");
    K.off_fc = (int)gi("off_fc");
    K.off_cal = (int)gi("off_cal");
    continuous = (K.flags & F_CONTINUOUS) != 0;

    K.initial_cash = gf("initial_cash");
    K.position_size = gf("position_size");
    K.commission = gf("commission");
    K.slippage = gf("slippage");
    K.leverage = gf("leverage");
    K.min_equity = gf("min_equity");
    K.cont_threshold = gf("cont_threshold");
    K.reward_scale = gf("reward_scale");
    K.annualization = gf("annualization");
    K.penalty_lambda = gf("penalty_lambda");
    K.sl_pips = gf("sl_pips");
    K.tp_pips = gf("tp_pips");
    K.pip_size = gf("pip_size");
    K.k_sl = gf("k_sl");
    K.k_tp = gf("k_tp");
    K.rel_volume = gf("rel_volume");
    K.min_order_volume = gf("min_order_volume");
    K.max_order_volume = gf("max_order_volume");
    K.min_sltp_frac = gf("min_sltp_frac");
    K.max_sltp_frac = gf("max_sltp_frac");
    K.baseline_rel_volume = gf("baseline_rel_volume");
    K.max_risk_rel_volume = gf("max_risk_rel_volume");
    K.sl_shrink_alpha = gf("sl_shrink_alpha");
    K.tp_shrink_alpha = gf("tp_shrink_alpha");
    K.min_k_sl = gf("min_k_sl");
    K.min_rr = gf("min_rr");
    K.mplf = gf("mplf");
    K.fc_pen_coef = gf("fc_pen_coef");
    K.fc_pen_window_hours = gf("fc_pen_window_hours");
    K.feature_clip = gf("feature_clip");
    K.overlay_threshold = gf("overlay_threshold");
    K.margin_init_rate = gf("margin_init_rate");

    // market pointers
    P.open_px = ptr<float>(market, "open");
    P.high_px = ptr<float>(market, "high");
    P.low_px = ptr<float>(market, "low");
    P.close_px = ptr<float>(market, "close");
    P.price_px = ptr<float>(market, "price");
    P.features = ptr<float>(market, "features", false);
    P.p1 = ptr<double>(market, "feat_prefix1", false);
    P.p2 = ptr<double>(market, "feat_prefix2", false);
    P.binary_mask = ptr<bool>(market, "binary_mask", false);
    P.ev_no_trade = ptr<float>(market, "ev_no_trade");
    P.force_close = ptr<float>(market, "force_close", false);
    P.calendar = ptr<float>(market, "calendar", false);
    P.sess_entry = ptr<bool>(market, "sess_entry");
    P.sess_close = ptr<bool>(market, "sess_close");
    P.roll_rate = ptr<float>(market, "roll_rate", false);

    // state pointers
    P.cursor = ptr<int>(state, "cursor");
    P.started = ptr<bool>(state, "started");
    P.terminated = ptr<bool>(state, "terminated");
    P.pos = ptr<double>(state, "pos");
    P.avg_entry = ptr<double>(state, "avg_entry");
    P.cash = ptr<double>(state, "cash");
    P.margin_used = ptr<double>(state, "margin_used");
    P.equity = ptr<double>(state, "equity");
    P.prev_equity = ptr<double>(state, "prev_equity");
    P.peak_equity = ptr<double>(state, "peak_equity");
    P.commission_paid = ptr<double>(state, "commission_paid");
    P.last_trade_cost = ptr<double>(state, "last_trade_cost");
    P.trade_count = ptr<int>(state, "trade_count");
    P.pend_close = ptr<bool>(state, "pend_close");
    P.pend_open_dir = ptr<int8_t>(state, "pend_open_dir");
    P.pend_open_size = ptr<float>(state, "pend_open_size");
    P.pend_sl = ptr<float>(state, "pend_sl");
    P.pend_tp = ptr<float>(state, "pend_tp");
    P.pend_wait = ptr<int>(state, "pend_wait");
    P.br_active = ptr<bool>(state, "br_active");
    P.br_armed = ptr<bool>(state, "br_armed");
    P.br_sl = ptr<float>(state, "br_sl");
    P.br_tp = ptr<float>(state, "br_tp");
    P.tr_ring = ptr<float>(state, "tr_ring");
    P.tr_count = ptr<int>(state, "tr_count");
    P.tr_sum = ptr<float>(state, "tr_sum");
    P.prev_close_atr = ptr<float>(state, "prev_close_atr");
    P.rew_ring = ptr<float>(state, "rew_ring");
    P.rew_count = ptr<int>(state, "rew_count");
    P.rew_s1 = ptr<double>(state, "rew_s1");
    P.rew_s2 = ptr<double>(state, "rew_s2");
    P.trade_won = ptr<int>(state, "trade_won");
    P.trade_lost = ptr<int>(state, "trade_lost");
    P.trade_pnl_sum = ptr<double>(state, "trade_pnl_sum");
    P.trade_pnl_sumsq = ptr<double>(state, "trade_pnl_sumsq");
    P.metric_peak = ptr<double>(state, "metric_peak");
    P.max_dd_money = ptr<double>(state, "max_dd_money");
    P.max_dd_pct = ptr<double>(state, "max_dd_pct");
    P.ret_sum = ptr<double>(state, "ret_sum");
    P.ret_sumsq = ptr<double>(state, "ret_sumsq");
    P.ret_count = ptr<int>(state, "ret_count");
    P.episode_step = ptr<int>(state, "episode_step");
    P.episode_return = ptr<double>(state, "episode_return");
    P.start_offset = ptr<int>(state, "start_offset");
    P.lo_bar = ptr<int>(state, "lo_bar");
    P.end_bar = ptr<int>(state, "end_bar");
    P.inst_id = ptr<int>(state, "inst_id");
    P.pip_env = ptr<float>(state, "pip_env");
    P.exec_diag = ptr<int>(state, "exec_diag");
    P.act_diag = ptr<int>(state, "act_diag");
    P.raw_abs_sum = ptr<float>(state, "raw_abs_sum");
    P.raw_min = ptr<float>(state, "raw_min");
    P.raw_max = ptr<float>(state, "raw_max");

    // persistent step outputs (allocated once, device)
    auto dev = state["cursor"].device();
    auto f32 = torch::TensorOptions().dtype(torch::kFloat32).device(dev);
    outputs["reward"] = torch::zeros({K.n_envs}, f32);
    outputs["base_reward"] = torch::zeros({K.n_envs}, f32);
    outputs["penalty"] = torch::zeros({K.n_envs}, f32);
    outputs["terminated"] = torch::zeros(
        {K.n_envs}, torch::TensorOptions().dtype(torch::kBool).device(dev));
    outputs["coerced"] = torch::zeros(
        {K.n_envs}, torch::TensorOptions().dtype(torch::kInt64).device(dev));
    P.reward_out = outputs["reward"].data_ptr<float>();
    P.base_reward_out = outputs["base_reward"].data_ptr<float>();
    P.penalty_out = outputs["penalty"].data_ptr<float>();
    P.terminated_out = outputs["terminated"].data_ptr<bool>();
    P.coerced_out = outputs["coerced"].data_ptr<int64_t>();
  }

  py::dict step(torch::Tensor actions, torch::Tensor obs_out,
                c10::optional<torch::Tensor> reward_out,
                c10::optional<torch::Tensor> terminated_out,
                c10::optional<torch::Tensor> obs_bf16_out,
                int64_t env_lo, int64_t env_hi,
                c10::optional<torch::Tensor> head,
                c10::optional<torch::Tensor> logp_out,
                c10::optional<torch::Tensor> value_out,
                c10::optional<torch::Tensor> step_base,
                int64_t sample_seed, int64_t sample_step, bool fuse_obs,
                c10::optional<torch::Tensor> h2,
                c10::optional<torch::Tensor> w3t,
                c10::optional<torch::Tensor> b3,
                c10::optional<torch::Tensor> rnn_h,
                c10::optional<torch::Tensor> rnn_c) {
    TORCH_CHECK(actions.is_cuda() == state["cursor"].is_cuda(),
                "actions must live on the env device");
    TORCH_CHECK(actions.numel() == K.n_envs, "actions numel != n_envs");
    if (env_hi <= 0) env_hi = K.n_envs;
    TORCH_CHECK(0 <= env_lo && env_lo < env_hi && env_hi <= K.n_envs,
                "bad env range");
    TORCH_CHECK(actions.is_contiguous(), "actions must be contiguous");
    if (continuous) {
      TORCH_CHECK(actions.scalar_type() == torch::kFloat32,
                  "continuous mode wants float32 actions");
    } else {
      TORCH_CHECK(actions.scalar_type() == torch::kInt64,
                  "discrete mode wants int64 actions");
    }
    P.actions = actions.data_ptr();
    // optional fused sampling: the step kernel samples from `head` and
    // writes actions/logp/value itself (one fewer launch per step).
    P.head = nullptr;
    P.h2 = nullptr;
    P.w3t = nullptr;
    P.b3 = nullptr;
    P.actions_out = nullptr;
    P.logp_out = nullptr;
    P.value_out = nullptr;
    P.step_base = nullptr;
    P.rnn_h = nullptr;
    P.rnn_c = nullptr;
    K.rnn_hidden = 0;
    if (rnn_h.has_value()) {
      // fused recurrent-state autoreset: the step kernel zeros the
      // terminated env's h/c rows itself (replaces the per-step
      // mask_reset launch in the recurrent rollout)
      TORCH_CHECK(rnn_c.has_value(), "rnn_h needs rnn_c");
      const int64_t cnt = env_hi - env_lo;
      TORCH_CHECK(rnn_h->is_contiguous() &&
                      rnn_h->scalar_type() == torch::kBFloat16 &&
                      rnn_h->dim() == 2 && rnn_h->size(0) == cnt,
                  "rnn_h must be contiguous bf16 [env_cnt, H]");
      TORCH_CHECK(rnn_c->is_contiguous() &&
                      rnn_c->scalar_type() == torch::kFloat32 &&
                      rnn_c->sizes() == rnn_h->sizes(),
                  "rnn_c must be contiguous f32 [env_cnt, H]");
      P.rnn_h = rnn_h->data_ptr();
      P.rnn_c = rnn_c->data_ptr<float>();
      K.rnn_hidden = (int)rnn_h->size(1);
    }
    if (h2.has_value()) {
      // head-in-step fusion: the kernel computes head = h2 @ W3 + b3
      TORCH_CHECK(!continuous, "fused sampling requires discrete actions");
      TORCH_CHECK(!head.has_value(), "pass either head or h2, not both");
      TORCH_CHECK(w3t.has_value() && b3.has_value(), "h2 needs w3t and b3");
      TORCH_CHECK(h2->is_contiguous() &&
                      h2->scalar_type() == torch::kBFloat16 && h2->dim() == 2,
                  "h2 must be contiguous bf16 [env_cnt, H]");
      TORCH_CHECK(w3t->is_contiguous() &&
                      w3t->scalar_type() == torch::kBFloat16 &&
                      w3t->dim() == 2 && w3t->size(1) == h2->size(1),
                  "w3t must be contiguous bf16 [A+1, H]");
      TORCH_CHECK(h2->size(1) % 8 == 0, "H must be a multiple of 8");
      TORCH_CHECK(w3t->size(0) <= 8, "head width must be <= 8");
      TORCH_CHECK(b3->scalar_type() == torch::kFloat32 &&
                      b3->numel() == w3t->size(0),
                  "b3 must be f32 [A+1]");
      TORCH_CHECK(logp_out.has_value() && value_out.has_value(),
                  "fused sampling needs logp_out and value_out");
      P.h2 = h2->data_ptr();
      P.w3t = w3t->data_ptr();
      P.b3 = b3->data_ptr<float>();
      P.actions_out = actions.data_ptr<int64_t>();
      P.logp_out = logp_out->data_ptr<float>();
      P.value_out = value_out->data_ptr<float>();
      K.head_hidden = (int)h2->size(1);
      K.sample_nact = (int)w3t->size(0) - 1;
      K.sample_seed = (unsigned long long)sample_seed;
      K.sample_step = (long long)sample_step;
      if (step_base.has_value()) {
        P.step_base = reinterpret_cast<const unsigned long long*>(
            step_base->data_ptr());
      }
    }
    if (head.has_value()) {
      TORCH_CHECK(!continuous, "fused sampling requires discrete actions");
      TORCH_CHECK(head->is_contiguous() &&
                      head->scalar_type() == torch::kFloat32 &&
                      head->dim() == 2 &&
                      head->size(0) >= (env_hi <= 0 ? K.n_envs : env_hi) - env_lo,
                  "head must be contiguous f32 [env_cnt, A+1]");
      TORCH_CHECK(logp_out.has_value() && value_out.has_value(),
                  "fused sampling needs logp_out and value_out");
      TORCH_CHECK(logp_out->is_contiguous() &&
                      logp_out->scalar_type() == torch::kFloat32 &&
                      logp_out->numel() == K.n_envs,
                  "logp_out must be contiguous f32 [n_envs]");
      TORCH_CHECK(value_out->is_contiguous() &&
                      value_out->scalar_type() == torch::kFloat32 &&
                      value_out->numel() == K.n_envs,
                  "value_out must be contiguous f32 [n_envs]");
      P.head = head->data_ptr<float>();
      P.actions_out = actions.data_ptr<int64_t>();
      P.logp_out = logp_out->data_ptr<float>();
      P.value_out = value_out->data_ptr<float>();
      K.sample_nact = (int)head->size(1) - 1;
      K.sample_seed = (unsigned long long)sample_seed;
      K.sample_step = (long long)sample_step;
      if (step_base.has_value()) {
        TORCH_CHECK(step_base->scalar_type() == torch::kUInt64 ||
                        step_base->scalar_type() == torch::kInt64,
                    "step_base must be a 64-bit counter tensor");
        P.step_base = reinterpret_cast<const unsigned long long*>(
            step_base->data_ptr());
      }
    }
    // optional external outputs (rollout slabs): the kernels write straight
    // into the trainer's [T, N] buffers — no copy kernels afterwards.
    torch::Tensor rew = outputs["reward"];
    if (reward_out.has_value()) {
      TORCH_CHECK(reward_out->is_contiguous() &&
                      reward_out->scalar_type() == torch::kFloat32 &&
                      reward_out->numel() == K.n_envs,
                  "reward_out must be contiguous f32 [n_envs]");
      rew = *reward_out;
    }
    torch::Tensor term = outputs["terminated"];
    if (terminated_out.has_value()) {
      TORCH_CHECK(terminated_out->is_contiguous() &&
                      terminated_out->scalar_type() == torch::kBool &&
                      terminated_out->numel() == K.n_envs,
                  "terminated_out must be contiguous bool [n_envs]");
      term = *terminated_out;
    }
    P.reward_out = rew.data_ptr<float>();
    P.terminated_out = term.data_ptr<bool>();
    hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
    if (fuse_obs) {
      // single fused launch: lane-0 step chain + wave-wide obs build
      set_obs_ptrs(obs_out, obs_bf16_out);
      launch_env_step_obs(P, K, (int)env_lo, (int)(env_hi - env_lo), stream);
    } else {
      launch_env_step(P, K, (int)env_lo, (int)(env_hi - env_lo), stream);
      build_obs(obs_out, obs_bf16_out, env_lo, env_hi);
    }
    py::dict out;
    out["reward"] = rew;
    out["base_reward"] = outputs["base_reward"];
    out["force_close_reward_penalty"] = outputs["penalty"];
    out["terminated"] = term;
    out["coerced_action"] = outputs["coerced"];
    return out;
  }

  void set_obs_ptrs(torch::Tensor& obs_out,
                    c10::optional<torch::Tensor>& obs_bf16_out) {
    TORCH_CHECK(obs_out.is_contiguous() && obs_out.scalar_type() == torch::kFloat32,
                "obs_out must be contiguous f32");
    TORCH_CHECK(obs_out.numel() == (int64_t)K.n_envs * K.obs_dim, "obs_out shape");
    P.obs_out = obs_out.data_ptr<float>();
    P.obs_bf16_out = nullptr;
    if (obs_bf16_out.has_value()) {
      TORCH_CHECK(obs_bf16_out->is_contiguous() &&
                      obs_bf16_out->scalar_type() == torch::kBFloat16 &&
                      obs_bf16_out->numel() == (int64_t)K.n_envs * K.obs_dim,
                  "obs_bf16_out must be contiguous bf16 [n_envs, obs_dim]");
      P.obs_bf16_out = obs_bf16_out->data_ptr();
    }
  }

  void build_obs(torch::Tensor obs_out,
                 c10::optional<torch::Tensor> obs_bf16_out,
                 int64_t env_lo = 0, int64_t env_hi = 0) {
    if (env_hi <= 0) env_hi = K.n_envs;
    set_obs_ptrs(obs_out, obs_bf16_out);
    hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
    launch_build_obs(P, K, (int)env_lo, (int)(env_hi - env_lo), stream);
  }
};

}  // namespace gymfx

// ---------------------------------------------------------------------------
// PPO op wrappers (free functions)
// ---------------------------------------------------------------------------
namespace {

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous() && t.scalar_type() == torch::kBFloat16,
              name, " must be contiguous bf16");
}

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous() && t.scalar_type() == torch::kFloat32,
              name, " must be contiguous f32");
}

// build a FeistelMap from the python-side dict-args of the gather fusion
static gymfx::FeistelMap make_fmap(int64_t n_rows, int M_mb, int64_t seed,
                                   int64_t minibatches, int64_t ctr_off,
                                   torch::Tensor step_base,
                                   torch::Tensor mb_ctr) {
  gymfx::FeistelMap fm;
  fm.n = (uint32_t)n_rows;
  int bits = 2;
  while ((1ll << bits) < n_rows) bits += 2;
  fm.half = bits / 2;
  fm.seed = (uint64_t)seed;
  fm.minibatches = (int)minibatches;
  fm.M_mb = M_mb;
  fm.ctr_off = (long long)ctr_off;
  fm.step_base = reinterpret_cast<const unsigned long long*>(
      step_base.data_ptr());
  fm.mb_ctr = reinterpret_cast<const unsigned long long*>(mb_ctr.data_ptr());
  return fm;
}

void gemm_op(torch::Tensor A, torch::Tensor B, c10::optional<torch::Tensor> bias,
             torch::Tensor C, c10::optional<torch::Tensor> Yact, bool trans_b,
             int64_t act, bool dact_tanh, bool accum,
             int64_t ap_seed, int64_t ap_minibatches, int64_t ap_ctr_off,
             c10::optional<torch::Tensor> ap_step_base,
             c10::optional<torch::Tensor> ap_mb_ctr) {
  TORCH_CHECK(!accum || (act == 0 && !dact_tanh && !bias.has_value()),
              "accum only supported for plain f32-out gemm");
  check_bf16(A, "A");
  check_bf16(B, "B");
  const int M = (int)A.size(0), K = (int)A.size(1);
  const int N = trans_b ? (int)B.size(0) : (int)B.size(1);
  TORCH_CHECK((trans_b ? B.size(1) : B.size(0)) == K, "gemm K mismatch");
  TORCH_CHECK(C.size(1) == N && (C.size(0) == M || ap_step_base.has_value()),
              "gemm C shape");
  if (act == 0) check_f32(C, "C"); else check_bf16(C, "C");
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    check_f32(*bias, "bias");
    TORCH_CHECK(bias->numel() == N, "bias numel");
    bias_p = bias->data_ptr<float>();
  }
  const void* y_p = nullptr;
  if (dact_tanh) {
    TORCH_CHECK(Yact.has_value(), "dact_tanh needs Yact");
    check_bf16(*Yact, "Yact");
    y_p = Yact->data_ptr();
  }
  if (ap_step_base.has_value()) {
    TORCH_CHECK(ap_mb_ctr.has_value(), "a_perm needs mb_ctr");
    const int M_mb = (int)C.size(0);
    gymfx::FeistelMap fm = make_fmap(A.size(0), M_mb, ap_seed,
                                     ap_minibatches, ap_ctr_off,
                                     *ap_step_base, *ap_mb_ctr);
    gymfx::launch_gemm(A.data_ptr(), B.data_ptr(), bias_p, C.data_ptr(), y_p,
                       M_mb, N, K, trans_b, (int)act, dact_tanh,
                       bias_p != nullptr, accum, cur_stream(), &fm);
    return;
  }
  gymfx::launch_gemm(A.data_ptr(), B.data_ptr(), bias_p, C.data_ptr(), y_p, M,
                     N, K, trans_b, (int)act, dact_tanh, bias_p != nullptr,
                     accum, cur_stream());
}

void lstm_cell_fwd_op(torch::Tensor gates_pre,
                      c10::optional<torch::Tensor> gates_h,
                      torch::Tensor c_prev,
                      torch::Tensor c_new, torch::Tensor h_new,
                      c10::optional<torch::Tensor> done,
                      c10::optional<torch::Tensor> h_masked,
                      c10::optional<torch::Tensor> c_masked,
                      c10::optional<torch::Tensor> acts_out) {
  check_bf16(gates_pre, "gates_pre");
  check_f32(c_prev, "c_prev");
  check_f32(c_new, "c_new");
  check_bf16(h_new, "h_new");
  const int64_t M = c_prev.size(0);
  const int H = (int)c_prev.size(1);
  TORCH_CHECK(gates_pre.size(0) == M && gates_pre.size(1) == 4 * H,
              "gates_pre shape");
  const void* gh = nullptr;
  if (gates_h.has_value()) {
    check_bf16(*gates_h, "gates_h");
    gh = gates_h->data_ptr();
  }
  const bool* done_p = nullptr;
  void* hm_p = nullptr;
  float* cm_p = nullptr;
  if (done.has_value()) {
    TORCH_CHECK(h_masked.has_value() && c_masked.has_value(),
                "done needs h_masked and c_masked outputs");
    TORCH_CHECK(done->scalar_type() == torch::kBool && done->numel() == M,
                "done must be bool [M]");
    check_bf16(*h_masked, "h_masked");
    check_f32(*c_masked, "c_masked");
    done_p = done->data_ptr<bool>();
    hm_p = h_masked->data_ptr();
    cm_p = c_masked->data_ptr<float>();
  }
  void* acts_p = nullptr;
  if (acts_out.has_value()) {
    check_bf16(*acts_out, "acts_out");
    acts_p = acts_out->data_ptr();
  }
  gymfx::launch_lstm_cell_fwd(gates_pre.data_ptr(), gh,
                              c_prev.data_ptr<float>(),
                              c_new.data_ptr<float>(), h_new.data_ptr(),
                              done_p, hm_p, cm_p, acts_p, M, H,
                              cur_stream());
}

void lstm_cell_bwd_op(torch::Tensor acts, torch::Tensor c_prev,
                      torch::Tensor c_new, torch::Tensor dh_head,
                      c10::optional<torch::Tensor> dh_next,
                      c10::optional<torch::Tensor> dc_next,
                      c10::optional<torch::Tensor> done,
                      torch::Tensor dgates, torch::Tensor dc_prev) {
  check_bf16(acts, "acts");
  check_f32(c_prev, "c_prev");
  check_f32(c_new, "c_new");
  check_bf16(dh_head, "dh_head");
  check_bf16(dgates, "dgates");
  check_f32(dc_prev, "dc_prev");
  const int64_t M = c_prev.size(0);
  const int H = (int)c_prev.size(1);
  TORCH_CHECK(acts.size(0) == M && acts.size(1) == 4 * H, "acts shape");
  TORCH_CHECK(H % 4 == 0, "H must be a multiple of 4");
  const float* dhn = dh_next.has_value() ? dh_next->data_ptr<float>() : nullptr;
  const float* dcn = dc_next.has_value() ? dc_next->data_ptr<float>() : nullptr;
  const bool* dn = done.has_value() ? done->data_ptr<bool>() : nullptr;
  gymfx::launch_lstm_cell_bwd(
      acts.data_ptr(), c_prev.data_ptr<float>(),
      c_new.data_ptr<float>(), dh_head.data_ptr(), dhn, dcn, dn,
      dgates.data_ptr(), dc_prev.data_ptr<float>(), M, H, cur_stream());
}

bool lstm_gemm_cell_fwd_op(torch::Tensor A, torch::Tensor B,
                           torch::Tensor gates_pre,
                           c10::optional<torch::Tensor> acts_out,
                           torch::Tensor c_prev, torch::Tensor c_new,
                           torch::Tensor h_new,
                           c10::optional<torch::Tensor> done,
                           c10::optional<torch::Tensor> h_masked,
                           c10::optional<torch::Tensor> c_masked) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  check_bf16(gates_pre, "gates_pre");
  check_f32(c_prev, "c_prev");
  check_f32(c_new, "c_new");
  check_bf16(h_new, "h_new");
  const int M = (int)A.size(0);
  const int K = (int)A.size(1);
  const int N = (int)B.size(0);
  TORCH_CHECK(B.size(1) == K, "B must be Wh^T [4H, H]");
  TORCH_CHECK(N == 4 * K, "N must equal 4*H");
  TORCH_CHECK(gates_pre.size(0) == M && gates_pre.size(1) == N,
              "gates_pre shape");
  void* acts_p = nullptr;
  if (acts_out.has_value()) {
    check_bf16(*acts_out, "acts_out");
    acts_p = acts_out->data_ptr();
  }
  const bool* done_p = nullptr;
  void* hm_p = nullptr;
  float* cm_p = nullptr;
  if (done.has_value()) {
    TORCH_CHECK(h_masked.has_value() && c_masked.has_value(),
                "done needs h_masked and c_masked outputs");
    done_p = done->data_ptr<bool>();
    hm_p = h_masked->data_ptr();
    cm_p = c_masked->data_ptr<float>();
  }
  return gymfx::launch_lstm_gemm_cell_fwd(
      A.data_ptr(), B.data_ptr(), gates_pre.data_ptr(), acts_p,
      c_prev.data_ptr<float>(), c_new.data_ptr<float>(), h_new.data_ptr(),
      done_p, hm_p, cm_p, M, N, K, cur_stream());
}

bool lstm_bwd_fused_op(torch::Tensor acts, torch::Tensor c_prev,
                       torch::Tensor c_new, torch::Tensor dh_head,
                       c10::optional<torch::Tensor> dh_next,
                       c10::optional<torch::Tensor> dc_next,
                       c10::optional<torch::Tensor> done, torch::Tensor B,
                       torch::Tensor dgates, torch::Tensor dc_prev,
                       c10::optional<torch::Tensor> dh_prev) {
  check_bf16(acts, "acts");
  check_f32(c_prev, "c_prev");
  check_f32(c_new, "c_new");
  check_bf16(dh_head, "dh_head");
  check_bf16(B, "B");
  check_bf16(dgates, "dgates");
  check_f32(dc_prev, "dc_prev");
  const int M = (int)c_prev.size(0);
  const int H = (int)c_prev.size(1);
  TORCH_CHECK(B.size(0) == H && B.size(1) == 4 * H, "B must be Wh [H, 4H]");
  const float* dhn = dh_next.has_value() ? dh_next->data_ptr<float>() : nullptr;
  const float* dcn = dc_next.has_value() ? dc_next->data_ptr<float>() : nullptr;
  const bool* dn = done.has_value() ? done->data_ptr<bool>() : nullptr;
  float* dhp = dh_prev.has_value() ? dh_prev->data_ptr<float>() : nullptr;
  return gymfx::launch_lstm_bwd_fused(
      acts.data_ptr(), c_prev.data_ptr<float>(),
      c_new.data_ptr<float>(), dh_head.data_ptr(), dhn, dcn, dn,
      B.data_ptr(), dgates.data_ptr(), dc_prev.data_ptr<float>(), dhp, M, H,
      cur_stream());
}

void mask_reset_op(torch::Tensor h, torch::Tensor c, torch::Tensor done) {
  check_bf16(h, "h");
  check_f32(c, "c");
  const int64_t M = c.size(0);
  const int H = (int)c.size(1);
  gymfx::launch_mask_reset(h.data_ptr(), c.data_ptr<float>(),
                           done.data_ptr<bool>(), M, H, cur_stream());
}

void masked_state_op(torch::Tensor h_raw, torch::Tensor c_raw,
                     torch::Tensor done, torch::Tensor h_in,
                     torch::Tensor c_in) {
  check_bf16(h_raw, "h_raw");
  check_f32(c_raw, "c_raw");
  check_bf16(h_in, "h_in");
  check_f32(c_in, "c_in");
  const int64_t M = c_raw.size(0);
  const int H = (int)c_raw.size(1);
  gymfx::launch_masked_state(h_raw.data_ptr(), c_raw.data_ptr<float>(),
                             done.data_ptr<bool>(), h_in.data_ptr(),
                             c_in.data_ptr<float>(), M, H, cur_stream());
}

void mb_gather_seq_op(torch::Tensor obs_src, torch::Tensor act_src,
                      torch::Tensor logp_src, torch::Tensor adv_src,
                      torch::Tensor ret_src, torch::Tensor done_src,
                      torch::Tensor h0_src,
                      torch::Tensor c0_src, torch::Tensor obs_mb,
                      torch::Tensor act_mb, torch::Tensor logp_mb,
                      torch::Tensor adv_mb, torch::Tensor ret_mb,
                      torch::Tensor done_mb,
                      torch::Tensor h0_mb, torch::Tensor c0_mb, int64_t L,
                      int64_t N, int64_t seed, int64_t minibatches,
                      torch::Tensor step_base, torch::Tensor mb_ctr) {
  check_bf16(obs_src, "obs_src");
  check_bf16(obs_mb, "obs_mb");
  check_bf16(h0_mb, "h0_mb");
  check_f32(h0_src, "h0_src");
  check_f32(c0_src, "c0_src");
  check_f32(c0_mb, "c0_mb");
  const int D = (int)obs_src.size(-1);
  const int H = (int)c0_mb.size(-1);
  const int Mseq = (int)c0_mb.size(0);
  const int64_t n_chunks = h0_src.numel() / ((int64_t)N * H);
  const int64_t n_seq = n_chunks * N;
  TORCH_CHECK((int64_t)Mseq * minibatches == n_seq,
              "Mseq * minibatches != n_seq");
  int bits = 2;
  while ((1ll << bits) < n_seq) bits += 2;
  gymfx::launch_mb_gather_seq(
      obs_src.data_ptr(), act_src.data_ptr<int64_t>(),
      logp_src.data_ptr<float>(), adv_src.data_ptr<float>(),
      ret_src.data_ptr<float>(), done_src.data_ptr<bool>(),
      h0_src.data_ptr<float>(),
      c0_src.data_ptr<float>(), obs_mb.data_ptr(),
      act_mb.data_ptr<int64_t>(), logp_mb.data_ptr<float>(),
      adv_mb.data_ptr<float>(), ret_mb.data_ptr<float>(),
      done_mb.data_ptr<bool>(), h0_mb.data_ptr(),
      c0_mb.data_ptr<float>(), Mseq, (int)L, D, H, (int)N, (uint32_t)n_seq,
      bits / 2, (uint64_t)seed, (int)minibatches,
      reinterpret_cast<const unsigned long long*>(step_base.data_ptr()),
      reinterpret_cast<const unsigned long long*>(mb_ctr.data_ptr()),
      cur_stream());
}



void wgrad_op(torch::Tensor X, torch::Tensor dY, torch::Tensor dW_part,
              c10::optional<torch::Tensor> db_part, torch::Tensor dW,
              c10::optional<torch::Tensor> db, int64_t slabs,
              int64_t ap_seed, int64_t ap_minibatches, int64_t ap_ctr_off,
              c10::optional<torch::Tensor> ap_step_base,
              c10::optional<torch::Tensor> ap_mb_ctr) {
  check_bf16(X, "X");
  check_bf16(dY, "dY");
  check_f32(dW, "dW");
  check_f32(dW_part, "dW_part");
  const bool aperm = ap_step_base.has_value();
  // gather fusion: X is the FULL rollout slab, rows selected through the
  // permutation; the logical reduction length is the minibatch (dY rows)
  const int M = aperm ? (int)dY.size(0) : (int)X.size(0);
  const int K = (int)X.size(1), N = (int)dY.size(1);
  TORCH_CHECK(dY.size(0) == M, "wgrad M mismatch");
  TORCH_CHECK(dW.size(0) == K && dW.size(1) == N, "dW shape");
  TORCH_CHECK(dW_part.numel() >= slabs * (int64_t)K * N, "dW_part too small");
  float* dbp = nullptr;
  float* dbo = nullptr;
  if (db_part.has_value() && db.has_value()) {
    check_f32(*db_part, "db_part");
    check_f32(*db, "db");
    dbp = db_part->data_ptr<float>();
    dbo = db->data_ptr<float>();
  }
  if (aperm) {
    TORCH_CHECK(ap_mb_ctr.has_value(), "a_perm needs mb_ctr");
    gymfx::FeistelMap fm = make_fmap(X.size(0), M, ap_seed, ap_minibatches,
                                     ap_ctr_off, *ap_step_base, *ap_mb_ctr);
    gymfx::launch_wgrad(X.data_ptr(), dY.data_ptr(),
                        dW_part.data_ptr<float>(), dbp, dW.data_ptr<float>(),
                        dbo, M, N, K, (int)slabs, cur_stream(), &fm);
    return;
  }
  gymfx::launch_wgrad(X.data_ptr(), dY.data_ptr(), dW_part.data_ptr<float>(),
                      dbp, dW.data_ptr<float>(), dbo, M, N, K, (int)slabs,
                      cur_stream());
}

void gae_op(torch::Tensor rewards, torch::Tensor values, torch::Tensor dones,
            torch::Tensor adv, torch::Tensor ret, double gamma, double lam) {
  check_f32(rewards, "rewards");
  check_f32(values, "values");
  check_f32(adv, "adv");
  check_f32(ret, "ret");
  TORCH_CHECK(dones.scalar_type() == torch::kBool && dones.is_contiguous());
  const int T = (int)rewards.size(0), N = (int)rewards.size(1);
  TORCH_CHECK(values.size(0) == T + 1 && values.size(1) == N, "values shape");
  gymfx::launch_gae(rewards.data_ptr<float>(), values.data_ptr<float>(),
                    dones.data_ptr<bool>(), adv.data_ptr<float>(),
                    ret.data_ptr<float>(), T, N, (float)gamma, (float)lam,
                    cur_stream());
}

void adam_op(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
             c10::optional<torch::Tensor> p_bf16, double lr, double beta1,
             double beta2, double eps, int64_t step,
             c10::optional<torch::Tensor> gscale,
             c10::optional<torch::Tensor> step_ctr,
             c10::optional<torch::Tensor> clip_part, double clip_max_norm) {
  check_f32(p, "p");
  check_f32(g, "g");
  check_f32(m, "m");
  check_f32(v, "v");
  const int64_t n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  void* pb = nullptr;
  if (p_bf16.has_value()) {
    check_bf16(*p_bf16, "p_bf16");
    pb = p_bf16->data_ptr();
  }
  const float* gs = nullptr;
  if (gscale.has_value()) gs = gscale->data_ptr<float>();
  const int* sc = nullptr;
  if (step_ctr.has_value()) {
    TORCH_CHECK(step_ctr->scalar_type() == torch::kInt32, "step_ctr int32");
    sc = step_ctr->data_ptr<int>();
  }
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  if (clip_part.has_value() && clip_max_norm > 0) {
    // fused clip: sumsq partials, then adam derives the scale per block
    check_f32(*clip_part, "clip_part");
    float* part = clip_part->data_ptr<float>();
    const int nparts = (int)clip_part->numel();
    gymfx::launch_grad_sumsq(g.data_ptr<float>(), n, part, nparts,
                             cur_stream());
    gymfx::launch_adam_clip(p.data_ptr<float>(), g.data_ptr<float>(),
                            m.data_ptr<float>(), v.data_ptr<float>(), pb, n,
                            (float)lr, (float)beta1, (float)beta2, (float)eps,
                            bc1, bc2, sc, part, nparts, (float)clip_max_norm,
                            cur_stream());
    return;
  }
  gymfx::launch_adam(p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), pb, n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, bc1, bc2, gs, sc,
                     cur_stream());
}

void grad_clip_op(torch::Tensor g, double max_norm, torch::Tensor part,
                  torch::Tensor scale) {
  check_f32(g, "g");
  check_f32(part, "part");
  check_f32(scale, "scale");
  gymfx::launch_grad_clip(g.data_ptr<float>(), g.numel(), (float)max_norm,
                          part.data_ptr<float>(), scale.data_ptr<float>(),
                          (int)part.numel(), cur_stream());
}

void sample_head_op(torch::Tensor head, int64_t seed, int64_t step,
                    torch::Tensor actions, torch::Tensor logp,
                    c10::optional<torch::Tensor> value,
                    c10::optional<torch::Tensor> entropy, bool greedy,
                    c10::optional<torch::Tensor> step_base,
                    int64_t row_offset) {
  check_f32(head, "head");
  const int M = (int)head.size(0);
  const int n_actions = (int)head.size(1) - 1;
  TORCH_CHECK(actions.scalar_type() == torch::kInt64);
  const unsigned long long* sb = nullptr;
  if (step_base.has_value()) {
    TORCH_CHECK(step_base->scalar_type() == torch::kUInt64 ||
                    step_base->scalar_type() == torch::kInt64,
                "step_base must be a 64-bit counter tensor");
    sb = reinterpret_cast<const unsigned long long*>(step_base->data_ptr());
  }
  gymfx::launch_sample_head(
      head.data_ptr<float>(), M, n_actions, (uint64_t)seed, (uint64_t)step,
      actions.data_ptr<int64_t>(), logp.data_ptr<float>(),
      value.has_value() ? value->data_ptr<float>() : nullptr,
      entropy.has_value() ? entropy->data_ptr<float>() : nullptr,
      greedy ? 1 : 0, sb, (int)row_offset, cur_stream());
}

void mlp_policy_rollout_op(torch::Tensor obs, torch::Tensor W1t,
                           torch::Tensor b1, torch::Tensor W2t,
                           torch::Tensor b2, torch::Tensor W3t,
                           torch::Tensor b3, torch::Tensor actions,
                           torch::Tensor logp, torch::Tensor value,
                           int64_t seed, int64_t step,
                           c10::optional<torch::Tensor> step_base,
                           int64_t row_offset, bool greedy) {
  check_bf16(obs, "obs");
  check_bf16(W1t, "W1t");
  check_bf16(W2t, "W2t");
  check_bf16(W3t, "W3t");
  const int N = (int)obs.size(0);
  const int D = (int)obs.size(1);
  const int H = (int)W1t.size(0);
  const int head_dim = (int)W3t.size(0);
  TORCH_CHECK(H == 256 && W2t.size(0) == 256 && W2t.size(1) == 256,
              "fused rollout requires hidden == 256");
  TORCH_CHECK(W1t.size(1) == D, "W1t shape");
  TORCH_CHECK(head_dim <= 16 && W3t.size(1) == 256, "W3t shape");
  const unsigned long long* sb = nullptr;
  if (step_base.has_value())
    sb = reinterpret_cast<const unsigned long long*>(step_base->data_ptr());
  gymfx::launch_mlp_policy_rollout(
      obs.data_ptr(), W1t.data_ptr(), b1.data_ptr<float>(), W2t.data_ptr(),
      b2.data_ptr<float>(), W3t.data_ptr(), b3.data_ptr<float>(),
      actions.data_ptr<int64_t>(), logp.data_ptr<float>(),
      value.data_ptr<float>(), N, D, head_dim - 1, (uint64_t)seed,
      (uint64_t)step, sb, (int)row_offset, greedy, cur_stream());
}

void increment_counter_op(torch::Tensor ctr, int64_t delta) {
  TORCH_CHECK(ctr.numel() == 1, "counter must be scalar");
  if (ctr.scalar_type() == torch::kInt32) {
    gymfx::launch_increment_i32(ctr.data_ptr<int>(), (int)delta, cur_stream());
  } else {
    TORCH_CHECK(ctr.scalar_type() == torch::kUInt64 ||
                    ctr.scalar_type() == torch::kInt64,
                "counter must be i32/i64/u64");
    gymfx::launch_increment_u64(
        reinterpret_cast<unsigned long long*>(ctr.data_ptr()),
        (unsigned long long)delta, cur_stream());
  }
}

void mb_gather_op(torch::Tensor obs_src, torch::Tensor act_src,
                  torch::Tensor logp_src, torch::Tensor adv_src,
                  torch::Tensor ret_src, torch::Tensor obs_mb,
                  torch::Tensor act_mb, torch::Tensor logp_mb,
                  torch::Tensor adv_mb, torch::Tensor ret_mb, int64_t seed,
                  int64_t minibatches, torch::Tensor step_base,
                  torch::Tensor mb_ctr, bool skip_obs) {
  check_bf16(obs_src, "obs_src");
  check_bf16(obs_mb, "obs_mb");
  const int64_t n = obs_src.size(0);
  const int D = (int)obs_src.size(1);
  const int M = (int)obs_mb.size(0);
  TORCH_CHECK(obs_mb.size(1) == D, "obs_mb D mismatch");
  TORCH_CHECK(act_src.numel() == n && logp_src.numel() == n &&
                  adv_src.numel() == n && ret_src.numel() == n,
              "src numel mismatch");
  TORCH_CHECK(act_mb.numel() == M && logp_mb.numel() == M &&
                  adv_mb.numel() == M && ret_mb.numel() == M,
              "mb numel mismatch");
  TORCH_CHECK((int64_t)M * minibatches == n, "M * minibatches != n");
  // smallest even-bit domain 2^(2*half) >= n
  int bits = 2;
  while ((1ll << bits) < n) bits += 2;
  gymfx::launch_mb_gather(
      obs_src.data_ptr(), act_src.data_ptr<int64_t>(),
      logp_src.data_ptr<float>(), adv_src.data_ptr<float>(),
      ret_src.data_ptr<float>(), obs_mb.data_ptr(),
      act_mb.data_ptr<int64_t>(), logp_mb.data_ptr<float>(),
      adv_mb.data_ptr<float>(), ret_mb.data_ptr<float>(), M, D, (uint32_t)n,
      bits / 2, (uint64_t)seed, (int)minibatches,
      reinterpret_cast<const unsigned long long*>(step_base.data_ptr()),
      reinterpret_cast<const unsigned long long*>(mb_ctr.data_ptr()),
      cur_stream(), skip_obs ? 1 : 0);
}

void ppo_loss_bwd_op(torch::Tensor head, torch::Tensor actions,
                     torch::Tensor old_logp, torch::Tensor adv,
                     torch::Tensor ret, torch::Tensor dhead, double clip_eps,
                     double ent_coef, double vf_coef, double inv_count,
                     c10::optional<torch::Tensor> losses) {
  check_f32(head, "head");
  check_bf16(dhead, "dhead");
  const int M = (int)head.size(0);
  const int n_actions = (int)head.size(1) - 1;
  gymfx::launch_ppo_loss_bwd(
      head.data_ptr<float>(), actions.data_ptr<int64_t>(),
      old_logp.data_ptr<float>(), adv.data_ptr<float>(), ret.data_ptr<float>(),
      dhead.data_ptr(), M, n_actions, (float)clip_eps, (float)ent_coef,
      (float)vf_coef, (float)inv_count,
      losses.has_value() ? losses->data_ptr<float>() : nullptr, cur_stream());
}

void adv_normalize_op(torch::Tensor adv, torch::Tensor part) {
  check_f32(adv, "adv");
  check_f32(part, "part");
  gymfx::launch_adv_normalize(adv.data_ptr<float>(), adv.numel(),
                              part.data_ptr<float>(), (int)(part.numel() / 2),
                              cur_stream());
}

void f32_to_bf16_op(torch::Tensor in, torch::Tensor out) {
  check_f32(in, "in");
  check_bf16(out, "out");
  gymfx::launch_f32_to_bf16(in.data_ptr<float>(), out.data_ptr(), in.numel(),
                            cur_stream());
}

void transpose_bf16_op(torch::Tensor src, torch::Tensor dst) {
  check_bf16(src, "src");
  check_bf16(dst, "dst");
  const int K = (int)src.size(0), N = (int)src.size(1);
  TORCH_CHECK(dst.size(0) == N && dst.size(1) == K, "transpose shape");
  gymfx::launch_transpose_bf16(src.data_ptr(), dst.data_ptr(), K, N,
                               cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm", &gemm_op,
        "C = act(A @ B + bias); trans_b reads B as [N,K]; act 0=f32 1=bf16 "
        "2=tanh-bf16; dact_tanh multiplies by (1-Yact^2); accum does C += ",
        py::arg("A"), py::arg("B"), py::arg("bias"), py::arg("C"),
        py::arg("Yact") = py::none(), py::arg("trans_b") = false,
        py::arg("act") = 1, py::arg("dact_tanh") = false,
        py::arg("accum") = false, py::arg("ap_seed") = 0,
        py::arg("ap_minibatches") = 1, py::arg("ap_ctr_off") = 0,
        py::arg("ap_step_base") = py::none(),
        py::arg("ap_mb_ctr") = py::none());
  m.def("lstm_cell_fwd", &lstm_cell_fwd_op, py::arg("gates_pre"),
        py::arg("gates_h"), py::arg("c_prev"), py::arg("c_new"),
        py::arg("h_new"), py::arg("done") = py::none(),
        py::arg("h_masked") = py::none(), py::arg("c_masked") = py::none(),
        py::arg("acts_out") = py::none());
  m.def("lstm_gemm_cell_fwd", &lstm_gemm_cell_fwd_op,
        "fused h@Wh^T recurrent GEMM + LSTM cell epilogue; returns False "
        "when the shape has no fused kernel (caller falls back)",
        py::arg("A"), py::arg("B"), py::arg("gates_pre"),
        py::arg("acts_out"), py::arg("c_prev"), py::arg("c_new"),
        py::arg("h_new"), py::arg("done") = py::none(),
        py::arg("h_masked") = py::none(), py::arg("c_masked") = py::none());
  m.def("lstm_bwd_fused", &lstm_bwd_fused_op,
        "fused LSTM cell backward + dgates @ Wh recurrent dgrad; returns "
        "False when the shape has no fused kernel (caller falls back)",
        py::arg("acts"), py::arg("c_prev"),
        py::arg("c_new"), py::arg("dh_head"), py::arg("dh_next"),
        py::arg("dc_next"), py::arg("done"), py::arg("B"), py::arg("dgates"),
        py::arg("dc_prev"), py::arg("dh_prev") = py::none());
  m.def("lstm_cell_bwd", &lstm_cell_bwd_op, py::arg("acts"),
        py::arg("c_prev"), py::arg("c_new"), py::arg("dh_head"),
        py::arg("dh_next"), py::arg("dc_next"), py::arg("done"),
        py::arg("dgates"), py::arg("dc_prev"));
  m.def("mask_reset", &mask_reset_op, py::arg("h"), py::arg("c"),
        py::arg("done"));
  m.def("masked_state", &masked_state_op, py::arg("h_raw"), py::arg("c_raw"),
        py::arg("done"), py::arg("h_in"), py::arg("c_in"));
  m.def("mb_gather_seq", &mb_gather_seq_op, py::arg("obs_src"),
        py::arg("act_src"), py::arg("logp_src"), py::arg("adv_src"),
        py::arg("ret_src"), py::arg("done_src"), py::arg("h0_src"),
        py::arg("c0_src"),
        py::arg("obs_mb"), py::arg("act_mb"), py::arg("logp_mb"),
        py::arg("adv_mb"), py::arg("ret_mb"), py::arg("done_mb"),
        py::arg("h0_mb"),
        py::arg("c0_mb"), py::arg("L"), py::arg("N"), py::arg("seed"),
        py::arg("minibatches"), py::arg("step_base"), py::arg("mb_ctr"));
  m.def("wgrad", &wgrad_op, py::arg("X"), py::arg("dY"), py::arg("dW_part"),
        py::arg("db_part"), py::arg("dW"), py::arg("db"), py::arg("slabs"),
        py::arg("ap_seed") = 0, py::arg("ap_minibatches") = 1,
        py::arg("ap_ctr_off") = 0, py::arg("ap_step_base") = py::none(),
        py::arg("ap_mb_ctr") = py::none());
  m.def("gae", &gae_op);
  m.def("adam", &adam_op, py::arg("p"), py::arg("g"), py::arg("m"),
        py::arg("v"), py::arg("p_bf16"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("step"),
        py::arg("gscale") = py::none(), py::arg("step_ctr") = py::none(),
        py::arg("clip_part") = py::none(), py::arg("clip_max_norm") = 0.0);
  m.def("grad_clip", &grad_clip_op);
  m.def("sample_head", &sample_head_op, py::arg("head"), py::arg("seed"),
        py::arg("step"), py::arg("actions"), py::arg("logp"),
        py::arg("value") = py::none(), py::arg("entropy") = py::none(),
        py::arg("greedy") = false, py::arg("step_base") = py::none(),
        py::arg("row_offset") = 0);
  m.def("increment_counter", &increment_counter_op, py::arg("ctr"),
        py::arg("delta"));
  m.def("mlp_policy_rollout", &mlp_policy_rollout_op, py::arg("obs"),
        py::arg("W1t"), py::arg("b1"), py::arg("W2t"), py::arg("b2"),
        py::arg("W3t"), py::arg("b3"), py::arg("actions"), py::arg("logp"),
        py::arg("value"), py::arg("seed"), py::arg("step"),
        py::arg("step_base") = py::none(), py::arg("row_offset") = 0,
        py::arg("greedy") = false);
  m.def("mb_gather", &mb_gather_op, py::arg("obs_src"), py::arg("act_src"),
        py::arg("logp_src"), py::arg("adv_src"), py::arg("ret_src"),
        py::arg("obs_mb"), py::arg("act_mb"), py::arg("logp_mb"),
        py::arg("adv_mb"), py::arg("ret_mb"), py::arg("seed"),
        py::arg("minibatches"), py::arg("step_base"), py::arg("mb_ctr"), py::arg("skip_obs") = false);
  m.def("ppo_loss_bwd", &ppo_loss_bwd_op, py::arg("head"), py::arg("actions"),
        py::arg("old_logp"), py::arg("adv"), py::arg("ret"), py::arg("dhead"),
        py::arg("clip_eps"), py::arg("ent_coef"), py::arg("vf_coef"),
        py::arg("inv_count"), py::arg("losses") = py::none());
  m.def("adv_normalize", &adv_normalize_op);
  m.def("f32_to_bf16", &f32_to_bf16_op);
  m.def("transpose_bf16", &transpose_bf16_op);
  py::class_<gymfx::GymFxEngine>(m, "GymFxEngine")
      .def(py::init<const py::dict&, const py::dict&, const py::dict&, const py::dict&>())
      .def("step", &gymfx::GymFxEngine::step, py::arg("actions"),
           py::arg("obs_out"), py::arg("reward_out") = py::none(),
           py::arg("terminated_out") = py::none(),
           py::arg("obs_bf16_out") = py::none(), py::arg("env_lo") = 0,
           py::arg("env_hi") = 0, py::arg("head") = py::none(),
           py::arg("logp_out") = py::none(), py::arg("value_out") = py::none(),
           py::arg("step_base") = py::none(), py::arg("sample_seed") = 0,
           py::arg("sample_step") = 0, py::arg("fuse_obs") = false,
           py::arg("h2") = py::none(), py::arg("w3t") = py::none(),
           py::arg("b3") = py::none(), py::arg("rnn_h") = py::none(),
           py::arg("rnn_c") = py::none())
      .def("build_obs", &gymfx::GymFxEngine::build_obs, py::arg("obs_out"),
           py::arg("obs_bf16_out") = py::none(), py::arg("env_lo") = 0,
           py::arg("env_hi") = 0);
  m.attr("EXEC_COUNTER_N") = (int)gymfx::EXEC_COUNTER_N;
  m.attr("ACT_COUNTER_N") = (int)gymfx::ACT_COUNTER_N;
}
