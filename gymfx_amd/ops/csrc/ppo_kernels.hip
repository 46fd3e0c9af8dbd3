// PPO compute stack for gfx950 (CDNA4): hand-written MFMA bf16 GEMMs with
// fused epilogues (bias+tanh fwd, tanh' dgrad), deterministic split-M wgrad,
// GAE backward scan, fused Adam, categorical sampling with counter-based RNG,
// and the PPO clipped-surrogate loss backward.
//
// The reference is agent-free by design (/root/reference/app/env.py:148-150);
// this stack is the BASELINE.json north-star requirement (PPO MLP(256,256)
// bf16 on MFMA, GAE scan, minibatch Adam).
//
// MFMA: v_mfma_f32_16x16x32_bf16 — per-wave 16x16 tile, K=32, fp32 accum.
// Lane mapping (cdna_hip_programming.md §3):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + i], i=0..7  (8 bf16)
//   B: lane l holds B[k = (l>>4)*8 + i][col = l&15]
//   C/D: lane l, reg r -> row = (l>>4)*4 + r, col = l&15
// Verified on hardware by tests/test_gpu_gemm.py (identity x asymmetric-B
// probes per guide rule G9).
#include "env_common.h"

#include <hip/hip_bf16.h>

#include <algorithm>

namespace gymfx {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

GFX_DEV float bf2f(__bf16 v) { return (float)v; }
GFX_DEV __bf16 f2bf(float v) { return (__bf16)v; }

// Inline tanh via the hardware exp (v_exp_f32).  libm tanhf is an actual
// CALL on gfx950: inside the GEMM epilogue it forced every live MFMA
// accumulator to scratch (272 B/thread of spill, 6x slowdown).  Accuracy is
// ~2 ulp of fp32 — far below the bf16 output rounding.
// fast_exp/fast_log/splitmix64 live in env_common.h (shared with the fused
// env_step sampling path so every sampler implementation agrees bitwise).
GFX_DEV float fast_tanh(float x) {
  x = fminf(fmaxf(x, -15.f), 15.f);
  // raw v_exp_f32 + v_rcp_f32, both single instructions.  An IEEE f32
  // divide here expands to a ~10-op guarded Newton sequence; unrolled 64x
  // in the wide epilogue that blew register pressure enough to demote the
  // MFMA accumulators to scratch for the whole kernel.
  const float e = __builtin_amdgcn_exp2f(x * 2.885390081777927f);
  return 1.f - 2.f * __builtin_amdgcn_rcpf(e + 1.f);
}

GFX_DEV uint32_t feistel_encrypt_once(uint32_t x, int half, uint32_t mask,
                                      uint64_t key) {
  uint32_t a = x & mask;        // low half
  uint32_t b = x >> half;       // high half
  for (int r = 0; r < 4; ++r) {
    const uint32_t f =
        (uint32_t)(splitmix64(key ^ ((uint64_t)r << 48) ^ (uint64_t)b)) & mask;
    const uint32_t t = b;
    b = a ^ f;
    a = t;
  }
  return (b << half) | a;
}

GFX_DEV uint32_t feistel_perm_idx(uint32_t i, uint32_t n, int half,
                                  uint64_t key) {
  const uint32_t mask = (1u << half) - 1u;
  uint32_t x = i;
  do {
    x = feistel_encrypt_once(x, half, mask, key);
  } while (x >= n);  // cycle-walk: terminates (bijection over 2^(2*half))
  return x;
}

// Row indirection for the gather+first-GEMM fusion: the L1 forward GEMM
// (and the W1 wgrad) read the rollout slab obs_flat THROUGH the epoch
// permutation instead of a materialized obs_mb copy (ROADMAP lever 3 —
// saves one full write+read pass over the minibatch observations).
// ctr_off: the gather that produced the OTHER minibatch fields has already
// advanced mb_ctr when these kernels run (-1 in the trainer's sequence).
// (FeistelMap struct lives in env_common.h — host code builds it.)
GFX_DEV uint64_t feistel_key(const FeistelMap& fm, uint32_t* mb) {
  const unsigned long long ctr =
      (unsigned long long)((long long)*fm.mb_ctr + fm.ctr_off);
  const uint32_t epoch = (uint32_t)(ctr / (unsigned)fm.minibatches);
  *mb = (uint32_t)(ctr % (unsigned)fm.minibatches);
  return splitmix64(fm.seed ^ (*fm.step_base * 0x9E3779B97F4A7C15ull) ^
                    ((uint64_t)epoch << 32));
}

GFX_DEV int64_t feistel_src_row(const FeistelMap& fm, uint64_t key,
                                uint32_t mb, int row) {
  return (int64_t)feistel_perm_idx(mb * (uint32_t)fm.M_mb + (uint32_t)row,
                                   fm.n, fm.half, key);
}

// ---------------------------------------------------------------------------
// GEMM: C[M,N] = act(A[M,K] @ B + bias)
//   TRANS_B = false: B stored [K, N] (ldb = N)
//   TRANS_B = true:  B stored [N, K] (ldb = K)  (used for dgrad: B = W^T)
// Block: 256 threads = 4 waves in a 2x2 wave grid; BK = 32; double-buffered
// LDS staging (global loads for tile k+1 issue in registers while MFMAs for
// tile k run — one __syncthreads per K-step).
// NFRAG = MFMA n-fragments per wave:
//   NFRAG=2 -> block tile 64x64  (the DEFAULT at every shape: occupancy 8
//              hides memory latency better than wider tiles save A-operand
//              re-streaming — L2 absorbs the re-reads; see launch_gemm)
//   NFRAG=4/8 -> 64x128 / 64x256 tiles, reachable via GYMFX_GEMM_WIDE
//              (measured slower: occupancy 2-3, 67% SQ_WAIT)
// Epilogues: ACT 0=none(f32 out) 1=none(bf16) 2=tanh(bf16)
//            DACT_TANH: multiply by (1 - Y^2) elementwise (dgrad fused tanh')
// ---------------------------------------------------------------------------
// min-waves floor 2: with the tanh epilogue the allocator otherwise chases
// a higher occupancy target and demotes the 64 MFMA accumulators of the
// NFRAG=8 variant to scratch — 1.2 GB of per-call scratch traffic, 4x
// slower than just running 2 waves/SIMD with accs in AGPRs.
// BKT: K-chunk depth of the LDS staging pipeline.  32 is the default; 64
// halves the barrier count for long-K tall shapes (the dgrad/fwd GEMMs of
// the update phase are latency-limited at 32 — ROADMAP lever 2), reachable
// via GYMFX_GEMM_BK64.  The MFMA k order is unchanged (two sequential
// 16x16x32 steps per 64-chunk), so results stay bitwise identical.
template <bool TRANS_B, int ACT, bool DACT_TANH, bool ADD_BIAS, int NFRAG,
          bool ACCUM = false, int BKT = 32, bool A_PERM = false>
__global__ __launch_bounds__(256, 2) void gemm_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C,
    const __bf16* __restrict__ Yact,  // activation output (DACT_TANH)
    int M, int N, int K, FeistelMap fm = FeistelMap{}) {
  constexpr int BM = 64, BK = BKT;
  constexpr int BN = 32 * NFRAG;
  // TRANS_B images are UNPADDED linear (required by the glds staging; the
  // b128 fragment-read bank multiplicity is already at its minimum for both
  // strides).  !TRANS_B keeps the +8 pad with register staging.
  constexpr int LDA = TRANS_B ? BK : (BK + 8);
  // B tile layout follows the GLOBAL layout so staging loads are always
  // contiguous bf16x8 (a strided 2-byte gather of B was 6x slower than the
  // whole GEMM): TRANS_B stages [n][k], !TRANS_B stages [k][n]; the
  // !TRANS_B fragment reads transpose out of LDS instead.
  constexpr int BS_ROWS = TRANS_B ? BN : BK;
  constexpr int BS_LD = TRANS_B ? BK : (BN + 8);
  __shared__ __align__(16) __bf16 As[2][BM][LDA];
  __shared__ __align__(16) __bf16 Bs[2][BS_ROWS][BS_LD];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;         // 0..3
  const int lane = tid & 63;
  const int wr = wave >> 1;          // wave row 0..1
  const int wc = wave & 1;           // wave col 0..1

  f32x4 acc[2][NFRAG] = {};

  // gather+first-GEMM fusion: A rows go through the epoch permutation
  uint64_t fkey = 0;
  uint32_t fmb = 0;
  if (A_PERM) fkey = feistel_key(fm, &fmb);
  auto a_row = [&](int gr) -> int64_t {
    return A_PERM ? feistel_src_row(fm, fkey, fmb, gr) : (int64_t)gr;
  };

  const int row_a = lane & 15;
  const int kseg = lane >> 4;  // 0..3 -> k-base = kseg*8

  // staging: element-offset addressed so any BKT works; for BKT=32 the
  // (row, col) decomposition reduces to the classic tid>>2 / (tid&3)*8
  constexpr int ALPT = (BM * BK) / 2048;  // bf16x8 loads per thread, A tile
  constexpr int BLPT = (BN * BK) / 2048;  // bf16x8 loads per thread, B tile
  // NFRAG=1 (BN=32) tiles need BKT=64 so every thread still stages >=1
  // bf16x8 of B (32*32/2048 would truncate to 0 loads)
  static_assert(ALPT >= 1 && BLPT >= 1, "tile too small for 256-thread staging");

  bf16x8 ra[ALPT], rb[BLPT];

  auto load_tile = [&](int k0) {
    // A tile [BM m][BK k]
    for (int j = 0; j < ALPT; ++j) {
      ra[j] = bf16x8{};
      const int e = (tid + j * 256) * 8;
      const int gr = bm + e / BK;
      if (gr < M) {
        const int64_t sr = a_row(gr);
        const int gk = k0 + e % BK;
        if (gk + 8 <= K) {
          ra[j] = *reinterpret_cast<const bf16x8*>(&A[sr * K + gk]);
        } else {
          for (int i = 0; i < 8; ++i)
            ra[j][i] = (gk + i < K) ? A[sr * K + gk + i] : (__bf16)0.f;
        }
      }
    }
    if (TRANS_B) {
      // stage [n][k]: contiguous along k in global [N, K]
      for (int j = 0; j < BLPT; ++j) {
        rb[j] = bf16x8{};
        const int e = (tid + j * 256) * 8;
        const int gn = bn + e / BK;
        if (gn < N) {
          const int gk = k0 + e % BK;
          if (gk + 8 <= K) {
            rb[j] = *reinterpret_cast<const bf16x8*>(&B[(int64_t)gn * K + gk]);
          } else {
            for (int i = 0; i < 8; ++i)
              rb[j][i] = (gk + i < K) ? B[(int64_t)gn * K + gk + i] : (__bf16)0.f;
          }
        }
      }
    } else {
      // stage [k][n]: contiguous along n in global [K, N]
      for (int j = 0; j < BLPT; ++j) {
        rb[j] = bf16x8{};
        const int c = tid + j * 256;             // chunk over [BK][BN/8]
        const int kk = c / (BN / 8);
        const int n8 = (c % (BN / 8)) * 8;
        const int gk = k0 + kk;
        const int gn = bn + n8;
        if (gk < K) {
          if (gn + 8 <= N) {
            rb[j] = *reinterpret_cast<const bf16x8*>(&B[(int64_t)gk * N + gn]);
          } else {
            for (int i = 0; i < 8; ++i)
              rb[j][i] = (gn + i < N) ? B[(int64_t)gk * N + gn + i] : (__bf16)0.f;
          }
        }
      }
    }
  };
  auto store_tile = [&](int buf) {
    for (int j = 0; j < ALPT; ++j) {
      const int e = (tid + j * 256) * 8;
      *reinterpret_cast<bf16x8*>(&As[buf][e / BK][e % BK]) = ra[j];
    }
    if (TRANS_B) {
      for (int j = 0; j < BLPT; ++j) {
        const int e = (tid + j * 256) * 8;
        *reinterpret_cast<bf16x8*>(&Bs[buf][e / BK][e % BK]) = rb[j];
      }
    } else {
      for (int j = 0; j < BLPT; ++j) {
        const int c = tid + j * 256;
        *reinterpret_cast<bf16x8*>(
            &Bs[buf][c / (BN / 8)][(c % (BN / 8)) * 8]) = rb[j];
      }
    }
  };

  const int ktiles = (K + BK - 1) / BK;
  // glds staging (TRANS_B, in-bounds blocks, full k-chunks): 16B per lane
  // into the linear images — no staging registers, loads still in flight
  // across the MFMA section and drained by the barrier's vmcnt(0).
  const bool can_glds = TRANS_B && (bm + BM <= M) && (bn + BN <= N);
  auto glds_tile = [&](int buf, int k0) {
    for (int j = 0; j < ALPT; ++j) {  // A tile [BM][BK]
      const int e0 = (wave * ALPT + j) * 512;
      const int row = (e0 + lane * 8) / BK;
      const int col = (e0 + lane * 8) % BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &A[a_row(bm + row) * K + k0 + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &As[buf][0][0] + e0),
          16, 0, 0);
    }
    for (int j = 0; j < BLPT; ++j) {  // B tile [BN][BK]
      const int e0 = (wave * BLPT + j) * 512;
      const int row = (e0 + lane * 8) / BK;
      const int col = (e0 + lane * 8) % BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &B[(int64_t)(bn + row) * K + k0 + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &Bs[buf][0][0] + e0),
          16, 0, 0);
    }
  };
  auto stage = [&](int buf, int k0) {
    if (can_glds && k0 + BK <= K) {
      glds_tile(buf, k0);
    } else {
      load_tile(k0);
      store_tile(buf);
    }
  };

  stage(0, 0);
  __syncthreads();

  for (int kt = 0; kt < ktiles; ++kt) {
    const int cur = kt & 1;
    const bool nxt_glds =
        kt + 1 < ktiles && can_glds && (kt + 1) * BK + BK <= K;
    if (nxt_glds) glds_tile(1 - cur, (kt + 1) * BK);
    else if (kt + 1 < ktiles) load_tile((kt + 1) * BK);

    for (int ks = 0; ks < BK / 32; ++ks) {  // sequential k order: bitwise
      bf16x8 af[2], bf[NFRAG];              // identical for any BKT
      for (int mi = 0; mi < 2; ++mi)
        af[mi] = *reinterpret_cast<const bf16x8*>(
            &As[cur][wr * 32 + mi * 16 + row_a][ks * 32 + kseg * 8]);
      if (TRANS_B) {
        for (int ni = 0; ni < NFRAG; ++ni)
          bf[ni] = *reinterpret_cast<const bf16x8*>(
              &Bs[cur][wc * (16 * NFRAG) + ni * 16 + row_a][ks * 32 + kseg * 8]);
      } else {
        // transpose at the LDS read: Bs holds [k][n]
        for (int ni = 0; ni < NFRAG; ++ni)
          for (int i = 0; i < 8; ++i)
            bf[ni][i] = Bs[cur][ks * 32 + kseg * 8 + i]
                          [wc * (16 * NFRAG) + ni * 16 + row_a];
      }
      for (int mi = 0; mi < 2; ++mi)
        for (int ni = 0; ni < NFRAG; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }

    if (kt + 1 < ktiles && !nxt_glds) store_tile(1 - cur);
    __syncthreads();
  }

  // ---- epilogue -------------------------------------------------------
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  for (int mi = 0; mi < 2; ++mi) {
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int gcol = bn + wc * (16 * NFRAG) + ni * 16 + ccol;
      if (gcol >= N) continue;
      for (int r = 0; r < 4; ++r) {
        const int grow = bm + wr * 32 + mi * 16 + crow_base + r;
        if (grow >= M) continue;
        float v = acc[mi][ni][r];
        if (ADD_BIAS) v += bias[gcol];
        if (ACT == 2) v = fast_tanh(v);
        if (DACT_TANH) {
          float y = bf2f(Yact[(int64_t)grow * N + gcol]);
          v *= (1.f - y * y);
        }
        if (ACT == 0) {
          float* cp = reinterpret_cast<float*>(C) + (int64_t)grow * N + gcol;
          *cp = ACCUM ? (*cp + v) : v;
        } else {
          reinterpret_cast<__bf16*>(C)[(int64_t)grow * N + gcol] = f2bf(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad partials: dW[K,N] = X^T @ dY, split over M into SLABS partial sums.
// Each block: one (ktile, ntile, slab); loops its M-chunk with MFMA where
// logical A = X^T [K, M] (i.e. A[k][m] = X[m][k]), logical B = dY [M, N].
// Deterministic: fixed slab count + ordered tree reduce (no atomics).
// Partials: f32 [S, K, N]; db partials f32 [S, N] (bias grad = colsum dY).
// Tiles are staged in NATURAL [m][k] / [m][n] layout with fully-coalesced
// bf16x8 global loads; the transpose happens at the MFMA fragment reads from
// LDS (scalar LDS reads are cheap; strided 2-byte GLOBAL reads were the
// bottleneck of the first version of this kernel).
// ---------------------------------------------------------------------------
// ---------------------------------------------------------------------------
// glds wgrad, 64x64 output tile, BKm=64 m-chunks: stages both operand
// chunks with __builtin_amdgcn_global_load_lds (16B per lane, LDS image
// linear in lane order — guide §5: 64²-tile step-3 structure), two LDS
// buffers, one __syncthreads per chunk (the barrier's vmcnt(0) drains the
// in-flight glds of the NEXT chunk — the simple 2-barrier recipe).
// Requires full, aligned tiles: bk+64<=K, bn+64<=N and a full 64-row
// m-chunk; the launcher falls back to the register-staged kernel for edge
// tiles (K%64 tails, the head layer's N=4, partial slabs).
// ---------------------------------------------------------------------------
template <bool WANT_DB, int FK, int FN, bool A_PERM = false>
__global__ __launch_bounds__(256) void wgrad_glds_kernel(
    const __bf16* __restrict__ X, const __bf16* __restrict__ dY,
    float* __restrict__ dW_part, float* __restrict__ db_part,
    int M, int N, int K, int slabs, FeistelMap fm = FeistelMap{}) {
  constexpr int BKm = 64;
  constexpr int TK = 32 * FK, TN = 32 * FN;
  // UNPADDED linear images (glds cannot scatter past a row pad)
  __shared__ __bf16 Xs[2][BKm][TK];
  __shared__ __bf16 Ys[2][BKm][TN];

  const int bk = blockIdx.x * TK;
  const int bn = blockIdx.y * TN;
  const int slab = blockIdx.z;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;

  const int m_per_slab = (M + slabs - 1) / slabs;
  const int m_begin = slab * m_per_slab;
  const int m_end = min(M, m_begin + m_per_slab);
  const int nchunks = (m_end - m_begin) / BKm;  // full chunks only (caller
                                                // guarantees divisibility)
  f32x4 acc[FK][FN] = {};
  float db_acc = 0.f;

  // gather+wgrad fusion: X rows go through the epoch permutation
  uint64_t fkey = 0;
  uint32_t fmb = 0;
  if (A_PERM) fkey = feistel_key(fm, &fmb);
  auto x_row = [&](int gm) -> int64_t {
    return A_PERM ? feistel_src_row(fm, fkey, fmb, gm) : (int64_t)gm;
  };

  // partial-K final tile (e.g. obs_dim=260 with TK=160: second tile covers
  // 100 live columns): X staged with guarded register loads instead of
  // glds — the K TAIL no longer needs a second full pass over dY
  // (profiles/lstm_kernel_stats_r2b.txt: the tail kernel re-read all of
  // dgates for 4 columns, ~1.5 ms/update).
  const bool k_tail = bk + TK > K;

  // one glds instruction stages 64 lanes x 16B = 1KB = 512 bf16; a 64xTW
  // chunk is 64*TW*2 B = TW/8 instructions = TW/32 per wave.
  auto glds_chunk = [&](int buf, int m0) {
    for (int j = 0; j < FK; ++j) {
      const int e0 = (wave * FK + j) * 512;
      const int row = (e0 + lane * 8) / TK;
      const int col = (e0 + lane * 8) % TK;
      if (k_tail) {
        bf16x8 v = bf16x8{};
        const int gk = bk + col;
        const int64_t xr = x_row(m0 + row) * K;
        if (gk + 8 <= K) {
          v = *reinterpret_cast<const bf16x8*>(&X[xr + gk]);
        } else {
          for (int i = 0; i < 8; ++i)
            v[i] = (gk + i < K) ? X[xr + gk + i] : (__bf16)0.f;
        }
        *reinterpret_cast<bf16x8*>(&Xs[buf][0][0] + e0 + lane * 8) = v;
        continue;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &X[x_row(m0 + row) * K + bk + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &Xs[buf][0][0] + e0),
          16, 0, 0);
    }
    for (int j = 0; j < FN; ++j) {
      const int e0 = (wave * FN + j) * 512;
      const int row = (e0 + lane * 8) / TN;
      const int col = (e0 + lane * 8) % TN;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &dY[(int64_t)(m0 + row) * N + bn + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &Ys[buf][0][0] + e0),
          16, 0, 0);
    }
  };

  const int row_a = lane & 15;
  const int kseg = lane >> 4;

  if (nchunks > 0) glds_chunk(0, m_begin);
  __syncthreads();  // vmcnt(0) inside the barrier drains the glds
  for (int ci = 0; ci < nchunks; ++ci) {
    const int cur = ci & 1;
    if (ci + 1 < nchunks) glds_chunk(1 - cur, m_begin + (ci + 1) * BKm);

    if (WANT_DB && tid < TN) {
      float s = 0.f;
      for (int i = 0; i < BKm; ++i) s += bf2f(Ys[cur][i][tid]);
      db_acc += s;
    }
    // two 32-row reduction batches per 64-row chunk
    for (int half = 0; half < 2; ++half) {
      const int mb = half * 32;
      bf16x8 af[FK], bf_[FN];
      for (int fi = 0; fi < FK; ++fi)
        for (int i = 0; i < 8; ++i)
          af[fi][i] =
              Xs[cur][mb + kseg * 8 + i][wr * (16 * FK) + fi * 16 + row_a];
      for (int ni = 0; ni < FN; ++ni)
        for (int i = 0; i < 8; ++i)
          bf_[ni][i] =
              Ys[cur][mb + kseg * 8 + i][wc * (16 * FN) + ni * 16 + row_a];
      for (int fi = 0; fi < FK; ++fi)
        for (int ni = 0; ni < FN; ++ni)
          acc[fi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi], bf_[ni], acc[fi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  float* out = dW_part + (int64_t)slab * K * N;
  for (int fi = 0; fi < FK; ++fi)
    for (int ni = 0; ni < FN; ++ni) {
      const int gcol = bn + wc * (16 * FN) + ni * 16 + ccol;
      for (int r = 0; r < 4; ++r) {
        const int grow = bk + wr * (16 * FK) + fi * 16 + crow_base + r;
        if (grow < K) out[(int64_t)grow * N + gcol] = acc[fi][ni][r];
      }
    }
  if (WANT_DB && tid < TN && blockIdx.x == 0)
    db_part[(int64_t)slab * N + bn + tid] = db_acc;
}

template <bool WANT_DB, int FK, int FN, bool A_PERM = false>
__global__ __launch_bounds__(256) void wgrad_partial_kernel(
    const __bf16* __restrict__ X, const __bf16* __restrict__ dY,
    float* __restrict__ dW_part, float* __restrict__ db_part,
    int M, int N, int K, int slabs, int kt0, FeistelMap fm = FeistelMap{}) {
  // FK/FN = MFMA fragments per wave along K/N: block tile (32*FK) x (32*FN)
  // over the dW output; reduction dim is M, chunked BKm=32 rows at a time
  // with double-buffered natural-layout LDS staging (coalesced bf16x8
  // global loads; the transpose happens at the fragment reads).
  constexpr int BKm = 32;
  constexpr int TK = 32 * FK, TN = 32 * FN;
  constexpr int LDX = TK + 8, LDY = TN + 8;
  __shared__ __bf16 Xs[2][BKm][LDX];   // X tile, natural layout: [m][k]
  __shared__ __bf16 Ys[2][BKm][LDY];   // dY tile, natural layout: [m][n]

  const int bk = (kt0 + blockIdx.x) * TK;
  const int bn = blockIdx.y * TN;
  const int slab = blockIdx.z;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;

  const int m_per_slab = (M + slabs - 1) / slabs;
  const int m_begin = slab * m_per_slab;
  const int m_end = min(M, m_begin + m_per_slab);

  f32x4 acc[FK][FN] = {};
  float db_acc = 0.f;

  uint64_t fkey = 0;
  uint32_t fmb = 0;
  if (A_PERM) fkey = feistel_key(fm, &fmb);

  bf16x8 rx[FK / 2], ry[FN / 2];
  auto load_chunk = [&](int m0) {
    // X chunk [32 m][TK k]: 32*TK elems = 256 threads * (FK/2) vec8
    for (int j = 0; j < FK / 2; ++j) {
      const int c = tid + j * 256;          // chunk id over [32][TK/8]
      const int m = c / (4 * FK);
      const int c8 = (c % (4 * FK)) * 8;
      rx[j] = bf16x8{};
      const int gm = m0 + m;
      if (gm < m_end) {
        const int64_t sm = A_PERM ? feistel_src_row(fm, fkey, fmb, gm)
                                  : (int64_t)gm;
        const int gk = bk + c8;
        if (gk + 8 <= K) {
          rx[j] = *reinterpret_cast<const bf16x8*>(&X[sm * K + gk]);
        } else {
          for (int i = 0; i < 8; ++i)
            rx[j][i] = (gk + i < K) ? X[sm * K + gk + i] : (__bf16)0.f;
        }
      }
    }
    for (int j = 0; j < FN / 2; ++j) {
      const int c = tid + j * 256;
      const int m = c / (4 * FN);
      const int c8 = (c % (4 * FN)) * 8;
      ry[j] = bf16x8{};
      const int gm = m0 + m;
      if (gm < m_end) {
        const int gn = bn + c8;
        if (gn + 8 <= N) {
          ry[j] = *reinterpret_cast<const bf16x8*>(&dY[(int64_t)gm * N + gn]);
        } else {
          for (int i = 0; i < 8; ++i)
            ry[j][i] = (gn + i < N) ? dY[(int64_t)gm * N + gn + i] : (__bf16)0.f;
        }
      }
    }
  };
  auto store_chunk = [&](int buf) {
    for (int j = 0; j < FK / 2; ++j) {
      const int c = tid + j * 256;
      *reinterpret_cast<bf16x8*>(&Xs[buf][c / (4 * FK)][(c % (4 * FK)) * 8]) = rx[j];
    }
    for (int j = 0; j < FN / 2; ++j) {
      const int c = tid + j * 256;
      *reinterpret_cast<bf16x8*>(&Ys[buf][c / (4 * FN)][(c % (4 * FN)) * 8]) = ry[j];
    }
  };

  const int row_a = lane & 15;
  const int kseg = lane >> 4;
  const int nchunks = (m_end - m_begin + BKm - 1) / BKm;

  if (nchunks > 0) {
    load_chunk(m_begin);
    store_chunk(0);
    __syncthreads();
  }
  for (int ci = 0; ci < nchunks; ++ci) {
    const int cur = ci & 1;
    const int m0 = m_begin + ci * BKm;
    if (ci + 1 < nchunks) load_chunk(m0 + BKm);

    if (WANT_DB && tid < TN) {
      float s = 0.f;
      for (int i = 0; i < BKm && m0 + i < m_end; ++i)
        s += bf2f(Ys[cur][i][tid]);
      db_acc += s;
    }
    // fragment reads do the transpose: logical A[kout][m] = X[m][kout],
    // logical B[m][n] = dY[m][n]; reduction index m = kseg*8 + i.
    bf16x8 af[FK], bf_[FN];
    for (int fi = 0; fi < FK; ++fi)
      for (int i = 0; i < 8; ++i)
        af[fi][i] = Xs[cur][kseg * 8 + i][wr * (16 * FK) + fi * 16 + row_a];
    for (int ni = 0; ni < FN; ++ni)
      for (int i = 0; i < 8; ++i)
        bf_[ni][i] = Ys[cur][kseg * 8 + i][wc * (16 * FN) + ni * 16 + row_a];
    for (int fi = 0; fi < FK; ++fi)
      for (int ni = 0; ni < FN; ++ni)
        acc[fi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[fi], bf_[ni], acc[fi][ni], 0, 0, 0);

    if (ci + 1 < nchunks) store_chunk(1 - cur);
    __syncthreads();
  }

  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  float* out = dW_part + (int64_t)slab * K * N;
  for (int fi = 0; fi < FK; ++fi)
    for (int ni = 0; ni < FN; ++ni) {
      const int gcol = bn + wc * (16 * FN) + ni * 16 + ccol;
      if (gcol >= N) continue;
      for (int r = 0; r < 4; ++r) {
        const int grow = bk + wr * (16 * FK) + fi * 16 + crow_base + r;
        if (grow >= K) continue;
        out[(int64_t)grow * N + gcol] = acc[fi][ni][r];
      }
    }
  if (WANT_DB && tid < TN && kt0 == 0 && blockIdx.x == 0) {
    const int gn = bn + tid;
    if (gn < N) db_part[(int64_t)slab * N + gn] = db_acc;
  }
}

__global__ void slab_reduce_kernel(const float* __restrict__ part,
                                   float* __restrict__ out, int64_t elems,
                                   int slabs) {
  // float4 per thread + 4-way unrolled slab loop (keeps >=16 loads in
  // flight; the scalar version was latency-bound on the slab chain).
  const int64_t nvec = elems >> 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    f32x4 s0 = {}, s1 = {}, s2 = {}, s3 = {};
    int k = 0;
    for (; k + 4 <= slabs; k += 4) {
      s0 += *reinterpret_cast<const f32x4*>(&part[(int64_t)k * elems + i * 4]);
      s1 += *reinterpret_cast<const f32x4*>(&part[(int64_t)(k + 1) * elems + i * 4]);
      s2 += *reinterpret_cast<const f32x4*>(&part[(int64_t)(k + 2) * elems + i * 4]);
      s3 += *reinterpret_cast<const f32x4*>(&part[(int64_t)(k + 3) * elems + i * 4]);
    }
    for (; k < slabs; ++k)
      s0 += *reinterpret_cast<const f32x4*>(&part[(int64_t)k * elems + i * 4]);
    // fixed association order: (s0 + s1) + (s2 + s3) — deterministic
    *reinterpret_cast<f32x4*>(&out[i * 4]) = (s0 + s1) + (s2 + s3);
  }
  // scalar tail
  for (int64_t i = (nvec << 2) + blockIdx.x * blockDim.x + threadIdx.x;
       i < elems; i += (int64_t)gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int k = 0; k < slabs; ++k) s += part[(int64_t)k * elems + i];
    out[i] = s;
  }
}

// tiny-elems variant (e.g. the head bias, elems=4, slabs up to 256): one
// block per element, threads parallel over slabs, deterministic tree reduce.
__global__ void slab_reduce_small_kernel(const float* __restrict__ part,
                                         float* __restrict__ out,
                                         int64_t elems, int slabs) {
  __shared__ float red[256];
  const int64_t e = blockIdx.x;
  if (e >= elems) return;
  float s = 0.f;
  for (int k = threadIdx.x; k < slabs; k += 256)
    s += part[(int64_t)k * elems + e];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[e] = red[0];
}

// ---------------------------------------------------------------------------
// bf16 transpose: dst[N,K] = src[K,N]^T.  Keeps per-weight transposed
// mirrors fresh after each Adam step so every model GEMM can stage its B
// operand with contiguous vector loads (TRANS_B path) — the strided
// alternative spilled registers and ran 6x slower.  LDS 64x64 tile.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const __bf16* __restrict__ src, __bf16* __restrict__ dst, int K, int N) {
  __shared__ __bf16 tile[64][64 + 4];
  const int kb = blockIdx.x * 64;
  const int nb = blockIdx.y * 64;
  // load 64x64 tile of src (rows k, cols n), coalesced along n
  for (int e = threadIdx.x; e < 64 * 8; e += 256) {
    const int r = e >> 3;            // 0..63 (k)
    const int c8 = (e & 7) * 8;      // 0..56 (n)
    const int gk = kb + r;
    const int gn = nb + c8;
    bf16x8 v = {};
    if (gk < K) {
      if (gn + 8 <= N) {
        v = *reinterpret_cast<const bf16x8*>(&src[(int64_t)gk * N + gn]);
      } else {
        for (int i = 0; i < 8; ++i)
          v[i] = (gn + i < N) ? src[(int64_t)gk * N + gn + i] : (__bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(&tile[r][c8]) = v;
  }
  __syncthreads();
  // write transposed, coalesced along k
  for (int e = threadIdx.x; e < 64 * 8; e += 256) {
    const int r = e >> 3;            // 0..63 (n)
    const int c8 = (e & 7) * 8;      // 0..56 (k)
    const int gn = nb + r;
    const int gk = kb + c8;
    if (gn >= N) continue;
    if (gk + 8 <= K) {
      bf16x8 v;
      for (int i = 0; i < 8; ++i) v[i] = tile[c8 + i][r];
      *reinterpret_cast<bf16x8*>(&dst[(int64_t)gn * K + gk]) = v;
    } else {
      for (int i = 0; i < 8 && gk + i < K; ++i)
        dst[(int64_t)gn * K + gk + i] = tile[c8 + i][r];
    }
  }
}

// ---------------------------------------------------------------------------
// elementwise casts
// ---------------------------------------------------------------------------
__global__ void f32_to_bf16_kernel(const float* __restrict__ in,
                                   __bf16* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = f2bf(in[i]);
}

// ---------------------------------------------------------------------------
// GAE backward scan (per env): adv[t] = delta_t + gamma*lam*(1-done_t)*adv[t+1]
// rewards/values/dones laid out [T, N] so lane reads coalesce across envs.
// ---------------------------------------------------------------------------
__global__ void gae_kernel(const float* __restrict__ rewards,
                           const float* __restrict__ values,   // [T+1, N]
                           const bool* __restrict__ dones,     // [T, N]
                           float* __restrict__ adv, float* __restrict__ ret,
                           int T, int N, float gamma, float lam) {
  const int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  float running = 0.f;
  for (int t = T - 1; t >= 0; --t) {
    const float nonterm = dones[(int64_t)t * N + n] ? 0.f : 1.f;
    const float delta = rewards[(int64_t)t * N + n] +
                        gamma * values[(int64_t)(t + 1) * N + n] * nonterm -
                        values[(int64_t)t * N + n];
    running = delta + gamma * lam * nonterm * running;
    adv[(int64_t)t * N + n] = running;
    ret[(int64_t)t * N + n] = running + values[(int64_t)t * N + n];
  }
}

// ---------------------------------------------------------------------------
// Fused Adam (f32 master params; bf16 mirror refreshed in the same pass)
// ---------------------------------------------------------------------------
// clip_part/nparts/max_norm: when given, every block tree-reduces the
// sumsq partials itself and derives the clip scale inline (the partials
// are ~1 KB, L2-resident) — this removed the separate one-block
// clip_scale launch from the 32x-per-update optimizer chain.
__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            __bf16* __restrict__ p_bf16, int64_t n, float lr,
                            float beta1, float beta2, float eps, float bc1,
                            float bc2, const float* __restrict__ gscale,
                            const int* __restrict__ step_ctr,
                            const float* __restrict__ clip_part, int nparts,
                            float max_norm) {
  if (step_ctr) {
    // device step counter (hipGraph-replayable): effective step = *ctr + 1;
    // the companion increment kernel advances it after this launch.
    const float st = (float)(*step_ctr + 1);
    bc1 = 1.f - powf(beta1, st);
    bc2 = 1.f - powf(beta2, st);
  }
  float s = 1.f;
  if (clip_part) {
    __shared__ float red[256];
    float acc = 0.f;
    for (int i = threadIdx.x; i < nparts; i += 256) acc += clip_part[i];
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
      __syncthreads();
    }
    const float norm = sqrtf(red[0]);
    s = (max_norm > 0.f && norm > max_norm) ? max_norm / norm : 1.f;
  } else if (gscale) {
    s = *gscale;
  }
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float gi = g[i] * s;
    const float mi = beta1 * m[i] + (1.f - beta1) * gi;
    const float vi = beta2 * v[i] + (1.f - beta2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    const float mhat = mi / bc1;
    const float vhat = vi / bc2;
    const float pi = p[i] - lr * mhat / (sqrtf(vhat) + eps);
    p[i] = pi;
    if (p_bf16) p_bf16[i] = f2bf(pi);
  }
}

// grad-norm^2: deterministic two-pass (fixed grid of partials, ordered sum)
__global__ void sumsq_partial_kernel(const float* __restrict__ g, int64_t n,
                                     float* __restrict__ part) {
  __shared__ float red[256];
  float s = 0.f;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    s += g[i] * g[i];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) part[blockIdx.x] = red[0];
}

__global__ void clip_scale_kernel(const float* __restrict__ part, int nparts,
                                  float max_norm, float* __restrict__ scale) {
  // one block of 256; tree reduce (deterministic, ~4x faster than the
  // single-thread serial loop this replaced)
  __shared__ float red[256];
  float s = 0.f;
  for (int i = threadIdx.x; i < nparts; i += 256) s += part[i];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const float norm = sqrtf(red[0]);
    *scale = (max_norm > 0.f && norm > max_norm) ? max_norm / norm : 1.f;
  }
}

// ---------------------------------------------------------------------------
// Categorical sampling + logp + entropy from the fused head output
// head[M, A+1] f32: cols 0..A-1 logits, col A value.
// Counter-based RNG (splitmix64): deterministic in (seed, step, env).
// ---------------------------------------------------------------------------
__global__ void sample_head_kernel(const float* __restrict__ head, int M,
                                   int n_actions, uint64_t seed, uint64_t step,
                                   int64_t* __restrict__ actions,
                                   float* __restrict__ logp,
                                   float* __restrict__ value,
                                   float* __restrict__ entropy,
                                   int greedy,
                                   const unsigned long long* __restrict__ step_base,
                                   int row_offset) {
  const int m = blockIdx.x * blockDim.x + threadIdx.x;
  if (m >= M) return;
  if (step_base) step += *step_base;  // device counter: hipGraph-replayable RNG
  const int64_t mg = m + row_offset;  // global row: split launches keep RNG
  const float* row = head + (int64_t)m * (n_actions + 1);
  const float logz = head_logz(row, n_actions);

  int a = 0;
  if (greedy) {
    float best = row[0];
    for (int j = 1; j < n_actions; ++j)
      if (row[j] > best) { best = row[j]; a = j; }
  } else {
    a = sample_categorical(row, n_actions, logz, seed, step, mg);
  }
  actions[m] = a;
  logp[m] = row[a] - logz;
  if (value) value[m] = row[n_actions];
  if (entropy) {
    float h = 0.f;
    for (int j = 0; j < n_actions; ++j) {
      const float lp = row[j] - logz;
      h -= fast_exp(lp) * lp;
    }
    entropy[m] = h;
  }
}

// ---------------------------------------------------------------------------
// PPO clipped-surrogate loss backward: head[M, A+1] f32 -> dhead[M, A+1] bf16
// losses (logging): [pi_loss, v_loss, entropy, approx_kl, clipfrac] via
// block-partial + atomicAdd (logging only; gradients are deterministic).
// ---------------------------------------------------------------------------
__global__ void ppo_loss_bwd_kernel(
    const float* __restrict__ head, const int64_t* __restrict__ actions,
    const float* __restrict__ old_logp, const float* __restrict__ adv,
    const float* __restrict__ ret, __bf16* __restrict__ dhead, int M,
    int n_actions, float clip_eps, float ent_coef, float vf_coef,
    float inv_count, float* __restrict__ losses) {
  const int m = blockIdx.x * blockDim.x + threadIdx.x;
  float l_pi = 0.f, l_v = 0.f, l_ent = 0.f, l_kl = 0.f, l_clip = 0.f;
  if (m < M) {
    const float* row = head + (int64_t)m * (n_actions + 1);
    __bf16* drow = dhead + (int64_t)m * (n_actions + 1);
    float mx = row[0];
    for (int j = 1; j < n_actions; ++j) mx = fmaxf(mx, row[j]);
    float z = 0.f;
    for (int j = 0; j < n_actions; ++j) z += expf(row[j] - mx);
    const float logz = logf(z) + mx;
    const int a = (int)actions[m];
    const float lp = row[a] - logz;
    const float ratio = expf(lp - old_logp[m]);
    const float A = adv[m];
    const bool clipped = (ratio > 1.f + clip_eps) || (ratio < 1.f - clip_eps);
    const float surr1 = ratio * A;
    const float rclip = fminf(fmaxf(ratio, 1.f - clip_eps), 1.f + clip_eps);
    const float surr2 = rclip * A;
    // d(-min(surr1,surr2))/dlogp = -A*ratio when surr1 <= surr2 (unclipped
    // branch active), else 0
    const float g_lp = (surr1 <= surr2) ? (-A * ratio) : 0.f;
    // entropy bonus: loss += -ent_coef * H
    float H = 0.f;
    for (int j = 0; j < n_actions; ++j) {
      const float lpj = row[j] - logz;
      H -= expf(lpj) * lpj;
    }
    for (int j = 0; j < n_actions; ++j) {
      const float pj = expf(row[j] - logz);
      const float onehot = (j == a) ? 1.f : 0.f;
      float d = g_lp * (onehot - pj);             // policy term
      d += ent_coef * pj * ((row[j] - logz) + H);  // -ent_coef*dH/dlogit
      drow[j] = f2bf(d * inv_count);
    }
    const float v = row[n_actions];
    const float dv = vf_coef * (v - ret[m]);
    drow[n_actions] = f2bf(dv * inv_count);

    l_pi = -fminf(surr1, surr2);
    l_v = 0.5f * (v - ret[m]) * (v - ret[m]);
    l_ent = H;
    l_kl = old_logp[m] - lp;
    l_clip = clipped ? 1.f : 0.f;
  }
  // block reduction for logging scalars
  __shared__ float red[256];
  float vals[5] = {l_pi, l_v, l_ent, l_kl, l_clip};
  for (int s = 0; s < 5; ++s) {
    red[threadIdx.x] = vals[s];
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
      __syncthreads();
    }
    if (threadIdx.x == 0 && losses) atomicAdd(&losses[s], red[0] * inv_count);
    __syncthreads();
  }
}

// normalize advantages in-place: (a - mean) / (std + 1e-8), deterministic
__global__ void adv_norm_stats_kernel(const float* __restrict__ adv, int64_t n,
                                      float* __restrict__ part /*[blocks*2]*/) {
  __shared__ float red[512];
  float s = 0.f, ss = 0.f;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    s += adv[i];
    ss += adv[i] * adv[i];
  }
  red[threadIdx.x] = s;
  red[256 + threadIdx.x] = ss;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) {
      red[threadIdx.x] += red[threadIdx.x + w];
      red[256 + threadIdx.x] += red[256 + threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    part[blockIdx.x] = red[0];
    part[gridDim.x + blockIdx.x] = red[256];
  }
}

__global__ void adv_norm_apply_kernel(float* __restrict__ adv, int64_t n,
                                      const float* __restrict__ part,
                                      int nparts) {
  __shared__ float mean_s, inv_std_s;
  if (threadIdx.x == 0) {
    float s = 0.f, ss = 0.f;
    for (int i = 0; i < nparts; ++i) { s += part[i]; ss += part[nparts + i]; }
    const float mean = s / (float)n;
    float var = ss / (float)n - mean * mean;
    var = fmaxf(var, 0.f);
    mean_s = mean;
    inv_std_s = 1.f / (sqrtf(var) + 1e-8f);
  }
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    adv[i] = (adv[i] - mean_s) * inv_std_s;
}

// ---------------------------------------------------------------------------
// LSTM cell, elementwise over [M, H] (the GEMM part — x@Wx + h@Wh + b — runs
// on the MFMA GEMM above; these kernels do the gate math).
// Gate layout in gates_pre [M, 4H]: [i | f | g | o].
//   c_new = sig(f)*c_prev + sig(i)*tanh(g);  h_new = sig(o)*tanh(c_new)
// fwd keeps gates_pre and c_prev/c_new around so bwd can recompute the
// activations (cheaper than storing five activation planes).
// ---------------------------------------------------------------------------
GFX_DEV float fast_sigmoid(float x) {
  x = fminf(fmaxf(x, -30.f), 30.f);
  return __builtin_amdgcn_rcpf(1.f + __builtin_amdgcn_exp2f(-x * 1.4426950408889634f));
}

// Gate layout is INTERLEAVED along the 4H axis (models/lstm.py): column
// 4*k + g holds gate g of hidden unit k (g: 0=i 1=f 2=g~ 3=o), so every
// per-unit gate access is ONE contiguous vector load (f32x4 / bf16x4)
// instead of four reads H apart.
//
// gates_h (optional, bf16): the recurrent projection h@Wh kept as a
// SEPARATE bf16 tensor and summed here — the fused alternative (f32
// accumulate-GEMM into gates_pre) read-modify-writes a 4x bigger buffer
// and was the dominant BPTT kernel.
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// 4-wide interleaved gate pre-activations of hidden unit k for row m
// (gates_pre is bf16 like gates_h; the sum runs in f32)
GFX_DEV f32x4 gate_pre4(const __bf16* __restrict__ gates_pre,
                        const __bf16* __restrict__ gates_h, int64_t idx4) {
  const bf16x4 gx = *reinterpret_cast<const bf16x4*>(&gates_pre[idx4]);
  f32x4 v;
  for (int j = 0; j < 4; ++j) v[j] = bf2f(gx[j]);
  if (gates_h) {
    const bf16x4 gh = *reinterpret_cast<const bf16x4*>(&gates_h[idx4]);
    for (int j = 0; j < 4; ++j) v[j] += bf2f(gh[j]);
  }
  return v;
}

// Cell forward math shared by the standalone cell kernel AND the fused
// GEMM epilogue: explicit fmaf pins the contraction so both compile to
// the same instruction sequence (a 1-ulp fma-vs-mul+add divergence
// between the two kernels broke bitwise fused==unfused once).
GFX_DEV void lstm_cell_math(const f32x4 gp, float cp, float* c_out,
                            float* h_out, bf16x4* acts_out) {
// contraction off: the compiler must not fuse mul+add differently in the
// different kernels that inline this helper (bitwise fused==unfused);
// the one fma we DO want is written explicitly below
#pragma clang fp contract(off)
  const float i = fast_sigmoid(gp[0]);
  const float f = fast_sigmoid(gp[1]);
  const float g = fast_tanh(gp[2]);
  const float o = fast_sigmoid(gp[3]);
  const float c = __builtin_fmaf(f, cp, i * g);
  *c_out = c;
  *h_out = o * fast_tanh(c);
  if (acts_out) {
    (*acts_out)[0] = f2bf(i);
    (*acts_out)[1] = f2bf(f);
    (*acts_out)[2] = f2bf(g);
    (*acts_out)[3] = f2bf(o);
  }
}


// Cell backward math shared by the standalone bwd kernel and the fused
// bwd kernel (same contraction-pinning rationale as lstm_cell_math).
// The four gate ACTIVATIONS arrive as saved bf16 (written by the forward
// cell) — the backward no longer touches gates_pre/gates_h at all, which
// removes 268 MB/minibatch of reads and all transcendental recompute.
// dh_next/dc_next arrive PRE-masked (0 when absent or across a reset).
GFX_DEV void lstm_cell_bwd_math(const bf16x4 acts, float cp, float c,
                                float dh_head_v, float dh_next_v,
                                float dc_next_v, bf16x4* dg, float* dcp) {
#pragma clang fp contract(off)
  const float i = bf2f(acts[0]);
  const float f = bf2f(acts[1]);
  const float g = bf2f(acts[2]);
  const float o = bf2f(acts[3]);
  const float tc = fast_tanh(c);
  const float dhv = dh_head_v + dh_next_v;
  float dc = dhv * o * (1.f - tc * tc);
  dc += dc_next_v;
  (*dg)[0] = f2bf(dc * g * i * (1.f - i));
  (*dg)[1] = f2bf(dc * cp * f * (1.f - f));
  (*dg)[2] = f2bf(dc * i * (1.f - g * g));
  (*dg)[3] = f2bf(dhv * tc * o * (1.f - o));
  *dcp = dc * f;
}

__global__ void lstm_cell_fwd_kernel(
    const __bf16* __restrict__ gates_pre,  // [M, 4H] (x-projection + bias)
    const __bf16* __restrict__ gates_h,   // [M, 4H] or null (h-projection)
    const float* __restrict__ c_prev,     // [M, H]
    float* __restrict__ c_new,            // [M, H]
    __bf16* __restrict__ h_new,           // [M, H] (bf16: feeds next GEMM)
    const bool* __restrict__ done,        // [M] or null: also emit the
    __bf16* __restrict__ h_masked,        //   reset-masked state that feeds
    float* __restrict__ c_masked,         //   the NEXT step (one fewer
    __bf16* __restrict__ acts_out,        // [M, 4H] saved activations or null
    int64_t M, int H) {                   //   launch than masked_state)
  const int64_t total = M * H;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t m = idx / H;
    const f32x4 gp = gate_pre4(gates_pre, gates_h, idx * 4);
    float c, hval;
    bf16x4 av;
    lstm_cell_math(gp, c_prev[idx], &c, &hval, acts_out ? &av : nullptr);
    if (acts_out) *reinterpret_cast<bf16x4*>(&acts_out[idx * 4]) = av;
    c_new[idx] = c;
    const __bf16 hb = f2bf(hval);
    h_new[idx] = hb;
    if (done) {
      const bool d = done[m];
      h_masked[idx] = d ? (__bf16)0.f : hb;
      c_masked[idx] = d ? 0.f : c;
    }
  }
}

// dh_head: gradient from the policy/value head (w.r.t. the RAW h output of
// this step).  dh_next: gradient arriving from step l+1's recurrent GEMM
// (w.r.t. the MASKED h input of step l+1) — masked by `done` (the episode
// boundary AFTER this step) along with dc_next, so no gradient crosses a
// reset.
__global__ void lstm_cell_bwd_kernel(
    const __bf16* __restrict__ acts,      // [M, 4H] saved activations
    const float* __restrict__ c_prev,     // [M, H] (masked input c)
    const float* __restrict__ c_new,      // [M, H] (raw output c)
    const __bf16* __restrict__ dh_head,   // [M, H] (head-dgrad slab, bf16)
    const float* __restrict__ dh_next,    // [M, H] or null
    const float* __restrict__ dc_next,    // [M, H] or null (last step)
    const bool* __restrict__ done,        // [M] or null
    __bf16* __restrict__ dgates,          // [M, 4H] out (bf16: feeds wgrad)
    float* __restrict__ dc_prev,          // [M, H] out
    int64_t M, int H) {
  // one thread owns FOUR consecutive hidden units: acts/dgates move as
  // bf16x8 pairs, the f32 streams as f32x4 (H % 4 == 0 enforced by the
  // binding) — the scalar-per-unit version ran at half its traffic floor
  const int64_t groups = M * (H / 4);
  for (int64_t gidx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       gidx < groups; gidx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t m = gidx / (H / 4);
    const int64_t idx0 = gidx * 4;  // == m * H + u0
    const float mask = (done && done[m]) ? 0.f : 1.f;
    const f32x4 cp = *reinterpret_cast<const f32x4*>(&c_prev[idx0]);
    const f32x4 cn = *reinterpret_cast<const f32x4*>(&c_new[idx0]);
    const bf16x4 dhb = *reinterpret_cast<const bf16x4*>(&dh_head[idx0]);
    f32x4 dhh;
    for (int j = 0; j < 4; ++j) dhh[j] = bf2f(dhb[j]);
    f32x4 dhn = {}, dcn = {};
    if (dh_next) dhn = *reinterpret_cast<const f32x4*>(&dh_next[idx0]);
    if (dc_next) dcn = *reinterpret_cast<const f32x4*>(&dc_next[idx0]);
    bf16x4 dg4[4];
    f32x4 dcp4;
    for (int j = 0; j < 4; ++j) {
      const bf16x4 av = *reinterpret_cast<const bf16x4*>(
          &acts[(idx0 + j) * 4]);
      float dcp;
      lstm_cell_bwd_math(av, cp[j], cn[j], dhh[j],
                         dh_next ? mask * dhn[j] : 0.f,
                         dc_next ? mask * dcn[j] : 0.f, &dg4[j], &dcp);
      dcp4[j] = dcp;
    }
    *reinterpret_cast<bf16x8*>(&dgates[idx0 * 4]) =
        *reinterpret_cast<const bf16x8*>(&dg4[0]);
    *reinterpret_cast<bf16x8*>(&dgates[idx0 * 4 + 8]) =
        *reinterpret_cast<const bf16x8*>(&dg4[2]);
    *reinterpret_cast<f32x4*>(&dc_prev[idx0]) = dcp4;
  }
}

// ---------------------------------------------------------------------------
// FUSED recurrent step: gates_h = h_in @ Wh^T computed by the standard MFMA
// pipeline (TRANS_B, NFRAG=2, 64x64 tile — same staging/double-buffering as
// gemm_kernel), with the LSTM CELL as the epilogue.  The interleaved gate
// layout makes each 64-column output tile hold 16 COMPLETE hidden units, so
// the cell needs no cross-tile exchange: accumulators go to an LDS tile,
// one barrier, then each thread does the gate math for 4 (row, unit) pairs.
// Removes the separate cell launch from the latency-bound sequential chain
// (profiles/lstm_kernel_stats_r2.txt: cell_fwd 11.3 us x 512/update).
//
// Bitwise parity with the unfused pair: the GEMM result is rounded to bf16
// BEFORE the gate sum (matching the bf16 gates_h tensor of the unfused
// path) and the MFMA K-chunk order is identical.
// ---------------------------------------------------------------------------
template <bool WRITE_GH>
__global__ __launch_bounds__(256, 2) void lstm_gemm_cell_fwd_kernel(
    const __bf16* __restrict__ A,         // h_in [M, K=H]
    const __bf16* __restrict__ B,         // Wh^T [N=4H, K=H]
    const __bf16* __restrict__ gates_pre,  // [M, 4H] (x-proj + bias)
    __bf16* __restrict__ acts_out,        // [M, 4H] saved activations (if WRITE_GH)
    const float* __restrict__ c_prev,     // [M, H]
    float* __restrict__ c_new,            // [M, H]
    __bf16* __restrict__ h_new,           // [M, H]
    const bool* __restrict__ done,        // [M] or null
    __bf16* __restrict__ h_masked,        // [M, H] (if done)
    float* __restrict__ c_masked,         // [M, H] (if done)
    int M, int N, int K) {
  constexpr int BM = 64, BK = 32, BN = 64;
  constexpr int LDA = BK;  // TRANS_B images are unpadded linear (glds)
  __shared__ __align__(16) __bf16 As[2][BM][LDA];
  __shared__ __align__(16) __bf16 Bs[2][BN][BK];
  __shared__ float Cs[BM][BN + 1];  // accum tile for the cell phase

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  f32x4 acc[2][2] = {};
  const int row_a = lane & 15;
  const int kseg = lane >> 4;
  const int a_r = tid >> 2;
  const int a_c8 = (tid & 3) * 8;

  bf16x8 ra, rb;
  auto load_tile = [&](int k0) {
    ra = bf16x8{};
    const int gr = bm + a_r;
    if (gr < M) {
      const int gk = k0 + a_c8;
      if (gk + 8 <= K) {
        ra = *reinterpret_cast<const bf16x8*>(&A[(int64_t)gr * K + gk]);
      } else {
        for (int i = 0; i < 8; ++i)
          ra[i] = (gk + i < K) ? A[(int64_t)gr * K + gk + i] : (__bf16)0.f;
      }
    }
    rb = bf16x8{};
    const int gn = bn + a_r;  // BN == BM: same staging coordinates
    if (gn < N) {
      const int gk = k0 + a_c8;
      if (gk + 8 <= K) {
        rb = *reinterpret_cast<const bf16x8*>(&B[(int64_t)gn * K + gk]);
      } else {
        for (int i = 0; i < 8; ++i)
          rb[i] = (gk + i < K) ? B[(int64_t)gn * K + gk + i] : (__bf16)0.f;
      }
    }
  };
  auto store_tile = [&](int buf) {
    *reinterpret_cast<bf16x8*>(&As[buf][a_r][a_c8]) = ra;
    *reinterpret_cast<bf16x8*>(&Bs[buf][a_r][a_c8]) = rb;
  };
  const bool can_glds = (bm + BM <= M) && (bn + BN <= N);
  auto glds_tile = [&](int buf, int k0) {
    {
      const int e0 = wave * 512;
      const int row = (e0 + lane * 8) / BK;
      const int col = (e0 + lane * 8) % BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &A[(int64_t)(bm + row) * K + k0 + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &As[buf][0][0] + e0),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(uintptr_t)(
              &B[(int64_t)(bn + row) * K + k0 + col]),
          (__attribute__((address_space(3))) void*)(uintptr_t)(
              &Bs[buf][0][0] + e0),
          16, 0, 0);
    }
  };
  auto stage = [&](int buf, int k0) {
    if (can_glds && k0 + BK <= K) {
      glds_tile(buf, k0);
    } else {
      load_tile(k0);
      store_tile(buf);
    }
  };

  const int ktiles = (K + BK - 1) / BK;
  stage(0, 0);
  __syncthreads();
  for (int kt = 0; kt < ktiles; ++kt) {
    const int cur = kt & 1;
    const bool nxt_glds = kt + 1 < ktiles && can_glds && (kt + 1) * BK + BK <= K;
    if (nxt_glds) glds_tile(1 - cur, (kt + 1) * BK);
    else if (kt + 1 < ktiles) load_tile((kt + 1) * BK);

    bf16x8 af[2], bf[2];
    for (int mi = 0; mi < 2; ++mi)
      af[mi] = *reinterpret_cast<const bf16x8*>(
          &As[cur][wr * 32 + mi * 16 + row_a][kseg * 8]);
    for (int ni = 0; ni < 2; ++ni)
      bf[ni] = *reinterpret_cast<const bf16x8*>(
          &Bs[cur][wc * 32 + ni * 16 + row_a][kseg * 8]);
    for (int mi = 0; mi < 2; ++mi)
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);

    if (kt + 1 < ktiles && !nxt_glds) store_tile(1 - cur);
    __syncthreads();
  }

  // accumulators -> LDS tile (bf16-round to match the unfused gates_h)
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < 2; ++ni)
      for (int r = 0; r < 4; ++r)
        Cs[wr * 32 + mi * 16 + crow_base + r][wc * 32 + ni * 16 + ccol] =
            bf2f(f2bf(acc[mi][ni][r]));
  __syncthreads();

  // ---- cell phase: one thread owns FOUR CONSECUTIVE units of one row
  // (4 threads/row x 64 rows), so every global access is a vector op:
  // gates/acts 32 B, c 16 B f32x4, h 8 B bf16x4.  (The first mapping —
  // 4 rows x 1 unit per thread — emitted 20 scattered scalar stores.)
  const int rl = tid >> 2;            // local row 0..63
  const int u0l = (tid & 3) * 4;      // first local unit (of 16)
  const int H = N / 4;
  const int grow = bm + rl;
  if (grow < M) {
    const int64_t idx0 = (int64_t)grow * H + (bn / 4 + u0l);
    const int64_t g0 = (int64_t)grow * N + (int64_t)(bn + u0l * 4);
    float c4[4], h4[4];
    bf16x4 av4[4];
    for (int j = 0; j < 4; ++j) {
      f32x4 gp;
      const bf16x4 gxb = *reinterpret_cast<const bf16x4*>(
          &gates_pre[g0 + j * 4]);
      for (int q = 0; q < 4; ++q)
        gp[q] = bf2f(gxb[q]) + Cs[rl][(u0l + j) * 4 + q];
      lstm_cell_math(gp, c_prev[idx0 + j], &c4[j], &h4[j],
                     WRITE_GH ? &av4[j] : nullptr);
    }
    if (WRITE_GH) {
      *reinterpret_cast<bf16x8*>(&acts_out[g0]) =
          *reinterpret_cast<const bf16x8*>(&av4[0]);
      *reinterpret_cast<bf16x8*>(&acts_out[g0 + 8]) =
          *reinterpret_cast<const bf16x8*>(&av4[2]);
    }
    f32x4 cv;
    bf16x4 hv;
    for (int j = 0; j < 4; ++j) {
      cv[j] = c4[j];
      hv[j] = f2bf(h4[j]);
    }
    *reinterpret_cast<f32x4*>(&c_new[idx0]) = cv;
    *reinterpret_cast<bf16x4*>(&h_new[idx0]) = hv;
    if (done) {
      const bool d = done[grow];
      f32x4 cm;
      bf16x4 hm;
      for (int j = 0; j < 4; ++j) {
        cm[j] = d ? 0.f : cv[j];
        hm[j] = d ? (__bf16)0.f : hv[j];
      }
      *reinterpret_cast<f32x4*>(&c_masked[idx0]) = cm;
      *reinterpret_cast<bf16x4*>(&h_masked[idx0]) = hm;
    }
  }
}

// ---------------------------------------------------------------------------
// FUSED BPTT backward step, tile-parallel: the cell backward (dgates) and
// the recurrent dgrad GEMM dh_prev = dgates @ Wh^T in ONE kernel with the
// standard 64x64 output tiling — each (bm, bn) block COMPUTES its A tile
// (dgates chunk) on the fly instead of reading a materialized tensor.
// The cell math is recomputed by each of the H/64 bn-blocks (the gate
// slabs are LLC-resident, the transcendentals are cheap), and only the
// bn==0 block writes dgates/dc_prev to global.  v1 of this fusion was a
// 16-row slab across the full 4H width: 4x the Wh traffic of this tiling
// and 1 workgroup/CU — measured SLOWER than the unfused pair
// (profiles/PERF_NOTES.md).  Bitwise parity: dgates are bf16-rounded
// before the MFMA, shared lstm_cell_bwd_math, gemm-identical K order.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void lstm_bwd_fused_kernel(
    const __bf16* __restrict__ acts,       // [M, 4H] saved activations
    const float* __restrict__ c_prev,      // [M, H]
    const float* __restrict__ c_new,       // [M, H]
    const __bf16* __restrict__ dh_head,    // [M, H] (bf16 head-dgrad slab)
    const float* __restrict__ dh_next,     // [M, H] or null
    const float* __restrict__ dc_next,     // [M, H] or null
    const bool* __restrict__ done,         // [M] or null
    const __bf16* __restrict__ B,          // Wh [N=H, K=4H] (trans_b layout)
    __bf16* __restrict__ dgates,           // [M, 4H] out
    float* __restrict__ dc_prev,           // [M, H] out
    float* __restrict__ dh_prev,           // [M, H] out, or null (step 0)
    int M, int H) {
  constexpr int BM = 64, BK = 32, BN = 64;
  const int K = 4 * H;
  const int N = H;
  __shared__ __align__(16) __bf16 As[2][BM][BK];
  __shared__ __align__(16) __bf16 Bs[2][BN][BK];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const bool write_dg = blockIdx.y == 0;

  // compute the dgates chunk for units [kt*8, kt*8+8) of rows [bm, bm+64)
  auto compute_a = [&](int buf, int kt) {
    const int u0 = kt * 8;
    for (int j = 0; j < 2; ++j) {
      const int p = tid + j * 256;       // (row, unit) pair
      const int rl = p >> 3;
      const int ul = p & 7;
      const int grow = bm + rl;
      bf16x4 dg = {};
      float dcp = 0.f;
      if (grow < M) {
        const int64_t idx = (int64_t)grow * H + (u0 + ul);
        const float mask = (done && done[grow]) ? 0.f : 1.f;
        const bf16x4 av = *reinterpret_cast<const bf16x4*>(&acts[idx * 4]);
        lstm_cell_bwd_math(av, c_prev[idx], c_new[idx], bf2f(dh_head[idx]),
                           dh_next ? mask * dh_next[idx] : 0.f,
                           dc_next ? mask * dc_next[idx] : 0.f, &dg, &dcp);
        if (write_dg) {
          *reinterpret_cast<bf16x4*>(&dgates[idx * 4]) = dg;
          dc_prev[idx] = dcp;
        }
      }
      *reinterpret_cast<bf16x4*>(&As[buf][rl][ul * 4]) = dg;
    }
  };
  if (dh_prev == nullptr) {
    // step 0: no recurrent-grad consumer — cell backward only (bn==0 grid)
    const int ktiles = K / BK;
    for (int kt = 0; kt < ktiles; ++kt) compute_a(0, kt);
    return;
  }
  auto glds_b = [&](int buf, int k0) {
    const int e0 = wave * 512;  // BN*BK = 2048 elems = 4 waves x 512
    const int row = (e0 + lane * 8) / BK;
    const int col = (e0 + lane * 8) % BK;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(uintptr_t)(
            &B[(int64_t)(bn + row) * K + k0 + col]),
        (__attribute__((address_space(3))) void*)(uintptr_t)(
            &Bs[buf][0][0] + e0),
        16, 0, 0);
  };

  f32x4 acc[2][2] = {};
  const int row_a = lane & 15;
  const int kseg = lane >> 4;
  const int ktiles = K / BK;
  glds_b(0, 0);
  compute_a(0, 0);
  __syncthreads();
  for (int kt = 0; kt < ktiles; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < ktiles) {
      glds_b(1 - cur, (kt + 1) * BK);
      compute_a(1 - cur, kt + 1);
    }
    bf16x8 af[2], bf[2];
    for (int mi = 0; mi < 2; ++mi)
      af[mi] = *reinterpret_cast<const bf16x8*>(
          &As[cur][wr * 32 + mi * 16 + row_a][kseg * 8]);
    for (int ni = 0; ni < 2; ++ni)
      bf[ni] = *reinterpret_cast<const bf16x8*>(
          &Bs[cur][wc * 32 + ni * 16 + row_a][kseg * 8]);
    for (int mi = 0; mi < 2; ++mi)
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
  }
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < 2; ++ni) {
      const int gcol = bn + wc * 32 + ni * 16 + ccol;
      for (int r = 0; r < 4; ++r) {
        const int grow = bm + wr * 32 + mi * 16 + crow + r;
        if (grow < M && gcol < N)
          dh_prev[(int64_t)grow * N + gcol] = acc[mi][ni][r];
      }
    }
}

// zero the recurrent state of terminated envs (rollout path: applied right
// after the fused env step, so the next policy step starts fresh)
__global__ void mask_reset_kernel(__bf16* __restrict__ h,
                                  float* __restrict__ c,
                                  const bool* __restrict__ done, int64_t M,
                                  int H) {
  const int64_t total = M * H;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    if (done[idx / H]) {
      h[idx] = (__bf16)0.f;
      c[idx] = 0.f;
    }
  }
}

// masked copy: out = in * (1 - done)  (BPTT fwd: the h/c INPUT of step l+1)
__global__ void masked_state_kernel(const __bf16* __restrict__ h_raw,
                                    const float* __restrict__ c_raw,
                                    const bool* __restrict__ done,
                                    __bf16* __restrict__ h_in,
                                    float* __restrict__ c_in, int64_t M,
                                    int H) {
  const int64_t total = M * H;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const bool d = done[idx / H];
    h_in[idx] = d ? (__bf16)0.f : h_raw[idx];
    c_in[idx] = d ? 0.f : c_raw[idx];
  }
}

// ---------------------------------------------------------------------------
// Fused MLP policy rollout step: obs -> tanh(obs@W1+b1) -> tanh(.@W2+b2)
// -> head(.@W3+b3) -> categorical sample, ONE kernel.
//
// The unfused rollout chain (3 GEMMs + sampler) is execution-floor bound
// (~8-10us per kernel at N=4096 regardless of bytes); fusing removes three
// kernel floors per env step (x128 steps per update).  The math replicates
// the GEMM kernel exactly (same BK=32 chunking, same MFMA sequence, bias
// added in f32, tanh via fast_tanh, h1/h2 rounded to bf16 between layers),
// so outputs are BITWISE identical to the unfused path — asserted by
// tests/test_gpu_trainer.py::test_fused_rollout_equals_unfused.
//
// Shapes: H = 256 (4 waves each own 32 rows x 128 cols), head_dim <= 16,
// D (obs_dim) arbitrary.  Block = 256 threads = 64 env rows; weights are
// the transposed mirrors (W1t [256, D], W2t [256, 256], W3t [A+1, 256]).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void mlp_policy_rollout_kernel(
    const __bf16* __restrict__ obs,   // [N, D]
    const __bf16* __restrict__ W1t,   // [256, D]
    const float* __restrict__ b1,     // [256]
    const __bf16* __restrict__ W2t,   // [256, 256]
    const float* __restrict__ b2,     // [256]
    const __bf16* __restrict__ W3t,   // [head_dim, 256]
    const float* __restrict__ b3,     // [head_dim]
    int64_t* __restrict__ actions, float* __restrict__ logp,
    float* __restrict__ value, int N, int D, int n_actions, uint64_t seed,
    uint64_t step, const unsigned long long* __restrict__ step_base,
    int row_offset, int greedy) {
  constexpr int BM = 64, BK = 32, H = 256;
  constexpr int LDT = BK + 8;     // staged-operand row (bf16)
  constexpr int LDH = H + 8;      // h1/h2 image row (bf16)
  __shared__ __bf16 As[2][BM][LDT];
  __shared__ __bf16 Bs[2][H][LDT];       // weight tile [n][k]
  __shared__ __bf16 Himg[2][BM][LDH];    // h1 / h2 images
  __shared__ float Head[BM][20];         // head values (row pad: 20 f32)
  const int bm = blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int row_a = lane & 15;
  const int kseg = lane >> 4;
  const int a_r = tid >> 2, a_c8 = (tid & 3) * 8;
  const int b_r = tid >> 2, b_c8 = (tid & 3) * 8;
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  f32x4 acc[2][8];
  bf16x8 ra, rb[4];

  // ONE shared dense-layer loop (layer selected at runtime): the first
  // version macro-duplicated the whole GEMM body per layer and the
  // allocator thrashed (~2.5k SGPR spills).  Math is bitwise identical to
  // the unfused GEMM kernel (same BK chunking, MFMA order, f32 bias,
  // fast_tanh, bf16 rounding).
  for (int layer = 0; layer < 2; ++layer) {
    const bool from_global = layer == 0;
    const __bf16* Wt = from_global ? W1t : W2t;
    const float* bias = from_global ? b1 : b2;
    const int K = from_global ? D : H;
    const __bf16(*Ximg)[LDH] = Himg[0];
    __bf16(*Yimg)[LDH] = Himg[layer];
    const int ktiles = (K + BK - 1) / BK;
    for (int mi = 0; mi < 2; ++mi)
      for (int ni = 0; ni < 8; ++ni) acc[mi][ni] = f32x4{};
    for (int kt = -1; kt + 1 < ktiles + 1; ++kt) {
      // stage chunk kt+1 (kt==-1 is the prologue), compute chunk kt
      const int knext = (kt + 1) * BK;
      if (kt + 1 < ktiles) {
        if (from_global) {
          ra = bf16x8{};
          const int gr = bm + a_r;
          if (gr < N) {
            const int gk = knext + a_c8;
            if (gk + 8 <= K) {
              ra = *reinterpret_cast<const bf16x8*>(&obs[(int64_t)gr * D + gk]);
            } else {
              for (int i = 0; i < 8; ++i)
                ra[i] = (gk + i < K) ? obs[(int64_t)gr * D + gk + i]
                                     : (__bf16)0.f;
            }
          }
        } else {
          ra = *reinterpret_cast<const bf16x8*>(&Ximg[a_r][knext + a_c8]);
        }
        for (int j = 0; j < 4; ++j) {
          const int gn = b_r + j * 64;
          const int gk = knext + b_c8;
          if (gk + 8 <= K) {
            rb[j] = *reinterpret_cast<const bf16x8*>(&Wt[(int64_t)gn * K + gk]);
          } else {
            rb[j] = bf16x8{};
            for (int i = 0; i < 8; ++i)
              rb[j][i] = (gk + i < K) ? Wt[(int64_t)gn * K + gk + i]
                                      : (__bf16)0.f;
          }
        }
      }
      if (kt >= 0) {
        const int cur = kt & 1;
        bf16x8 af[2], bf[8];
        for (int mi = 0; mi < 2; ++mi)
          af[mi] = *reinterpret_cast<const bf16x8*>(
              &As[cur][wr * 32 + mi * 16 + row_a][kseg * 8]);
        for (int ni = 0; ni < 8; ++ni)
          bf[ni] = *reinterpret_cast<const bf16x8*>(
              &Bs[cur][wc * 128 + ni * 16 + row_a][kseg * 8]);
        for (int mi = 0; mi < 2; ++mi)
          for (int ni = 0; ni < 8; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      }
      if (kt + 1 < ktiles) {
        const int nxt = (kt + 1) & 1;
        *reinterpret_cast<bf16x8*>(&As[nxt][a_r][a_c8]) = ra;
        for (int j = 0; j < 4; ++j)
          *reinterpret_cast<bf16x8*>(&Bs[nxt][b_r + j * 64][b_c8]) = rb[j];
      }
      __syncthreads();
    }
    for (int mi = 0; mi < 2; ++mi)
      for (int ni = 0; ni < 8; ++ni) {
        const int lcol = wc * 128 + ni * 16 + ccol;
        for (int r = 0; r < 4; ++r) {
          const int lrow = wr * 32 + mi * 16 + crow_base + r;
          float v = acc[mi][ni][r] + bias[lcol];
          Yimg[lrow][lcol] = f2bf(fast_tanh(v));
        }
      }
    __syncthreads();
  }

  // ---- head: [64, head_dim] = h2 @ W3 + b3 via one MFMA column tile ----
  {
    // wave w handles rows w*16..w*16+15 (16x16 MFMA, cols 0..15)
    f32x4 hacc = {};
    const int hrow = wave * 16 + row_a;
    for (int kt = 0; kt < H / BK; ++kt) {
      bf16x8 af, bf;
      af = *reinterpret_cast<const bf16x8*>(
          &Himg[1][hrow][kt * BK + kseg * 8]);
      for (int i = 0; i < 8; ++i) {
        const int k = kt * BK + kseg * 8 + i;
        bf[i] = (row_a <= n_actions) ? W3t[(int64_t)row_a * H + k] : (__bf16)0.f;
      }
      hacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, hacc, 0, 0, 0);
    }
    for (int r = 0; r < 4; ++r) {
      const int lrow = wave * 16 + crow_base + r;
      if (ccol <= n_actions)
        Head[lrow][ccol] = hacc[r] + b3[ccol];
    }
  }
  __syncthreads();

  // ---- categorical sample (== sample_head_kernel math) ----------------
  if (tid < BM) {
    const int gr = bm + tid;
    if (gr < N) {
      const float* row = Head[tid];
      float mx = row[0];
      for (int j = 1; j < n_actions; ++j) mx = fmaxf(mx, row[j]);
      float z = 0.f;
      for (int j = 0; j < n_actions; ++j) z += fast_exp(row[j] - mx);
      const float logz = fast_log(z) + mx;
      uint64_t st = step;
      if (step_base) st += *step_base;
      int a = 0;
      if (greedy) {
        float best = row[0];
        for (int j = 1; j < n_actions; ++j)
          if (row[j] > best) { best = row[j]; a = j; }
      } else {
        const int64_t mg = gr + row_offset;
        const uint64_t r =
            splitmix64(seed ^ (st * 0x51E1F5ull + (uint64_t)mg * 0x9E37ull));
        float u = (float)((r >> 11) * (1.0 / 9007199254740992.0));
        u = fminf(u, 0.999999f);
        float c = 0.f;
        a = n_actions - 1;
        for (int j = 0; j < n_actions; ++j) {
          c += fast_exp(row[j] - logz);
          if (u < c) { a = j; break; }
        }
      }
      actions[gr] = a;
      logp[gr] = row[a] - logz;
      if (value) value[gr] = row[n_actions];
    }
  }
}

// ---------------------------------------------------------------------------
// Device counters (hipGraph support): tiny kernels that advance the RNG /
// optimizer-step counters inside a captured graph, so one captured update
// replays with fresh randomness and correct Adam bias correction.
// ---------------------------------------------------------------------------
__global__ void increment_u64_kernel(unsigned long long* __restrict__ ctr,
                                     unsigned long long delta) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *ctr += delta;
}

__global__ void increment_i32_kernel(int* __restrict__ ctr, int delta) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *ctr += delta;
}

// ---------------------------------------------------------------------------
// Minibatch shuffle + gather, fused.
//
// The epoch permutation is a 4-round balanced Feistel network over the
// smallest even-bit power-of-two domain >= n, cycle-walked back into [0, n):
// a stateless bijection keyed by (seed, rollout counter, epoch).  This
// replaces host randperm + five giant index-gathers (and is exactly
// reproducible on the CPU oracle: ops/api.feistel_perm).
// ---------------------------------------------------------------------------
// (feistel helpers moved above gemm_kernel: the gather+first-GEMM
// fusion indexes A rows through the permutation)

// One wavefront copies one minibatch row. grid.x = ceil(M/4), block = 256.
// mb_ctr (device) = epoch * minibatches + mb, advanced by increment after
// the gather inside the captured graph; step_base keys the epoch permutation
// to the rollout counter so every update reshuffles.
__global__ __launch_bounds__(256) void mb_gather_kernel(
    const __bf16* __restrict__ obs_src, const int64_t* __restrict__ act_src,
    const float* __restrict__ logp_src, const float* __restrict__ adv_src,
    const float* __restrict__ ret_src, __bf16* __restrict__ obs_mb,
    int64_t* __restrict__ act_mb, float* __restrict__ logp_mb,
    float* __restrict__ adv_mb, float* __restrict__ ret_mb, int M, int D,
    uint32_t n, int half, uint64_t seed, int minibatches,
    const unsigned long long* __restrict__ step_base,
    const unsigned long long* __restrict__ mb_ctr, int skip_obs) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (row >= M) return;
  const unsigned long long ctr = *mb_ctr;
  const uint32_t epoch = (uint32_t)(ctr / (unsigned)minibatches);
  const uint32_t mb = (uint32_t)(ctr % (unsigned)minibatches);
  const uint64_t key = splitmix64(seed ^ (*step_base * 0x9E3779B97F4A7C15ull) ^
                                  ((uint64_t)epoch << 32));
  const uint32_t dst = mb * (uint32_t)M + (uint32_t)row;
  const uint32_t src = feistel_perm_idx(dst, n, half, key);

  const __bf16* src_row = obs_src + (int64_t)src * D;
  __bf16* dst_row = obs_mb + (int64_t)row * D;
  if (skip_obs) {
    // gather+first-GEMM fusion: the L1 GEMM / W1 wgrad read obs rows
    // through the permutation themselves — only the small fields copy
  } else if ((D & 3) == 0) {
    // 8-byte chunks (bf16 x4)
    const int chunks = D >> 2;
    const uint64_t* s64 = reinterpret_cast<const uint64_t*>(src_row);
    uint64_t* d64 = reinterpret_cast<uint64_t*>(dst_row);
    for (int c = lane; c < chunks; c += 64) d64[c] = s64[c];
  } else {
    for (int c = lane; c < D; c += 64) dst_row[c] = src_row[c];
  }
  if (lane == 0) {
    act_mb[row] = act_src[src];
    logp_mb[row] = logp_src[src];
    adv_mb[row] = adv_src[src];
    ret_mb[row] = ret_src[src];
  }
}

// ---------------------------------------------------------------------------
// Sequence minibatch gather for recurrent PPO (chunked BPTT): sequences are
// (chunk, env) pairs; sequence s of this minibatch maps through the Feistel
// permutation to a source sequence, and all L of its timesteps are gathered
// so the minibatch tensors are [L, Mseq, ...] (time-major for the BPTT
// loop).  h0/c0 are the saved chunk-boundary recurrent states.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void mb_gather_seq_kernel(
    const __bf16* __restrict__ obs_src,   // [T, N, D] flat
    const int64_t* __restrict__ act_src,  // [T, N]
    const float* __restrict__ logp_src, const float* __restrict__ adv_src,
    const float* __restrict__ ret_src,
    const bool* __restrict__ done_src,    // [T, N]
    const float* __restrict__ h0_src,     // [n_chunks, N, H] f32
    const float* __restrict__ c0_src,     // [n_chunks, N, H]
    __bf16* __restrict__ obs_mb,          // [L, Mseq, D]
    int64_t* __restrict__ act_mb,         // [L, Mseq]
    float* __restrict__ logp_mb, float* __restrict__ adv_mb,
    float* __restrict__ ret_mb,
    bool* __restrict__ done_mb,           // [L, Mseq]
    __bf16* __restrict__ h0_mb,           // [Mseq, H] bf16 (GEMM operand)
    float* __restrict__ c0_mb,            // [Mseq, H]
    int Mseq, int L, int D, int H, int N, uint32_t n_seq, int half,
    uint64_t seed, int minibatches,
    const unsigned long long* __restrict__ step_base,
    const unsigned long long* __restrict__ mb_ctr) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);  // seq within mb
  const int lane = threadIdx.x & 63;
  if (row >= Mseq) return;
  const unsigned long long ctr = *mb_ctr;
  const uint32_t epoch = (uint32_t)(ctr / (unsigned)minibatches);
  const uint32_t mb = (uint32_t)(ctr % (unsigned)minibatches);
  const uint64_t key = splitmix64(seed ^ (*step_base * 0x9E3779B97F4A7C15ull) ^
                                  ((uint64_t)epoch << 32));
  const uint32_t dst = mb * (uint32_t)Mseq + (uint32_t)row;
  const uint32_t src = feistel_perm_idx(dst, n_seq, half, key);
  const int chunk = (int)(src / (uint32_t)N);
  const int env = (int)(src % (uint32_t)N);
  // recurrent initial state
  {
    const float* hs = h0_src + ((int64_t)chunk * N + env) * H;
    const float* cs = c0_src + ((int64_t)chunk * N + env) * H;
    __bf16* hd = h0_mb + (int64_t)row * H;
    float* cd = c0_mb + (int64_t)row * H;
    for (int j = lane; j < H; j += 64) {
      hd[j] = f2bf(hs[j]);
      cd[j] = cs[j];
    }
  }
  for (int l = 0; l < L; ++l) {
    const int64_t t = (int64_t)chunk * L + l;
    const __bf16* srow = obs_src + (t * N + env) * D;
    __bf16* drow = obs_mb + ((int64_t)l * Mseq + row) * D;
    if ((D & 3) == 0) {
      const int chunks8 = D >> 2;
      const uint64_t* s64 = reinterpret_cast<const uint64_t*>(srow);
      uint64_t* d64 = reinterpret_cast<uint64_t*>(drow);
      for (int c = lane; c < chunks8; c += 64) d64[c] = s64[c];
    } else {
      for (int c = lane; c < D; c += 64) drow[c] = srow[c];
    }
    if (lane == 0) {
      const int64_t si = t * N + env;
      const int64_t di = (int64_t)l * Mseq + row;
      act_mb[di] = act_src[si];
      logp_mb[di] = logp_src[si];
      adv_mb[di] = adv_src[si];
      ret_mb[di] = ret_src[si];
      done_mb[di] = done_src[si];
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

void launch_lstm_cell_fwd(const void* gates_pre, const void* gates_h,
                          const float* c_prev, float* c_new, void* h_new,
                          const bool* done, void* h_masked, float* c_masked,
                          void* acts_out, int64_t M, int H,
                          hipStream_t stream) {
  int64_t total = M * H;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(lstm_cell_fwd_kernel, dim3(blocks), dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(gates_pre),
                     reinterpret_cast<const __bf16*>(gates_h),
                     c_prev, c_new, reinterpret_cast<__bf16*>(h_new), done,
                     reinterpret_cast<__bf16*>(h_masked), c_masked,
                     reinterpret_cast<__bf16*>(acts_out), M, H);
}

void launch_lstm_cell_bwd(const void* acts, const float* c_prev,
                          const float* c_new, const void* dh_head,
                          const float* dh_next, const float* dc_next,
                          const bool* done, void* dgates, float* dc_prev,
                          int64_t M, int H, hipStream_t stream) {
  int64_t total = M * (H / 4);
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(lstm_cell_bwd_kernel, dim3(blocks), dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(acts),
                     c_prev, c_new, reinterpret_cast<const __bf16*>(dh_head),
                     dh_next, dc_next, done,
                     reinterpret_cast<__bf16*>(dgates), dc_prev, M, H);
}

bool launch_lstm_gemm_cell_fwd(const void* A, const void* B,
                               const void* gates_pre, void* acts_out,
                               const float* c_prev, float* c_new, void* h_new,
                               const bool* done, void* h_masked,
                               float* c_masked, int M, int N, int K,
                               hipStream_t stream) {
  if (N % 64 != 0) return false;  // 64-col tiles must hold whole units
  dim3 grid(ceil_div(M, 64), N / 64);
  if (acts_out) {
    hipLaunchKernelGGL((lstm_gemm_cell_fwd_kernel<true>), grid, dim3(256), 0,
                       stream, reinterpret_cast<const __bf16*>(A),
                       reinterpret_cast<const __bf16*>(B),
                       reinterpret_cast<const __bf16*>(gates_pre),
                       reinterpret_cast<__bf16*>(acts_out), c_prev, c_new,
                       reinterpret_cast<__bf16*>(h_new), done,
                       reinterpret_cast<__bf16*>(h_masked), c_masked, M, N, K);
  } else {
    hipLaunchKernelGGL((lstm_gemm_cell_fwd_kernel<false>), grid, dim3(256), 0,
                       stream, reinterpret_cast<const __bf16*>(A),
                       reinterpret_cast<const __bf16*>(B),
                       reinterpret_cast<const __bf16*>(gates_pre), nullptr,
                       c_prev, c_new, reinterpret_cast<__bf16*>(h_new), done,
                       reinterpret_cast<__bf16*>(h_masked), c_masked, M, N, K);
  }
  return true;
}

bool launch_lstm_bwd_fused(const void* acts, const float* c_prev,
                           const float* c_new, const void* dh_head,
                           const float* dh_next, const float* dc_next,
                           const bool* done, const void* B, void* dgates,
                           float* dc_prev, float* dh_prev, int M, int H,
                           hipStream_t stream) {
  if (H % 64 != 0) return false;
  dim3 grid(ceil_div(M, 64), dh_prev ? H / 64 : 1);
  hipLaunchKernelGGL(lstm_bwd_fused_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(acts), c_prev, c_new,
                     reinterpret_cast<const __bf16*>(dh_head), dh_next,
                     dc_next, done,
                     reinterpret_cast<const __bf16*>(B),
                     reinterpret_cast<__bf16*>(dgates), dc_prev, dh_prev, M,
                     H);
  return true;
}

void launch_mask_reset(void* h, float* c, const bool* done, int64_t M, int H,
                       hipStream_t stream) {
  int64_t total = M * H;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(mask_reset_kernel, dim3(blocks), dim3(256), 0, stream,
                     reinterpret_cast<__bf16*>(h), c, done, M, H);
}

void launch_masked_state(const void* h_raw, const float* c_raw,
                         const bool* done, void* h_in, float* c_in, int64_t M,
                         int H, hipStream_t stream) {
  int64_t total = M * H;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(masked_state_kernel, dim3(blocks), dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(h_raw), c_raw, done,
                     reinterpret_cast<__bf16*>(h_in), c_in, M, H);
}

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
                          hipStream_t stream) {
  hipLaunchKernelGGL(mb_gather_seq_kernel, dim3(ceil_div(Mseq, 4)), dim3(256),
                     0, stream, reinterpret_cast<const __bf16*>(obs_src),
                     act_src, logp_src, adv_src, ret_src, done_src, h0_src,
                     c0_src, reinterpret_cast<__bf16*>(obs_mb), act_mb,
                     logp_mb, adv_mb, ret_mb, done_mb,
                     reinterpret_cast<__bf16*>(h0_mb), c0_mb,
                     Mseq, L, D, H, N, n_seq, half, seed, minibatches,
                     step_base, mb_ctr);
}

void launch_mlp_policy_rollout(const void* obs, const void* W1t,
                               const float* b1, const void* W2t,
                               const float* b2, const void* W3t,
                               const float* b3, int64_t* actions, float* logp,
                               float* value, int N, int D, int n_actions,
                               uint64_t seed, uint64_t step,
                               const unsigned long long* step_base,
                               int row_offset, int greedy,
                               hipStream_t stream) {
  hipLaunchKernelGGL(mlp_policy_rollout_kernel, dim3(ceil_div(N, 64)),
                     dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(obs),
                     reinterpret_cast<const __bf16*>(W1t), b1,
                     reinterpret_cast<const __bf16*>(W2t), b2,
                     reinterpret_cast<const __bf16*>(W3t), b3, actions, logp,
                     value, N, D, n_actions, seed, step, step_base, row_offset,
                     greedy);
}

void launch_increment_u64(unsigned long long* ctr, unsigned long long delta,
                          hipStream_t stream) {
  hipLaunchKernelGGL(increment_u64_kernel, dim3(1), dim3(1), 0, stream, ctr,
                     delta);
}

void launch_increment_i32(int* ctr, int delta, hipStream_t stream) {
  hipLaunchKernelGGL(increment_i32_kernel, dim3(1), dim3(1), 0, stream, ctr,
                     delta);
}

void launch_mb_gather(const void* obs_src, const int64_t* act_src,
                      const float* logp_src, const float* adv_src,
                      const float* ret_src, void* obs_mb, int64_t* act_mb,
                      float* logp_mb, float* adv_mb, float* ret_mb, int M,
                      int D, uint32_t n, int half, uint64_t seed,
                      int minibatches, const unsigned long long* step_base,
                      const unsigned long long* mb_ctr, hipStream_t stream,
                      int skip_obs) {
  hipLaunchKernelGGL(mb_gather_kernel, dim3(ceil_div(M, 4)), dim3(256), 0,
                     stream, reinterpret_cast<const __bf16*>(obs_src), act_src,
                     logp_src, adv_src, ret_src,
                     reinterpret_cast<__bf16*>(obs_mb), act_mb, logp_mb, adv_mb,
                     ret_mb, M, D, n, half, seed, minibatches, step_base,
                     mb_ctr, skip_obs);
}

void launch_gemm(const void* A, const void* B, const float* bias, void* C,
                 const void* Yact, int M, int N, int K, bool trans_b, int act,
                 bool dact_tanh, bool add_bias, bool accum,
                 hipStream_t stream, const FeistelMap* fmp = nullptr) {
  const __bf16* a = reinterpret_cast<const __bf16*>(A);
  const __bf16* b = reinterpret_cast<const __bf16*>(B);
  const __bf16* y = reinterpret_cast<const __bf16*>(Yact);
  // Tile width: 64x64 (NFRAG=2) tiles run at occupancy 8 and beat the
  // 64x256 (NFRAG=8, occupancy 3) wide tiles even at tall update shapes
  // (M=65536: 20.3 vs 22.0 ms/update whole-trainer) — the extra occupancy
  // hides HBM latency better than wide tiles save A-operand re-streaming
  // (L2 absorbs the re-reads).  GYMFX_GEMM_WIDE=1 forces the wide tile,
  // =4 the 64x128 middle tile (tuning knobs).
  static const int wide_env = [] {
    const char* e = getenv("GYMFX_GEMM_WIDE");
    return e ? atoi(e) : -1;
  }();
  // BK=64 staging pipeline (half the barriers) for long-K trans_b shapes —
  // ROADMAP lever 2; GYMFX_GEMM_BK64=1 forces on, =0 forces off
  static const int bk64_env = [] {
    const char* e = getenv("GYMFX_GEMM_BK64");
    return e ? atoi(e) : -1;
  }();
  // Tile-width heuristic: at N >= 1024 the 64-wide tile re-streams the A
  // operand N/64 times (the LSTM Wx gates GEMM read its 34 MB A slab 16x
  // = 544 MB/call); the 64x256 tile cuts that 4x and wins big there.
  // The round-1 "wide loses" measurement holds only for N <= 256 outputs.
  const bool mid = (N >= 96) &&
                   (wide_env == 4 ||
                    (wide_env == -1 && N >= 1024 && M >= 16384));
  const bool wide = (N >= 192) && !mid && wide_env == 1;
  // measured: BK=64 wins for long-K shapes (LSTM dgates dgrad K=1024:
  // update 41.5 -> 39.9 ms) and LOSES at K<=260 (MLP headline 28.4 ->
  // 27.4M) — auto-enable only at K >= 512
  const bool bk64 = trans_b && !wide && !mid &&
                    (bk64_env == 1 || (bk64_env != 0 && K >= 512));
  dim3 grid(ceil_div(M, 64), ceil_div(N, wide ? 256 : (mid ? 128 : 64)));
  dim3 block(256);
  // Narrow 64x32 tile (NFRAG=1, BK=64 only — see the BLPT static_assert):
  // at N=256 the 64x64 tile fills exactly 1 block/CU (the LSTM bwd-chain
  // dgrad: 64x4 = 256 workgroups) while occupancy allows more; BN=32
  // doubles the grid for latency hiding at 2x B re-streaming.
  // GYMFX_GEMM_N1=1 forces on, =0 off, unset = auto (M*N small, long K).
  static const int n1_env = [] {
    const char* e = getenv("GYMFX_GEMM_N1");
    return e ? atoi(e) : -1;
  }();
  if (bk64 && act == 0 && !dact_tanh && !add_bias && !accum && !fmp &&
      N <= 256 &&
      (n1_env == 1 || (n1_env == -1 && (int64_t)M * N <= 64 * 64 * 256))) {
    dim3 grid_n1(ceil_div(M, 64), ceil_div(N, 32));
    hipLaunchKernelGGL((gemm_kernel<true, 0, false, false, 1, false, 64>),
                       grid_n1, block, 0, stream, a, b, bias, C, y, M, N, K);
    return;
  }
  if (fmp) {
    // gather+first-GEMM fusion: only the L1 forward combo is instantiated
    // (trans_b, tanh epilogue, bias, 64x64 tile)
    if (!(trans_b && act == 2 && !dact_tanh && add_bias && !accum)) {
      // unreachable by construction (models/mlp.py passes a_feistel only
      // on the first layer); abort loudly if it ever is
      abort();
    }
    hipLaunchKernelGGL((gemm_kernel<true, 2, false, true, 2, false, 32, true>),
                       grid, block, 0, stream, a, b, bias, C, y, M, N, K,
                       *fmp);
    return;
  }

#define GEMM_LAUNCH(TB, ACT, DT, AB)                                          \
  do {                                                                        \
    if (wide)                                                                 \
      hipLaunchKernelGGL((gemm_kernel<TB, ACT, DT, AB, 8>), grid, block, 0,   \
                         stream, a, b, bias, C, y, M, N, K);                  \
    else if (mid)                                                             \
      hipLaunchKernelGGL((gemm_kernel<TB, ACT, DT, AB, 4>), grid, block, 0,   \
                         stream, a, b, bias, C, y, M, N, K);                  \
    else if (bk64)                                                            \
      hipLaunchKernelGGL((gemm_kernel<TB, ACT, DT, AB, 2, false, 64>), grid,  \
                         block, 0, stream, a, b, bias, C, y, M, N, K);        \
    else                                                                      \
      hipLaunchKernelGGL((gemm_kernel<TB, ACT, DT, AB, 2>), grid, block, 0,   \
                         stream, a, b, bias, C, y, M, N, K);                  \
  } while (0)
  if (accum) {
    // C += A@B (f32 out, no bias/activation): the LSTM recurrent GEMM
    if (trans_b)
      if (wide) hipLaunchKernelGGL((gemm_kernel<true, 0, false, false, 8, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
      else if (mid) hipLaunchKernelGGL((gemm_kernel<true, 0, false, false, 4, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
      else hipLaunchKernelGGL((gemm_kernel<true, 0, false, false, 2, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
    else
      if (wide) hipLaunchKernelGGL((gemm_kernel<false, 0, false, false, 8, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
      else if (mid) hipLaunchKernelGGL((gemm_kernel<false, 0, false, false, 4, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
      else hipLaunchKernelGGL((gemm_kernel<false, 0, false, false, 2, true>), grid, block, 0, stream, a, b, bias, C, y, M, N, K);
    return;
  }
  if (!trans_b && !dact_tanh) {
    if (act == 0) { if (add_bias) GEMM_LAUNCH(false, 0, false, true); else GEMM_LAUNCH(false, 0, false, false); }
    else if (act == 1) { if (add_bias) GEMM_LAUNCH(false, 1, false, true); else GEMM_LAUNCH(false, 1, false, false); }
    else { if (add_bias) GEMM_LAUNCH(false, 2, false, true); else GEMM_LAUNCH(false, 2, false, false); }
  } else if (trans_b && !dact_tanh) {
    if (act == 0) { if (add_bias) GEMM_LAUNCH(true, 0, false, true); else GEMM_LAUNCH(true, 0, false, false); }
    else if (act == 1) { if (add_bias) GEMM_LAUNCH(true, 1, false, true); else GEMM_LAUNCH(true, 1, false, false); }
    else { if (add_bias) GEMM_LAUNCH(true, 2, false, true); else GEMM_LAUNCH(true, 2, false, false); }
  } else if (trans_b && dact_tanh) {
    GEMM_LAUNCH(true, 1, true, false);
  } else {
    GEMM_LAUNCH(false, 1, true, false);
  }
#undef GEMM_LAUNCH
}

void launch_wgrad(const void* X, const void* dY, float* dW_part, float* db_part,
                  float* dW, float* db, int M, int N, int K, int slabs,
                  hipStream_t stream, const FeistelMap* fmp = nullptr) {
  const __bf16* x = reinterpret_cast<const __bf16*>(X);
  const __bf16* dy = reinterpret_cast<const __bf16*>(dY);
  // Tile choice: 128x128 halves the HBM re-streaming of X/dY — wgrad runs
  // at ~5.7 TB/s with 64x64 tiles, i.e. bandwidth-bound, so the bigger tile
  // wins whenever the grid still fills the chip (MLP 260x256 shapes:
  // 19.3 -> 18.5 ms/update whole-trainer).  384+ workgroups = 1.5 waves/SIMD
  // at the kernel's occupancy 2 keeps every CU fed.
  // GYMFX_WGRAD_BIG=0/1 overrides (tuning knob).
  static const int big_env = [] {
    const char* e = getenv("GYMFX_WGRAD_BIG");
    return e ? atoi(e) : -1;
  }();
  const bool big = K >= 128 && N >= 128 &&
                   (big_env >= 0
                        ? big_env != 0
                        : (int64_t)ceil_div(K, 128) * ceil_div(N, 128) * slabs >=
                              384);
  const int m_per_slab = (M + slabs - 1) / slabs;
  // glds path (both tile configs): full tiles only; the K tail runs the
  // skinny per-slab streaming kernel.
  const int TW = big ? 128 : 64;
  const bool glds_ok = N % TW == 0 && K >= TW && m_per_slab % 64 == 0 &&
                       M % slabs == 0;
  (void)0;
  if (glds_ok) {
    // the glds kernel masks a partial final K tile in-kernel (guarded
    // register staging + grow<K epilogue guard), so NO separate tail pass
    // over dY is needed.  For K=260 the 5x4 tile (TK=160) covers K in TWO
    // passes where 128-wide tiles need three — one fewer full dY read
    // (~1.5 ms/update on the LSTM wgrads).
#define WG_GLDS(WDB, FK, FN, GRID)                                            \
  do {                                                                        \
    if (fmp)                                                                  \
      hipLaunchKernelGGL((wgrad_glds_kernel<WDB, FK, FN, true>), GRID,        \
                         dim3(256), 0, stream, x, dy, dW_part, db_part, M, N, \
                         K, slabs, *fmp);                                     \
    else                                                                      \
      hipLaunchKernelGGL((wgrad_glds_kernel<WDB, FK, FN>), GRID, dim3(256),   \
                         0, stream, x, dy, dW_part, db_part, M, N, K, slabs); \
  } while (0)
    if (big) {
      // 5x4 (TK=160) saves a dY pass for K=260 but its 80-VGPR accs spill
      // at N >= 1024 (LSTM Wx wgrad measured 187 us vs 144 for 4x4+tail)
      const int t44 = ceil_div(K, 128), t54 = ceil_div(K, 160);
      if (t54 < t44 && N <= 512) {
        dim3 g0(t54, N / TW, slabs);
        if (db_part) WG_GLDS(true, 5, 4, g0);
        else WG_GLDS(false, 5, 4, g0);
      } else {
        dim3 g0(t44, N / TW, slabs);
        if (db_part) WG_GLDS(true, 4, 4, g0);
        else WG_GLDS(false, 4, 4, g0);
      }
    } else {
      dim3 g0(ceil_div(K, 64), N / TW, slabs);
      if (db_part) WG_GLDS(true, 2, 2, g0);
      else WG_GLDS(false, 2, 2, g0);
    }
#undef WG_GLDS
  } else {
    dim3 grid(ceil_div(K, big ? 128 : 64), ceil_div(N, big ? 128 : 64), slabs);
#define WG_PART(WDB, FK, FN)                                                  \
  do {                                                                        \
    if (fmp)                                                                  \
      hipLaunchKernelGGL((wgrad_partial_kernel<WDB, FK, FN, true>), grid,     \
                         dim3(256), 0, stream, x, dy, dW_part, db_part, M, N, \
                         K, slabs, 0, *fmp);                                  \
    else                                                                      \
      hipLaunchKernelGGL((wgrad_partial_kernel<WDB, FK, FN>), grid,           \
                         dim3(256), 0, stream, x, dy, dW_part, db_part, M, N, \
                         K, slabs, 0);                                        \
  } while (0)
    if (db_part) {
      if (big) WG_PART(true, 4, 4);
      else WG_PART(true, 2, 2);
    } else {
      if (big) WG_PART(false, 4, 4);
      else WG_PART(false, 2, 2);
    }
#undef WG_PART
  }
  int64_t elems = (int64_t)K * N;
  if (elems <= 4096) {
    hipLaunchKernelGGL(slab_reduce_small_kernel, dim3((int)elems), dim3(256),
                       0, stream, dW_part, dW, elems, slabs);
  } else {
    int blocks = (int)std::min<int64_t>((elems + 1023) / 1024, 1024);
    hipLaunchKernelGGL(slab_reduce_kernel, dim3(blocks), dim3(256), 0, stream,
                       dW_part, dW, elems, slabs);
  }
  if (db_part && db)
    hipLaunchKernelGGL(slab_reduce_small_kernel, dim3(N), dim3(256), 0, stream,
                       db_part, db, (int64_t)N, slabs);
}

void launch_gae(const float* rewards, const float* values, const bool* dones,
                float* adv, float* ret, int T, int N, float gamma, float lam,
                hipStream_t stream) {
  hipLaunchKernelGGL(gae_kernel, dim3(ceil_div(N, 256)), dim3(256), 0, stream,
                     rewards, values, dones, adv, ret, T, N, gamma, lam);
}

void launch_adam(float* p, const float* g, float* m, float* v, void* p_bf16,
                 int64_t n, float lr, float beta1, float beta2, float eps,
                 float bc1, float bc2, const float* gscale, const int* step_ctr,
                 hipStream_t stream) {
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, stream, p, g, m,
                     v, reinterpret_cast<__bf16*>(p_bf16), n, lr, beta1, beta2,
                     eps, bc1, bc2, gscale, step_ctr, nullptr, 0, 0.f);
}

void launch_adam_clip(float* p, const float* g, float* m, float* v,
                      void* p_bf16, int64_t n, float lr, float beta1,
                      float beta2, float eps, float bc1, float bc2,
                      const int* step_ctr, const float* clip_part, int nparts,
                      float max_norm, hipStream_t stream) {
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, stream, p, g, m,
                     v, reinterpret_cast<__bf16*>(p_bf16), n, lr, beta1, beta2,
                     eps, bc1, bc2, nullptr, step_ctr, clip_part, nparts,
                     max_norm);
}

void launch_grad_clip(const float* g, int64_t n, float max_norm, float* part,
                      float* scale, int nparts, hipStream_t stream) {
  hipLaunchKernelGGL(sumsq_partial_kernel, dim3(nparts), dim3(256), 0, stream,
                     g, n, part);
  hipLaunchKernelGGL(clip_scale_kernel, dim3(1), dim3(256), 0, stream, part,
                     nparts, max_norm, scale);
}

void launch_grad_sumsq(const float* g, int64_t n, float* part, int nparts,
                       hipStream_t stream) {
  hipLaunchKernelGGL(sumsq_partial_kernel, dim3(nparts), dim3(256), 0, stream,
                     g, n, part);
}

void launch_sample_head(const float* head, int M, int n_actions, uint64_t seed,
                        uint64_t step, int64_t* actions, float* logp,
                        float* value, float* entropy, int greedy,
                        const unsigned long long* step_base, int row_offset,
                        hipStream_t stream) {
  hipLaunchKernelGGL(sample_head_kernel, dim3(ceil_div(M, 256)), dim3(256), 0,
                     stream, head, M, n_actions, seed, step, actions, logp,
                     value, entropy, greedy, step_base, row_offset);
}

void launch_ppo_loss_bwd(const float* head, const int64_t* actions,
                         const float* old_logp, const float* adv,
                         const float* ret, void* dhead, int M, int n_actions,
                         float clip_eps, float ent_coef, float vf_coef,
                         float inv_count, float* losses, hipStream_t stream) {
  hipLaunchKernelGGL(ppo_loss_bwd_kernel, dim3(ceil_div(M, 256)), dim3(256), 0,
                     stream, head, actions, old_logp, adv, ret,
                     reinterpret_cast<__bf16*>(dhead), M, n_actions, clip_eps,
                     ent_coef, vf_coef, inv_count, losses);
}

void launch_adv_normalize(float* adv, int64_t n, float* part, int nparts,
                          hipStream_t stream) {
  hipLaunchKernelGGL(adv_norm_stats_kernel, dim3(nparts), dim3(256), 0, stream,
                     adv, n, part);
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(adv_norm_apply_kernel, dim3(blocks), dim3(256), 0, stream,
                     adv, n, part, nparts);
}

void launch_transpose_bf16(const void* src, void* dst, int K, int N,
                           hipStream_t stream) {
  dim3 grid(ceil_div(K, 64), ceil_div(N, 64));
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const __bf16*>(src),
                     reinterpret_cast<__bf16*>(dst), K, N);
}

void launch_f32_to_bf16(const float* in, void* out, int64_t n,
                        hipStream_t stream) {
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(blocks), dim3(256), 0, stream, in,
                     reinterpret_cast<__bf16*>(out), n);
}

}  // namespace gymfx
