// dsac_kernels.hip — CDNA4 (gfx950) kernels for the SAC hot path.
//
// Replaces the PyTorch op-sequences inventoried in SURVEY.md §2.6:
//   K1  linear_act_fwd[_g]   — GEMM + bias + ReLU (MFMA f32 16x16x4),
//                              grouped over grid.z (twin critics in one
//                              launch, shared activations)
//   K8  linear_bwd_dx/dwdb   — backward GEMMs with fused ReLU masking;
//                              dW/db is split-K over the batch (partials +
//                              reduce) so the grid fills 256 CUs
//   K4  squashed_gaussian_*  — fused clamp/exp/rsample/tanh/log-prob (+bwd)
//   K5  td_target            — Bellman backup elementwise
//   K6  critic/actor/alpha loss — single-workgroup fused loss reductions
//                              (incl. softmax(-alpha) task weights, per-task
//                              alpha gather, entropy) with analytic backward
//   K9  adam_step_[dev_]     — fused Adam over one flat parameter buffer
//   K10 polyak_              — fused soft target update over flat buffers
//   K12 replay_sample        — stratified gather from the HBM-resident
//                              sharded replay in ONE kernel
//
// Design notes (see /opt/skills guides):
// - fp32 end-to-end like the reference; GEMMs use the exact f32-input MFMA
//   v_mfma_f32_16x16x4_f32 (155 TF peak — these latency-bound tiny GEMMs
//   are grid/launch-bound, not FLOP-bound, so exact fp32 costs nothing).
// - tiles are 64x64x32 with LDS staging; row pads chosen so the MFMA
//   fragment gathers are LDS-bank-conflict-free (+2 on 32-wide rows:
//   bank = (34*r + k) % 32 = (2r + k) % 32 distinct for r<16, k<2;
//   +16 on 64-wide rows: (80*m + j) % 32 = (16m + j) % 32 distinct).
// - one workgroup = 4 waves (wave64), each wave owns a 32x32 output
//   sub-tile as a 2x2 grid of 16x16 MFMA fragments.
// - everything is hipGraph-capturable: no host-side state in the hot path
//   (Adam step counter lives on-device).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <cmath>
#include <vector>

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;

static constexpr int BM = 64;   // batch-tile rows
static constexpr int BN = 64;   // out-tile cols
static constexpr int BK = 32;   // reduction tile
static constexpr int PAD_K = BK + 2;   // 34: conflict-free [*][BK] frag reads
static constexpr int PAD_N = 80;       // 64+16: conflict-free [BK][*] frag reads

#define CHECK_IN(t) TORCH_CHECK((t).is_cuda() && (t).scalar_type() == torch::kFloat32, \
                                #t " must be a fp32 HIP tensor")

static inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// K1: y[g,M,N] = act(x[M,K] @ w[g,N,K]^T + b[g,N])   act: 0=none, 1=relu
// x is SHARED across groups (twin critics consume the same activations).
// grid: (ceil(M/64), ceil(N/64), G)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_act_fwd(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ b, float* __restrict__ y,
    int M, int N, int K, int act, long xgs) {
  // double-buffered: global loads for tile t+1 issue BEFORE the MFMAs of
  // tile t (HBM latency hides under compute — guide T14); ds_writes into
  // the other buffer, ONE barrier per K-tile.
  __shared__ float sx[2][BM][PAD_K];
  __shared__ float sw[2][BN][PAD_K];
  const long g = blockIdx.z;
  x += g * xgs;            // 0 = activations shared across groups
  w += g * (long)N * K;
  b += g * (long)N;
  y += g * (long)M * N;
  const int m0 = blockIdx.x * BM, n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;  // fragment row / k index
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  const int lr = (tid * 8) >> 5, lc = (tid * 8) & 31;  // this thread's slot
  // 2-tiles-ahead prefetch: two register sets; loads for tile t+2 issue
  // while tile t computes, so each load gets ~2 MFMA phases of latency
  // cover (the 1-deep version stalled: HBM latency > one MFMA phase).
  float rxa[8], rwa[8], rxb[8], rwb[8];
#define LOAD_TILE_FWD(k0, rx, rw)                                           \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    const int gk = (k0) + lc + j;                                           \
    rx[j] = (m0 + lr < M && gk < K) ? x[(long)(m0 + lr) * K + gk] : 0.f;    \
    rw[j] = (n0 + lr < N && gk < K) ? w[(long)(n0 + lr) * K + gk] : 0.f;    \
  }
#define STORE_TILE_FWD(buf, rx, rw)                                         \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    sx[buf][lr][lc + j] = rx[j];                                            \
    sw[buf][lr][lc + j] = rw[j];                                            \
  }
  LOAD_TILE_FWD(0, rxa, rwa)
  STORE_TILE_FWD(0, rxa, rwa)
  if (BK < K) LOAD_TILE_FWD(BK, rxa, rwa)
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    // issue loads for tile t+2 into the register set whose data for tile
    // t+1 has NOT yet been written to LDS?  No: set A holds t+1 (issued
    // last iter / prologue); set B receives t+2 now; at loop end we write
    // t+1 (set A) into the other LDS buffer and swap the sets.
    const bool have_next = k0 + BK < K;
    const bool have_next2 = k0 + 2 * BK < K;
    if (have_next2) LOAD_TILE_FWD(k0 + 2 * BK, rxb, rwb)
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sx[cur][wr + fi][kk + fk];
      const float a1 = sx[cur][wr + 16 + fi][kk + fk];
      const float b0 = sw[cur][wc + fi][kk + fk];
      const float b1 = sw[cur][wc + 16 + fi][kk + fk];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    if (have_next) {
      STORE_TILE_FWD(cur ^ 1, rxa, rwa)
#pragma unroll
      for (int j = 0; j < 8; ++j) {  // swap register sets (compiled away)
        const float tx = rxa[j], tw = rwa[j];
        rxa[j] = rxb[j]; rwa[j] = rwb[j];
        rxb[j] = tx; rwb[j] = tw;
      }
    }
    __syncthreads();
    cur ^= 1;
  }
  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg
  const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const f32x4 a = *accs[mi][ni];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr + mi * 16 + fk * 4 + r;
        const int col = n0 + wc + ni * 16 + fi;
        if (row < M && col < N) {
          float v = a[r] + b[col];
          if (act == 1) v = fmaxf(v, 0.f);
          y[(long)row * N + col] = v;
        }
      }
    }
}

// ---------------------------------------------------------------------------
// K8a: dx[M,K] = sum_g (dy*mask)[g,M,N] @ w[g,N,K]  (mask = yout>0 if act)
// The G-sum is the twin-critic case: both Qs consume the same x, so dL/dx
// accumulates over groups inside the K-loop (no extra kernel or atomics).
// grid: (ceil(M/64), ceil(K/64))
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_bwd_dx(
    const float* __restrict__ dy, const float* __restrict__ w,
    const float* __restrict__ yout, float* __restrict__ dx,
    int M, int N, int K, int act, int G, int S) {
  __shared__ float sdy[2][BM][PAD_K];   // [m][n-slice]
  __shared__ float sw[2][BK][PAD_N];     // [n-slice][k]
  // grid.z = Gz*S: Gz per-group slots (1 in summed mode), S slices over
  // the flattened (g, n0) reduction-tile loop (split-reduce fills the
  // chip for narrow-K layers; partials folded by k_reduce_partials).
  const long z = blockIdx.z;
  const long gz = z / S, sl = z % S;
  dy += gz * (long)M * N;
  w += gz * (long)N * K;
  yout += gz * (long)M * N;
  dx += z * (long)M * K;   // one output slab per (gz, slice)
  const int m0 = blockIdx.x * BM, c0 = blockIdx.y * BN;  // c over K
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  const int nt_per_g = (N + BK - 1) / BK;
  const int nt_all = G * nt_per_g;
  const int per_s = (nt_all + S - 1) / S;
  const int t_lo = (int)sl * per_s;
  const int nt = min(nt_all, t_lo + per_s);
  const int ar = (tid * 8) >> 5, ac = (tid * 8) & 31;  // dy slot (64x32)
  const int br = (tid * 8) >> 6, bc = (tid * 8) & 63;  // w slot (32x64)
  float rdy[8], rw[8];
#define LOAD_TILE_DX(ti)                                                    \
  {                                                                         \
    const int g_ = (ti) / nt_per_g;                                         \
    const int n0_ = ((ti) % nt_per_g) * BK;                                 \
    const float* dyg = dy + (long)g_ * M * N;                               \
    const float* wg = w + (long)g_ * N * K;                                 \
    const float* yg = yout + (long)g_ * M * N;                              \
    _Pragma("unroll") for (int j = 0; j < 8; ++j) {                         \
      const int gm = m0 + ar, gn = n0_ + ac + j;                            \
      float v = 0.f;                                                        \
      if (gm < M && gn < N) {                                               \
        v = dyg[(long)gm * N + gn];                                         \
        if (act == 1 && yg[(long)gm * N + gn] <= 0.f) v = 0.f;              \
      }                                                                     \
      rdy[j] = v;                                                           \
      const int wn = n0_ + br, wk = c0 + bc + j;                            \
      rw[j] = (wn < N && wk < K) ? wg[(long)wn * K + wk] : 0.f;             \
    }                                                                       \
  }
#define STORE_TILE_DX(buf)                                                  \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    sdy[buf][ar][ac + j] = rdy[j];                                          \
    sw[buf][br][bc + j] = rw[j];                                            \
  }
  if (t_lo >= nt) {  // empty slice: still must write zeros
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = m0 + wr + mi * 16 + fk * 4 + r;
          const int col = c0 + wc + ni * 16 + fi;
          if (row < M && col < K) dx[(long)row * K + col] = 0.f;
        }
    return;
  }
  LOAD_TILE_DX(t_lo)
  STORE_TILE_DX(0)
  __syncthreads();
  int cur = 0;
  for (int ti = t_lo; ti < nt; ++ti) {
    const bool more = ti + 1 < nt;
    if (more) LOAD_TILE_DX(ti + 1)
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sdy[cur][wr + fi][kk + fk];
      const float a1 = sdy[cur][wr + 16 + fi][kk + fk];
      const float b0 = sw[cur][kk + fk][wc + fi];
      const float b1 = sw[cur][kk + fk][wc + 16 + fi];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    if (more) STORE_TILE_DX(cur ^ 1)
    __syncthreads();
    cur ^= 1;
  }
  const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const f32x4 a = *accs[mi][ni];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr + mi * 16 + fk * 4 + r;
        const int col = c0 + wc + ni * 16 + fi;
        if (row < M && col < K) dx[(long)row * K + col] = a[r];
      }
    }
}

// ---------------------------------------------------------------------------
// K8b: dw[g,N,K] = (dy*mask)^T @ x per group, SPLIT-K over the batch:
// grid.z = G*S; slice s covers batch rows [s*chunk, (s+1)*chunk).  Partials
// land in ws[g*S+s][N][K] / ws_db[g*S+s][N]; k_reduce_partials folds S.
// This turns the 49-block (or 7-block, for head layers) serial-1280-batch
// kernel that dominated the baseline profile (104 us, 46% of GPU time —
// profiles/r01_baseline_NOTES.md) into a chip-filling grid.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_bwd_dwdb_splitk(
    const float* __restrict__ dy, const float* __restrict__ x,
    const float* __restrict__ yout, float* __restrict__ ws,
    float* __restrict__ ws_db, int M, int N, int K, int act, int S,
    int chunk, long xgs) {
  __shared__ float sa[2][BN][PAD_K];   // dy^T tile: [n][m-slice]
  __shared__ float sb[2][BK][PAD_N];    // x tile:    [m-slice][k]
  const int gs = blockIdx.z;        // g*S + s
  const int g = gs / S, s = gs % S;
  const float* dyg = dy + (long)g * M * N;
  const float* yg = yout + (long)g * M * N;
  const float* xg = x + (long)g * xgs;
  float* wsp = ws + (long)gs * N * K;
  float* dbp = ws_db + (long)gs * N;
  const int m_lo = s * chunk;
  const int m_hi = min(M, m_lo + chunk);
  const int n0 = blockIdx.x * BM, c0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  const bool do_db = (blockIdx.y == 0);
  float db_acc = 0.f;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  const int an = (tid * 8) & 63, am = (tid * 8) >> 6;  // dy slot (n fast)
  const int bm = (tid * 8) >> 6, bc = (tid * 8) & 63;  // x slot (k fast)
  float rdy[8], rx[8];
#define LOAD_TILE_DW(m0_)                                                   \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    const int gm = (m0_) + am, gn = n0 + an + j;                            \
    float v = 0.f;                                                          \
    if (gm < m_hi && gn < N) {                                              \
      v = dyg[(long)gm * N + gn];                                           \
      if (act == 1 && yg[(long)gm * N + gn] <= 0.f) v = 0.f;                \
    }                                                                       \
    rdy[j] = v;                                                             \
    const int xm = (m0_) + bm, xk = c0 + bc + j;                            \
    rx[j] = (xm < m_hi && xk < K) ? xg[(long)xm * K + xk] : 0.f;            \
  }
#define STORE_TILE_DW(buf)                                                  \
  _Pragma("unroll") for (int j = 0; j < 8; ++j) {                           \
    sa[buf][an + j][am] = rdy[j];                                           \
    sb[buf][bm][bc + j] = rx[j];                                            \
  }
  LOAD_TILE_DW(m_lo)
  STORE_TILE_DW(0)
  __syncthreads();
  int cur = 0;
  for (int m0 = m_lo; m0 < m_hi; m0 += BK) {
    const bool more = m0 + BK < m_hi;
    if (more) LOAD_TILE_DW(m0 + BK)
    if (do_db && tid < BN) {
#pragma unroll
      for (int m = 0; m < BK; ++m) db_acc += sa[cur][tid][m];
    }
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sa[cur][wr + fi][kk + fk];
      const float a1 = sa[cur][wr + 16 + fi][kk + fk];
      const float b0 = sb[cur][kk + fk][wc + fi];
      const float b1 = sb[cur][kk + fk][wc + 16 + fi];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    if (more) STORE_TILE_DW(cur ^ 1)
    __syncthreads();
    cur ^= 1;
  }
  if (do_db && tid < BN && n0 + tid < N) dbp[n0 + tid] = db_acc;
  const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const f32x4 a = *accs[mi][ni];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = n0 + wr + mi * 16 + fk * 4 + r;
        const int col = c0 + wc + ni * 16 + fi;
        if (row < N && col < K) wsp[(long)row * K + col] = a[r];
      }
    }
}

// fold S split-K partials: out[g][i] = sum_s ws[(g*S+s)*stride + i]
__global__ __launch_bounds__(256) void k_reduce_partials(
    const float* __restrict__ ws, float* __restrict__ out, long stride,
    int S, long n_per_g) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int g = blockIdx.y;
  if (i >= n_per_g) return;
  const float* base = ws + ((long)g * S) * stride + i;
  float acc = 0.f;
  for (int s = 0; s < S; ++s) acc += base[(long)s * stride];
  out[(long)g * n_per_g + i] = acc;
}

// ---------------------------------------------------------------------------
// K12: stratified replay sample — ONE kernel replaces the per-shard
// index_select/cat chains (~65 torch kernels per update in the baseline
// profile).  Fields are stacked [T, cap, D]; output row i draws from shard
// t = i / (B/T) at index floor(rand[i] * size[t]).
// ---------------------------------------------------------------------------
// counter-based device RNG (splitmix64 hash of (counter, index, salt)):
// every RNG-consuming kernel derives its noise from one persistent int64
// counter that k_adam_prolog_many (or prolog3) bumps ONCE per update
// (single-block kernels, so the bump is race-free by stream ordering).
// Replaces the per-update
// torch rand/randn launches AND the hipGraph RNG-offset bookkeeping
// kernels torch inserts around them.
__device__ __forceinline__ unsigned long long dsac_sm64(
    unsigned long long x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
__device__ __forceinline__ float dsac_u01(unsigned long long h) {
  // top 24 bits + 1 -> uniform in (0, 1]
  return (float)((h >> 40) + 1ull) * 5.9604644775390625e-8f;
}

__global__ __launch_bounds__(256) void k_replay_sample(
    const float* __restrict__ states, const float* __restrict__ actions,
    const float* __restrict__ rewards, const float* __restrict__ next_states,
    const float* __restrict__ dones, const float* __restrict__ sizes,
    const float* __restrict__ rnd, const long long* __restrict__ rng,
    float* __restrict__ o_states,
    float* __restrict__ o_actions, float* __restrict__ o_rewards,
    float* __restrict__ o_next_states, float* __restrict__ o_dones,
    int B, int T, int per, long cap, int Ds, int Da) {
  // one wave per batch row: lanes stride the row's columns (coalesced on
  // both the gathered source row and the packed destination row)
  const int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (i >= B) return;
  const int t = min(i / per, T - 1);
  float u;
  if (rng != nullptr) {
    const unsigned long long c = (unsigned long long)rng[0];
    u = dsac_u01(dsac_sm64(c * 0x100000001ull
                           + (unsigned long long)i * 2ull + 0x2ull));
    u = u < 1.f ? u : 0.99999994f;  // keep floor(u*size) < size
  } else {
    u = rnd[i];
  }
  const long idx = (long)(u * sizes[t]);
  const long src = (long)t * cap + idx;
  for (int j = lane; j < Ds; j += 64) {
    o_states[(long)i * Ds + j] = states[src * Ds + j];
    o_next_states[(long)i * Ds + j] = next_states[src * Ds + j];
  }
  for (int j = lane; j < Da; j += 64)
    o_actions[(long)i * Da + j] = actions[src * Da + j];
  if (lane == 0) {
    o_rewards[i] = rewards[src];
    o_dones[i] = dones[src];
  }
}

// ---------------------------------------------------------------------------
// K4: fused tanh-squashed Gaussian sample + log-prob (fwd + bwd).
// One thread per batch row; A = action_dim <= 32.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_squash_fwd(
    const float* __restrict__ mu, const float* __restrict__ lsr,
    float* __restrict__ eps, float* __restrict__ act,
    float* __restrict__ logp, float* __restrict__ tanh_u,
    float* __restrict__ ls_out, const long long* __restrict__ rng,
    long mu_ld, long ls_ld, int B, int A, float k) {
  // mu/lsr may be strided row views (the (mu|lsr) head output sliced in
  // half) — reading through the stride kills the .contiguous() copies
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  constexpr float C = 0.9189385332046727f;  // 0.5*log(2*pi)
  float lp = 0.f;
  for (int a = 0; a < A; ++a) {
    const long idx = (long)i * A + a;
    const float ls = fminf(fmaxf(lsr[(long)i * ls_ld + a], -20.f), 2.f);
    const float s = __expf(ls);
    float e;
    if (rng != nullptr) {
      // Box-Muller from two hashed uniforms; eps SAVED for the backward
      const unsigned long long c = (unsigned long long)rng[0];
      const unsigned long long h1 = dsac_sm64(
          c * 0x100000001ull + (unsigned long long)idx * 2ull + 0x1ull);
      const unsigned long long h2 = dsac_sm64(h1);
      e = sqrtf(-2.f * __logf(dsac_u01(h1)))
          * __cosf(6.283185307179586f * dsac_u01(h2));
      eps[idx] = e;
    } else {
      e = eps[idx];
    }
    const float u = mu[(long)i * mu_ld + a] + s * e;
    const float t = tanhf(u);
    act[idx] = k * t;
    tanh_u[idx] = t;
    ls_out[idx] = ls;
    lp += -0.5f * e * e - ls - C - __logf(k * (1.f - t * t + 1e-6f));
  }
  logp[i] = lp;
}

__global__ __launch_bounds__(256) void k_squash_bwd(
    const float* __restrict__ ga, const float* __restrict__ gl,
    const float* __restrict__ lsr, const float* __restrict__ ls,
    const float* __restrict__ eps, const float* __restrict__ tanh_u,
    float* __restrict__ dmu, float* __restrict__ dlsr,
    int B, int A, float k) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const float g = gl[i];
  for (int a = 0; a < A; ++a) {
    const long idx = (long)i * A + a;
    const float t = tanh_u[idx];
    const float omt2 = 1.f - t * t;
    const float dlp_du = 2.f * t * omt2 / (omt2 + 1e-6f);
    const float s = __expf(ls[idx]);
    const float e = eps[idx];
    const float du = ga[idx] * k * omt2 + g * dlp_du;   // dL/du
    dmu[idx] = du;
    const float raw = lsr[idx];
    const float mask = (raw >= -20.f && raw <= 2.f) ? 1.f : 0.f;
    dlsr[idx] = mask * (du * e * s - g);
  }
}

// ---------------------------------------------------------------------------
// K5: y = rs*r + gamma*(1-d)*(min(q1,q2) - alpha*lp).
// alpha comes per-sample from log_alpha[t_i] (one_hots suffix of mtobs) —
// no separate gather matmul (reference learner.get_log_alpha).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_td_target(
    const float* __restrict__ r, const float* __restrict__ d,
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ lp, const float* __restrict__ alpha,
    float* __restrict__ y, int n, float gamma, float rs) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  y[i] = rs * r[i] + gamma * (1.f - d[i]) * (fminf(q1[i], q2[i]) - alpha[i] * lp[i]);
}

// K5b: MT variant — per-sample alpha computed in-kernel from
// log_alpha[t_i] (one-hot suffix), removing the gather matmul
// (reference learner.get_log_alpha, MT10…MTSAC/src/learner.py:213-233).
__global__ __launch_bounds__(256) void k_td_target_mt(
    const float* __restrict__ r, const float* __restrict__ d,
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ lp, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, float* __restrict__ y,
    int n, int T, int oh_stride, float gamma, float rs) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int t_i = 0;
  if (T > 1) {
    float best = -1e30f;
    for (int t = 0; t < T; ++t) {
      const float v = onehot[(long)i * oh_stride + t];
      if (v > best) { best = v; t_i = t; }
    }
  }
  const float alpha = __expf(log_alpha[t_i]);
  y[i] = rs * r[i] + gamma * (1.f - d[i]) * (fminf(q1[i], q2[i]) - alpha * lp[i]);
}

// ---------------------------------------------------------------------------
// K6: fused SAC loss reductions.  Single 256-thread workgroup loops the
// batch; LDS tree reductions.  T <= 32 tasks.
//
// Task machinery (matches reference MT10_Distributed_MTSAC/src/model.py:
// 99-116 and learner.py:213-233 exactly):
//   t_i      = argmax over one-hot suffix of states rows
//   alpha_i  = exp(log_alpha[t_i])           (per-sample temperature)
//   w_raw_i  = softmax(-exp(log_alpha))[t_i] (weighted-loss path)
//   coeff_i  = use_w ? (w_raw_i/sum w_raw)/B : 1/B
//
// The critic loss runs BEFORE the critic Adam step and the actor/alpha
// losses AFTER it (reference update_SAC ordering), so they are separate
// kernel pairs.
// ---------------------------------------------------------------------------

__device__ __forceinline__ int task_of_row(const float* __restrict__ onehot,
                                           long i, int oh_stride, int T) {
  if (T <= 1) return 0;
  int t_i = 0;
  float best = -1e30f;
  for (int t = 0; t < T; ++t) {
    const float v = onehot[i * oh_stride + t];
    if (v > best) { best = v; t_i = t; }
  }
  return t_i;
}

// softmax(-exp(log_alpha)) gathered at task t (recomputed per thread; T<=32)
__device__ __forceinline__ float task_weight(const float* __restrict__ la,
                                             int T, int t_i) {
  float mx = -1e30f;
  for (int t = 0; t < T; ++t) mx = fmaxf(mx, -__expf(la[t]));
  float den = 0.f;
  for (int t = 0; t < T; ++t) den += __expf(-__expf(la[t]) - mx);
  return __expf(-__expf(la[t_i]) - mx) / den;
}


// thread 0 fills smw[0..T) with softmax(-exp(log_alpha)); call before use,
// followed by __syncthreads().
__device__ __forceinline__ void fill_task_weights(
    float* __restrict__ smw, const float* __restrict__ la, int T) {
  if (threadIdx.x == 0) {
    float mx = -1e30f;
    for (int t = 0; t < T; ++t) mx = fmaxf(mx, -__expf(la[t]));
    float den = 0.f;
    for (int t = 0; t < T; ++t) {
      smw[t] = __expf(-__expf(la[t]) - mx);
      den += smw[t];
    }
    for (int t = 0; t < T; ++t) smw[t] /= den;
  }
}




// single-workgroup fused variant: 1024 threads cover B<=few-thousand rows
// in <=4 strides, LDS tree reduce, thread 0 writes the FINALIZED values —
// one launch replaces zero + multi-block-atomic fwd + finalize (the
// multi-block version was only ever launch-bound at these batch sizes)
// wavefront sum: 6 DPP/shuffle steps, no LDS, no barrier
__device__ inline float wave_sum64(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__global__ __launch_bounds__(256) void k_critic_loss_fwd_1wg(
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ y, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, float* __restrict__ out,
    int B, int T, int oh_stride, int use_w) {
  __shared__ float part[3][4];
  __shared__ float smw[32];
  const int tid = threadIdx.x;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  if (use_w) __syncthreads();
  float s_l1 = 0.f, s_l2 = 0.f, s_w = 0.f;
  for (int i = tid; i < B; i += 256) {
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    s_w += w_raw;
    const float d1 = y[i] - q1[i], d2 = y[i] - q2[i];
    s_l1 += w_raw * d1 * d1;
    s_l2 += w_raw * d2 * d2;
  }
  // wave-level shuffles + ONE barrier (a 256-thread LDS reduce tree
  // measured 16-19 us from barrier latency alone on one CU)
  s_l1 = wave_sum64(s_l1); s_l2 = wave_sum64(s_l2); s_w = wave_sum64(s_w);
  const int wid = tid >> 6;
  if ((tid & 63) == 0) {
    part[0][wid] = s_l1; part[1][wid] = s_l2; part[2][wid] = s_w;
  }
  __syncthreads();
  if (tid == 0) {
    const float r0 = part[0][0] + part[0][1] + part[0][2] + part[0][3];
    const float r1 = part[1][0] + part[1][1] + part[1][2] + part[1][3];
    const float r2 = part[2][0] + part[2][1] + part[2][2] + part[2][3];
    const float wsum = use_w ? r2 : 1.f;
    const float denom = wsum * (float)B;
    out[3] = r0; out[4] = r1; out[5] = r2;
    out[0] = r0 / denom;
    out[1] = r1 / denom;
    out[2] = wsum;
    out[6] = (r0 + r1) / denom;
  }
}

// bf16 stacked-output variant for the manual backward: dq[2,B] bf16
__global__ __launch_bounds__(256) void k_critic_loss_bwd2(
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ y, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, const float* __restrict__ saved,
    unsigned short* __restrict__ dq, int B, int T, int oh_stride,
    int use_w) {
  __shared__ float smw[32];
  if (use_w) fill_task_weights(smw, log_alpha, T);
  __syncthreads();
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const int t_i = task_of_row(onehot, i, oh_stride, T);
  const float w_raw = use_w ? smw[t_i] : 1.f;
  const float coeff = (use_w ? w_raw / saved[2] : 1.f) / (float)B;
  auto cvt = [](float f) -> unsigned short {
    union { float f; unsigned u; } v{f};
    unsigned u = v.u;
    u += 0x7FFFu + ((u >> 16) & 1u);
    return (unsigned short)(u >> 16);
  };
  dq[i] = cvt(coeff * -2.f * (y[i] - q1[i]));
  dq[B + i] = cvt(coeff * -2.f * (y[i] - q2[i]));
}

// dq1_i = g1 * coeff_i * -2 (y_i - q1_i);  dq2 likewise with g2
__global__ __launch_bounds__(256) void k_critic_loss_bwd(
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ y, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, const float* __restrict__ saved,
    const float* __restrict__ gscale, float* __restrict__ dq1,
    float* __restrict__ dq2, int B, int T, int oh_stride, int use_w) {
  __shared__ float smw[32];
  if (use_w) fill_task_weights(smw, log_alpha, T);
  __syncthreads();
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const int t_i = task_of_row(onehot, i, oh_stride, T);
  const float w_raw = use_w ? smw[t_i] : 1.f;
  const float coeff = (use_w ? w_raw / saved[2] : 1.f) / (float)B;
  dq1[i] = gscale[0] * coeff * -2.f * (y[i] - q1[i]);
  dq2[i] = gscale[1] * coeff * -2.f * (y[i] - q2[i]);
}



// single-workgroup fused variant (see k_critic_loss_fwd_1wg); writes
// finalized out[0..3] + raw sums out[4..7] in one launch.  If dla is
// non-null its first dla_n floats are zeroed here — the alpha gradient
// buffer the subsequent k_actor_alpha_loss_bwd* atomics target — so the
// engine needs no separate fill launch either.
// multi-block variant: each block reduces its stripe with wave shuffles
// (one barrier), writes a private partial slot in ws, and the LAST block
// (ticket in ws[0], self-resetting so graph replays stay correct) sums
// the slots and writes the finalized outputs.  One launch, full CU
// parallelism — the 1wg variant above is the fallback when no workspace
// is provided (eager/legacy callers) and is latency-bound at ~4 waves.
__global__ __launch_bounds__(256) void k_critic_loss_fwd_mb(
    const float* __restrict__ q1, const float* __restrict__ q2,
    const float* __restrict__ y, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, float* __restrict__ out,
    float* __restrict__ ws, int B, int T, int oh_stride, int use_w) {
  __shared__ float part[3][4];
  __shared__ float smw[32];
  const int tid = threadIdx.x;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  if (use_w) __syncthreads();
  float s_l1 = 0.f, s_l2 = 0.f, s_w = 0.f;
  for (int i = blockIdx.x * 256 + tid; i < B; i += 256 * gridDim.x) {
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    s_w += w_raw;
    const float d1 = y[i] - q1[i], d2 = y[i] - q2[i];
    s_l1 += w_raw * d1 * d1;
    s_l2 += w_raw * d2 * d2;
  }
  s_l1 = wave_sum64(s_l1); s_l2 = wave_sum64(s_l2); s_w = wave_sum64(s_w);
  const int wid = tid >> 6;
  if ((tid & 63) == 0) {
    part[0][wid] = s_l1; part[1][wid] = s_l2; part[2][wid] = s_w;
  }
  __syncthreads();
  if (tid == 0) {
    float* slots = ws + 1;
    slots[blockIdx.x * 3 + 0] =
        part[0][0] + part[0][1] + part[0][2] + part[0][3];
    slots[blockIdx.x * 3 + 1] =
        part[1][0] + part[1][1] + part[1][2] + part[1][3];
    slots[blockIdx.x * 3 + 2] =
        part[2][0] + part[2][1] + part[2][2] + part[2][3];
    __threadfence();
    const unsigned old = atomicAdd((unsigned*)ws, 1u);
    if (old == (unsigned)gridDim.x - 1u) {
      *(unsigned*)ws = 0u;  // reset the ticket for the next replay
      __threadfence();
      float r0 = 0.f, r1 = 0.f, r2 = 0.f;
      for (int b = 0; b < (int)gridDim.x; ++b) {
        r0 += slots[b * 3 + 0];
        r1 += slots[b * 3 + 1];
        r2 += slots[b * 3 + 2];
      }
      const float wsum = use_w ? r2 : 1.f;
      const float denom = wsum * (float)B;
      out[3] = r0; out[4] = r1; out[5] = r2;
      out[0] = r0 / denom;
      out[1] = r1 / denom;
      out[2] = wsum;
      out[6] = (r0 + r1) / denom;  // logged sum (no separate add launch)
    }
  }
}

__global__ __launch_bounds__(256) void k_actor_alpha_loss_fwd_mb(
    const float* __restrict__ aq1, const float* __restrict__ aq2,
    const float* __restrict__ lp, const float* __restrict__ ls,
    const float* __restrict__ onehot, const float* __restrict__ log_alpha,
    float* __restrict__ out, float* __restrict__ ws,
    float* __restrict__ dla, int dla_n,
    int B, int T, int A, int oh_stride, int use_w, float H_bar) {
  __shared__ float part[4][4];
  __shared__ float smw[32];
  const int tid = threadIdx.x;
  if (blockIdx.x == 0 && dla != nullptr && tid < dla_n) dla[tid] = 0.f;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  if (use_w) __syncthreads();
  constexpr float CE = 1.4189385332046727f;  // 0.5*(1+log(2*pi))
  float s_pl = 0.f, s_w = 0.f, s_al = 0.f, s_en = 0.f;
  for (int i = blockIdx.x * 256 + tid; i < B; i += 256 * gridDim.x) {
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float la = log_alpha[t_i];
    const float alpha_i = __expf(la);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    s_w += w_raw;
    const float qmin = fminf(aq1[i], aq2[i]);
    s_pl += w_raw * -(qmin - alpha_i * lp[i]);
    s_al += la * (lp[i] + H_bar);
    float ent = CE * A;
    for (int a = 0; a < A; ++a) ent += ls[(long)i * A + a];
    s_en += ent;
  }
  s_pl = wave_sum64(s_pl); s_w = wave_sum64(s_w);
  s_al = wave_sum64(s_al); s_en = wave_sum64(s_en);
  const int wid = tid >> 6;
  if ((tid & 63) == 0) {
    part[0][wid] = s_pl; part[1][wid] = s_w;
    part[2][wid] = s_al; part[3][wid] = s_en;
  }
  __syncthreads();
  if (tid == 0) {
    float* slots = ws + 1;
#pragma unroll
    for (int r = 0; r < 4; ++r)
      slots[blockIdx.x * 4 + r] =
          part[r][0] + part[r][1] + part[r][2] + part[r][3];
    __threadfence();
    const unsigned old = atomicAdd((unsigned*)ws, 1u);
    if (old == (unsigned)gridDim.x - 1u) {
      *(unsigned*)ws = 0u;
      __threadfence();
      float r0 = 0.f, r1 = 0.f, r2 = 0.f, r3 = 0.f;
      for (int b = 0; b < (int)gridDim.x; ++b) {
        r0 += slots[b * 4 + 0]; r1 += slots[b * 4 + 1];
        r2 += slots[b * 4 + 2]; r3 += slots[b * 4 + 3];
      }
      const float wsum = use_w ? r1 : 1.f;
      out[4] = r0; out[5] = r1; out[6] = r2; out[7] = r3;
      out[0] = r0 / (wsum * (float)B);
      out[1] = wsum;
      out[2] = -r2 / (float)B;
      out[3] = r3 / (float)B;
    }
  }
}

__global__ __launch_bounds__(256) void k_actor_alpha_loss_fwd_1wg(
    const float* __restrict__ aq1, const float* __restrict__ aq2,
    const float* __restrict__ lp, const float* __restrict__ ls,
    const float* __restrict__ onehot, const float* __restrict__ log_alpha,
    float* __restrict__ out, float* __restrict__ dla, int dla_n,
    int B, int T, int A, int oh_stride, int use_w, float H_bar) {
  __shared__ float part[4][4];
  __shared__ float smw[32];
  const int tid = threadIdx.x;
  if (dla != nullptr && tid < dla_n) dla[tid] = 0.f;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  if (use_w) __syncthreads();
  constexpr float CE = 1.4189385332046727f;  // 0.5*(1+log(2*pi))
  float s_pl = 0.f, s_w = 0.f, s_al = 0.f, s_en = 0.f;
  for (int i = tid; i < B; i += 256) {
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float la = log_alpha[t_i];
    const float alpha_i = __expf(la);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    s_w += w_raw;
    const float qmin = fminf(aq1[i], aq2[i]);
    s_pl += w_raw * -(qmin - alpha_i * lp[i]);
    s_al += la * (lp[i] + H_bar);
    float ent = CE * A;
    for (int a = 0; a < A; ++a) ent += ls[(long)i * A + a];
    s_en += ent;
  }
  s_pl = wave_sum64(s_pl); s_w = wave_sum64(s_w);
  s_al = wave_sum64(s_al); s_en = wave_sum64(s_en);
  const int wid = tid >> 6;
  if ((tid & 63) == 0) {
    part[0][wid] = s_pl; part[1][wid] = s_w;
    part[2][wid] = s_al; part[3][wid] = s_en;
  }
  __syncthreads();
  if (tid == 0) {
    const float r0 = part[0][0] + part[0][1] + part[0][2] + part[0][3];
    const float r1 = part[1][0] + part[1][1] + part[1][2] + part[1][3];
    const float r2 = part[2][0] + part[2][1] + part[2][2] + part[2][3];
    const float r3 = part[3][0] + part[3][1] + part[3][2] + part[3][3];
    const float wsum = use_w ? r1 : 1.f;
    out[4] = r0; out[5] = r1; out[6] = r2; out[7] = r3;
    out[0] = r0 / (wsum * (float)B);
    out[1] = wsum;
    out[2] = -r2 / (float)B;
    out[3] = r3 / (float)B;
  }
}

// daq1_i = gp*coeff_i*-(aq1<=aq2); daq2_i = gp*coeff_i*-(aq2<aq1)
// dlp_i  = gp*coeff_i*alpha_i      (alpha detached in actor loss)
// dla[t] = gal * -(1/B) sum_{i in t}(lp_i + H_bar)   (lp detached)
__global__ __launch_bounds__(256) void k_actor_alpha_loss_bwd(
    const float* __restrict__ aq1, const float* __restrict__ aq2,
    const float* __restrict__ lp, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, const float* __restrict__ saved,
    const float* __restrict__ gscale, float* __restrict__ daq1,
    float* __restrict__ daq2, float* __restrict__ dlp,
    float* __restrict__ dla, int B, int T, int oh_stride, int use_w,
    float H_bar) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int tid = threadIdx.x;
  __shared__ float s_dla[32];
  __shared__ float smw[32];
  if (tid < T) s_dla[tid] = 0.f;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  __syncthreads();
  if (i < B) {
    const float gp = gscale[0], gal = gscale[1];
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float alpha_i = __expf(log_alpha[t_i]);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    const float coeff = (use_w ? w_raw / saved[1] : 1.f) / (float)B;
    const bool first = aq1[i] <= aq2[i];
    daq1[i] = first ? gp * coeff * -1.f : 0.f;
    daq2[i] = first ? 0.f : gp * coeff * -1.f;
    dlp[i] = gp * coeff * alpha_i;
    atomicAdd(&s_dla[t_i], gal * -(lp[i] + H_bar) / (float)B);
  }
  __syncthreads();
  if (tid < T) atomicAdd(&dla[tid], s_dla[tid]);
}

// manual-backward actor/alpha variant: daq[2,B] bf16 stacked, dlp fp32,
// dla accumulated into the caller buffer (zeroed by caller).
__global__ __launch_bounds__(256) void k_actor_alpha_loss_bwd2(
    const float* __restrict__ aq1, const float* __restrict__ aq2,
    const float* __restrict__ lp, const float* __restrict__ onehot,
    const float* __restrict__ log_alpha, const float* __restrict__ saved,
    unsigned short* __restrict__ daq, float* __restrict__ dlp,
    float* __restrict__ dla, int B, int T, int oh_stride, int use_w,
    float H_bar) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int tid = threadIdx.x;
  __shared__ float s_dla[32];
  __shared__ float smw[32];
  if (tid < T) s_dla[tid] = 0.f;
  if (use_w) fill_task_weights(smw, log_alpha, T);
  __syncthreads();
  if (i < B) {
    const int t_i = task_of_row(onehot, i, oh_stride, T);
    const float alpha_i = __expf(log_alpha[t_i]);
    const float w_raw = use_w ? smw[t_i] : 1.f;
    const float coeff = (use_w ? w_raw / saved[1] : 1.f) / (float)B;
    const bool first = aq1[i] <= aq2[i];
    auto cvt = [](float f) -> unsigned short {
      union { float f; unsigned u; } v{f};
      unsigned u = v.u;
      u += 0x7FFFu + ((u >> 16) & 1u);
      return (unsigned short)(u >> 16);
    };
    daq[i] = cvt(first ? coeff * -1.f : 0.f);
    daq[B + i] = cvt(first ? 0.f : coeff * -1.f);
    dlp[i] = coeff * alpha_i;
    atomicAdd(&s_dla[t_i], -(lp[i] + H_bar) / (float)B);
  }
  __syncthreads();
  if (tid < T) atomicAdd(&dla[tid], s_dla[tid]);
}

// squash backward emitting the joined actor-head gradient [B, 2A] bf16
// ([dmu | dlog_std_raw] — the mu_log_std_layer output layout)
__global__ __launch_bounds__(256) void k_squash_bwd2(
    const float* __restrict__ ga, const float* __restrict__ gl,
    const float* __restrict__ lsr, const float* __restrict__ ls,
    const float* __restrict__ eps, const float* __restrict__ tanh_u,
    unsigned short* __restrict__ dhead, long lsr_ld, int ga_twin, int B,
    int A, float k) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const float g = gl[i];
  auto cvt = [](float f) -> unsigned short {
    union { float f; unsigned u; } v{f};
    unsigned u = v.u;
    u += 0x7FFFu + ((u >> 16) & 1u);
    return (unsigned short)(u >> 16);
  };
  for (int a = 0; a < A; ++a) {
    const long idx = (long)i * A + a;
    const float t = tanh_u[idx];
    const float omt2 = 1.f - t * t;
    const float dlp_du = 2.f * t * omt2 / (omt2 + 1e-6f);
    const float s = __expf(ls[idx]);
    const float e = eps[idx];
    // ga_twin: ga is the [2,B,A] action-column grad straight from the
    // twin-critic chain backward; summing the two heads here kills the
    // dx0[0]+dx0[1] add launch
    const float gav = ga_twin ? ga[idx] + ga[(long)B * A + idx] : ga[idx];
    const float du = gav * k * omt2 + g * dlp_du;
    const float raw = lsr[(long)i * lsr_ld + a];
    const float mask = (raw >= -20.f && raw <= 2.f) ? 1.f : 0.f;
    dhead[(long)i * 2 * A + a] = cvt(du);
    dhead[(long)i * 2 * A + A + a] = cvt(mask * (du * e * s - g));
  }
}

// ---------------------------------------------------------------------------
// K9: fused Adam over one flat buffer (torch.optim.Adam numerics).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_adam(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v, long n,
    float b1, float b2, float step_size, float inv_sqrt_bc2, float eps) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float gi = g[i];
  const float mi = b1 * m[i] + (1.f - b1) * gi;
  const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
  m[i] = mi;
  v[i] = vi;
  p[i] -= step_size * mi / (sqrtf(vi) * inv_sqrt_bc2 + eps);
}

// K9b: graph-capturable Adam — the step counter lives on-device so a
// captured update's bias correction advances across hipGraph replays.
__global__ void k_adam_prolog(float* __restrict__ state, float lr, float b1,
                              float b2) {
  const float step = state[0] + 1.f;
  state[0] = step;
  state[1] = lr / (1.f - __powf(b1, step));          // step_size
  state[2] = 1.f / sqrtf(1.f - __powf(b2, step));    // inv_sqrt_bc2
}

__device__ __forceinline__ unsigned short f32_bf16_rne_d(float f) {
  union { float f; unsigned u; } v{f};
  unsigned u = v.u;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return (unsigned short)(u >> 16);
}

// mir (optional): the group's bf16 compute mirror — emitting the refreshed
// mirror from the SAME kernel kills the separate full-buffer cast launch.
__global__ __launch_bounds__(256) void k_adam_dev(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ state, long n, float b1, float b2, float eps,
    unsigned short* __restrict__ mir) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float step_size = state[1], inv_sqrt_bc2 = state[2];
  const float gi = g[i];
  const float mi = b1 * m[i] + (1.f - b1) * gi;
  const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
  m[i] = mi;
  v[i] = vi;
  const float pn = p[i] - step_size * mi / (sqrtf(vi) * inv_sqrt_bc2 + eps);
  p[i] = pn;
  if (mir != nullptr) mir[i] = f32_bf16_rne_d(pn);
}

// K9c: all three optimizer groups (critic was stepped earlier; this one
// covers actor+alpha+optional context in ONE launch) — or any <=3 groups.
struct AdamGroup {
  float* p; const float* g; float* m; float* v; const float* st; long n;
};

// one launch advances EVERY optimizer's Adam state for this update
// (critic + actor + alpha [+ context]) and bumps the device RNG counter:
// the engine calls it once at the top of seg2, and the per-optimizer
// steppers skip their own prologs (skip_prolog)
__global__ void k_adam_prolog_many(long long* rng,
                                   float* s0, float* s1, float* s2,
                                   float* s3,
                                   float lr0, float lr1, float lr2,
                                   float lr3, float b1, float b2) {
  if (rng != nullptr && threadIdx.x == 0) rng[0] += 1;
  const int i = threadIdx.x;
  float* sv[4] = {s0, s1, s2, s3};
  const float lrv[4] = {lr0, lr1, lr2, lr3};
  if (i >= 4 || sv[i] == nullptr) return;
  float* st = sv[i];
  const float step = st[0] + 1.f;
  st[0] = step;
  st[1] = lrv[i] / (1.f - __powf(b1, step));
  st[2] = 1.f / sqrtf(1.f - __powf(b2, step));
}

__global__ void k_adam_prolog3(long long* rng, float* s0, float* s1,
                               float* s2,
                               float lr0, float lr1, float lr2,
                               float b1, float b2) {
  if (rng != nullptr && threadIdx.x == 0) rng[0] += 1;
  const int i = threadIdx.x;
  float* s = i == 0 ? s0 : (i == 1 ? s1 : s2);
  const float lr = i == 0 ? lr0 : (i == 1 ? lr1 : lr2);
  if (s == nullptr) return;
  const float step = s[0] + 1.f;
  s[0] = step;
  s[1] = lr / (1.f - __powf(b1, step));
  s[2] = 1.f / sqrtf(1.f - __powf(b2, step));
}

__global__ __launch_bounds__(256) void k_adam_multi(
    float* p0, const float* g0, float* m0, float* v0, const float* st0, long n0,
    float* p1, const float* g1, float* m1, float* v1, const float* st1, long n1,
    float* p2, const float* g2, float* m2, float* v2, const float* st2, long n2,
    float b1, float b2, float eps, unsigned short* mr0, unsigned short* mr1,
    unsigned short* mr2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  float* p; const float* g; float* m; float* v; const float* st;
  unsigned short* mr;
  if (i < n0) { p = p0; g = g0; m = m0; v = v0; st = st0; mr = mr0; }
  else if ((i -= n0) < n1) { p = p1; g = g1; m = m1; v = v1; st = st1; mr = mr1; }
  else if ((i -= n1) < n2) { p = p2; g = g2; m = m2; v = v2; st = st2; mr = mr2; }
  else return;
  const float gi = g[i];
  const float mi = b1 * m[i] + (1.f - b1) * gi;
  const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
  m[i] = mi;
  v[i] = vi;
  const float pn = p[i] - st[1] * mi / (sqrtf(vi) * st[2] + eps);
  p[i] = pn;
  if (mr != nullptr) mr[i] = f32_bf16_rne_d(pn);
}

// ---------------------------------------------------------------------------
// K10: t = (1-tau)*t + tau*s over flat buffers.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_polyak(
    float* __restrict__ t, const float* __restrict__ s, long n, float tau,
    unsigned short* __restrict__ mir) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float tn = (1.f - tau) * t[i] + tau * s[i];
  t[i] = tn;
  if (mir != nullptr) mir[i] = f32_bf16_rne_d(tn);
}

// ===========================================================================
// Host wrappers
// ===========================================================================

static torch::Tensor linear_act_fwd_g(torch::Tensor x, torch::Tensor w,
                                      torch::Tensor b, long act, long G) {
  CHECK_IN(x); CHECK_IN(w); CHECK_IN(b);
  auto xc = x.contiguous(); auto wc = w.contiguous(); auto bc = b.contiguous();
  const bool per_group_x = xc.dim() == 3;  // [G,B,K] vs shared [B,K]
  const long M = per_group_x ? xc.size(1) : xc.size(0);
  const long K = xc.size(-1);
  const long N = wc.numel() / (G * K);
  TORCH_CHECK(wc.numel() == G * N * K && bc.numel() == G * N,
              "grouped weight shape mismatch");
  const long xgs = per_group_x ? M * K : 0;
  auto y = G == 1 ? torch::empty({M, N}, xc.options())
                  : torch::empty({G, M, N}, xc.options());
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN, G);
  hipLaunchKernelGGL(k_linear_act_fwd, grid, dim3(256), 0, cur_stream(),
                     xc.data_ptr<float>(), wc.data_ptr<float>(),
                     bc.data_ptr<float>(), y.data_ptr<float>(),
                     (int)M, (int)N, (int)K, (int)act, xgs);
  return y;
}

static torch::Tensor linear_act_fwd(torch::Tensor x, torch::Tensor w,
                                    torch::Tensor b, long act) {
  return linear_act_fwd_g(x, w, b, act, 1);
}

static torch::Tensor linear_bwd_dx_g(torch::Tensor dy, torch::Tensor w,
                                     torch::Tensor yout, long act, long G,
                                     long sum_over_g) {
  CHECK_IN(dy); CHECK_IN(w); CHECK_IN(yout);
  auto dyc = dy.contiguous(); auto wc = w.contiguous();
  auto yc = yout.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  const long N = G == 1 ? dyc.size(1) : dyc.size(2);
  const long K = wc.size(-1);
  // sum mode: one grid, g-loop accumulates (shared-x layer).  per-group
  // mode: grid.z = G independent dx outputs [G,M,K].
  const long Gz = (G == 1 || sum_over_g) ? 1 : G;
  const int Gin = (int)((G > 1 && sum_over_g) ? G : 1);
  const long nbx = (M + BM - 1) / BM, nby = (K + BN - 1) / BN;
  const long nt_all = Gin * ((N + BK - 1) / BK);
  // split-reduce the tile loop when the grid underfills the 256-CU chip
  long S = 1;
  const long base_blocks = nbx * nby * Gz;
  if (base_blocks < 256) {
    S = std::min<long>(nt_all, std::max<long>(1, 512 / base_blocks));
  }
  auto dx = Gz == 1 ? torch::empty({M, K}, dyc.options())
                    : torch::empty({Gz, M, K}, dyc.options());
  auto out = dx;
  if (S > 1) out = torch::empty({Gz * S, M, K}, dyc.options());
  dim3 grid(nbx, nby, Gz * S);
  hipLaunchKernelGGL(k_linear_bwd_dx, grid, dim3(256), 0, cur_stream(),
                     dyc.data_ptr<float>(), wc.data_ptr<float>(),
                     yc.data_ptr<float>(), out.data_ptr<float>(),
                     (int)M, (int)N, (int)K, (int)act, Gin, (int)S);
  if (S > 1) {
    const long nmk = M * K;
    dim3 g1((nmk + 255) / 256, Gz);
    hipLaunchKernelGGL(k_reduce_partials, g1, dim3(256), 0, cur_stream(),
                       out.data_ptr<float>(), dx.data_ptr<float>(), nmk,
                       (int)S, nmk);
  }
  return dx;
}

static torch::Tensor linear_bwd_dx(torch::Tensor dy, torch::Tensor w,
                                   torch::Tensor yout, long act) {
  return linear_bwd_dx_g(dy, w, yout, act, 1, 1);
}

static std::vector<torch::Tensor> linear_bwd_dwdb_g(torch::Tensor dy,
                                                    torch::Tensor x,
                                                    torch::Tensor yout,
                                                    long act, long G) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(yout);
  auto dyc = dy.contiguous(); auto xc = x.contiguous();
  auto yc = yout.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  const long N = G == 1 ? dyc.size(1) : dyc.size(2);
  const long K = xc.size(-1);
  const long xgs = xc.dim() == 3 ? M * K : 0;
  // split-K over the batch: pick S so total blocks ~ 2x CU count (just
  // enough to fill the chip; larger S doubles the partial-slab traffic)
  const long nbx = (N + BM - 1) / BM, nby = (K + BN - 1) / BN;
  long S = std::max<long>(1, 512 / std::max<long>(1, nbx * nby * G));
  S = std::min<long>(S, (M + BK - 1) / BK);
  const int chunk = (int)(((M + S - 1) / S + BK - 1) / BK * BK);
  S = (M + chunk - 1) / chunk;
  auto ws = torch::empty({G * S, N, K}, dyc.options());
  auto ws_db = torch::empty({G * S, N}, dyc.options());
  dim3 grid((N + BM - 1) / BM, (K + BN - 1) / BN, G * S);
  hipLaunchKernelGGL(k_linear_bwd_dwdb_splitk, grid, dim3(256), 0,
                     cur_stream(), dyc.data_ptr<float>(), xc.data_ptr<float>(),
                     yc.data_ptr<float>(), ws.data_ptr<float>(),
                     ws_db.data_ptr<float>(), (int)M, (int)N, (int)K,
                     (int)act, (int)S, chunk, xgs);
  auto dw = G == 1 ? torch::empty({N, K}, dyc.options())
                   : torch::empty({G, N, K}, dyc.options());
  auto db = G == 1 ? torch::empty({N}, dyc.options())
                   : torch::empty({G, N}, dyc.options());
  if (S == 1) {
    dw.copy_(ws.view_as(dw));
    db.copy_(ws_db.view_as(db));
  } else {
    const long nw = N * K;
    dim3 g1((nw + 255) / 256, G);
    hipLaunchKernelGGL(k_reduce_partials, g1, dim3(256), 0, cur_stream(),
                       ws.data_ptr<float>(), dw.data_ptr<float>(), nw,
                       (int)S, nw);
    dim3 g2((N + 255) / 256, G);
    hipLaunchKernelGGL(k_reduce_partials, g2, dim3(256), 0, cur_stream(),
                       ws_db.data_ptr<float>(), db.data_ptr<float>(), N,
                       (int)S, N);
  }
  return {dw, db};
}

static std::vector<torch::Tensor> linear_bwd_dwdb(torch::Tensor dy,
                                                  torch::Tensor x,
                                                  torch::Tensor yout,
                                                  long act) {
  return linear_bwd_dwdb_g(dy, x, yout, act, 1);
}

static std::vector<torch::Tensor> replay_sample(
    torch::Tensor states, torch::Tensor actions, torch::Tensor rewards,
    torch::Tensor next_states, torch::Tensor dones, torch::Tensor sizes,
    torch::Tensor rnd, long B,
    c10::optional<torch::Tensor> rng = c10::nullopt) {
  CHECK_IN(states); CHECK_IN(sizes);
  const bool krng = rng.has_value() && rng->numel() > 0;
  if (!krng) CHECK_IN(rnd);
  TORCH_CHECK(!krng || rng->scalar_type() == torch::kInt64,
              "rng counter must be int64");
  const long T = states.size(0), cap = states.size(1);
  const long Ds = states.size(2), Da = actions.size(2);
  const int per = (int)(B / T);
  TORCH_CHECK(per * T == B, "batch must divide by num_tasks");
  auto opt = states.options();
  auto o_s = torch::empty({B, Ds}, opt);
  auto o_a = torch::empty({B, Da}, opt);
  auto o_r = torch::empty({B, 1}, opt);
  auto o_ns = torch::empty({B, Ds}, opt);
  auto o_d = torch::empty({B, 1}, opt);
  hipLaunchKernelGGL(k_replay_sample, dim3((B + 3) / 4), dim3(256), 0,
                     cur_stream(), states.data_ptr<float>(),
                     actions.data_ptr<float>(), rewards.data_ptr<float>(),
                     next_states.data_ptr<float>(), dones.data_ptr<float>(),
                     sizes.data_ptr<float>(),
                     krng ? nullptr : rnd.data_ptr<float>(),
                     krng ? (const long long*)rng->data_ptr<long>()
                          : nullptr,
                     o_s.data_ptr<float>(), o_a.data_ptr<float>(),
                     o_r.data_ptr<float>(), o_ns.data_ptr<float>(),
                     o_d.data_ptr<float>(), (int)B, (int)T, per, cap,
                     (int)Ds, (int)Da);
  return {o_s, o_a, o_r, o_ns, o_d};
}

static std::vector<torch::Tensor> squashed_gaussian_fwd(
    torch::Tensor mu, torch::Tensor lsr, torch::Tensor eps, double k,
    c10::optional<torch::Tensor> rng = c10::nullopt) {
  CHECK_IN(mu); CHECK_IN(lsr); CHECK_IN(eps);
  const bool krng = rng.has_value() && rng->numel() > 0;
  TORCH_CHECK(!krng || (rng->scalar_type() == torch::kInt64
                        && eps.is_contiguous()),
              "krng mode needs an int64 counter and a contiguous eps out");
  const bool mu_ok = mu.dim() == 2 && mu.stride(1) == 1;
  const bool ls_ok = lsr.dim() == 2 && lsr.stride(1) == 1;
  auto muc = mu_ok ? mu : mu.contiguous();
  auto lc = ls_ok ? lsr : lsr.contiguous();
  auto ec = eps.contiguous();
  const long B = muc.size(0), A = muc.size(1);
  const long mu_ld = mu_ok ? mu.stride(0) : A;
  const long ls_ld = ls_ok ? lsr.stride(0) : A;
  TORCH_CHECK(A <= 32, "action_dim too large for fused kernel");
  auto opt = muc.options();
  auto act = torch::empty({B, A}, opt);
  auto logp = torch::empty({B, 1}, opt);
  auto tanh_u = torch::empty({B, A}, opt);
  auto ls_out = torch::empty({B, A}, opt);
  const int grid = (B + 255) / 256;
  hipLaunchKernelGGL(k_squash_fwd, dim3(grid), dim3(256), 0, cur_stream(),
                     muc.data_ptr<float>(), lc.data_ptr<float>(),
                     ec.data_ptr<float>(), act.data_ptr<float>(),
                     logp.data_ptr<float>(), tanh_u.data_ptr<float>(),
                     ls_out.data_ptr<float>(),
                     krng ? (const long long*)rng->data_ptr<long>()
                          : nullptr,
                     mu_ld, ls_ld, (int)B, (int)A, (float)k);
  return {act, logp, tanh_u, ls_out};
}

static std::vector<torch::Tensor> squashed_gaussian_bwd(
    torch::Tensor ga, torch::Tensor gl, torch::Tensor lsr, torch::Tensor ls,
    torch::Tensor eps, torch::Tensor tanh_u, double k) {
  CHECK_IN(ga); CHECK_IN(gl);
  auto gac = ga.contiguous(); auto glc = gl.contiguous();
  auto lsrc = lsr.contiguous(); auto lsc = ls.contiguous();
  auto ec = eps.contiguous(); auto tc = tanh_u.contiguous();
  const long B = gac.size(0), A = gac.size(1);
  auto dmu = torch::empty_like(gac);
  auto dlsr = torch::empty_like(gac);
  const int grid = (B + 255) / 256;
  hipLaunchKernelGGL(k_squash_bwd, dim3(grid), dim3(256), 0, cur_stream(),
                     gac.data_ptr<float>(), glc.data_ptr<float>(),
                     lsrc.data_ptr<float>(), lsc.data_ptr<float>(),
                     ec.data_ptr<float>(), tc.data_ptr<float>(),
                     dmu.data_ptr<float>(), dlsr.data_ptr<float>(),
                     (int)B, (int)A, (float)k);
  return {dmu, dlsr};
}

static torch::Tensor td_target(torch::Tensor r, torch::Tensor d,
                               torch::Tensor q1, torch::Tensor q2,
                               torch::Tensor lp, torch::Tensor alpha,
                               double gamma, double rs) {
  CHECK_IN(r);
  auto rc = r.contiguous(); auto dc = d.contiguous();
  auto q1c = q1.contiguous(); auto q2c = q2.contiguous();
  auto lpc = lp.contiguous(); auto ac = alpha.contiguous();
  const long n = rc.numel();
  TORCH_CHECK(ac.numel() == n, "alpha must be per-sample");
  auto y = torch::empty_like(rc);
  hipLaunchKernelGGL(k_td_target, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), rc.data_ptr<float>(), dc.data_ptr<float>(),
                     q1c.data_ptr<float>(), q2c.data_ptr<float>(),
                     lpc.data_ptr<float>(), ac.data_ptr<float>(),
                     y.data_ptr<float>(), (int)n, (float)gamma, (float)rs);
  return y;
}

static torch::Tensor td_target_mt(torch::Tensor r, torch::Tensor d,
                                  torch::Tensor q1, torch::Tensor q2,
                                  torch::Tensor lp, torch::Tensor states,
                                  torch::Tensor log_alpha, long T,
                                  double gamma, double rs) {
  CHECK_IN(r); CHECK_IN(states); CHECK_IN(log_alpha);
  const long n = r.numel();
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto y = torch::empty_like(r.contiguous());
  hipLaunchKernelGGL(k_td_target_mt, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), r.contiguous().data_ptr<float>(),
                     d.contiguous().data_ptr<float>(),
                     q1.contiguous().data_ptr<float>(),
                     q2.contiguous().data_ptr<float>(),
                     lp.contiguous().data_ptr<float>(), oh,
                     log_alpha.data_ptr<float>(), y.data_ptr<float>(),
                     (int)n, (int)T, (int)oh_stride, (float)gamma,
                     (float)rs);
  return y;
}

static std::vector<torch::Tensor> critic_loss_fwd(
    torch::Tensor q1, torch::Tensor q2, torch::Tensor y,
    torch::Tensor states, torch::Tensor log_alpha, long T, long use_w,
    c10::optional<torch::Tensor> ws = c10::nullopt) {
  CHECK_IN(q1); CHECK_IN(states); CHECK_IN(log_alpha);
  const long B = q1.size(0);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto out = torch::empty({8}, q1.options());
  if (ws.has_value() && ws->numel() >= 1 + 32 * 3) {
    // persistent zero-initialized workspace -> multi-block one-launch
    // path with last-block finalize (ticket self-resets per launch)
    const int nblk = (int)std::min<long>((B + 255) / 256, 32);
    hipLaunchKernelGGL(k_critic_loss_fwd_mb, dim3(nblk), dim3(256), 0,
                       cur_stream(), q1.data_ptr<float>(),
                       q2.data_ptr<float>(), y.data_ptr<float>(), oh,
                       log_alpha.data_ptr<float>(), out.data_ptr<float>(),
                       ws->data_ptr<float>(), (int)B, (int)T,
                       (int)oh_stride, (int)use_w);
  } else {
    hipLaunchKernelGGL(k_critic_loss_fwd_1wg, dim3(1), dim3(256), 0,
                       cur_stream(), q1.data_ptr<float>(),
                       q2.data_ptr<float>(), y.data_ptr<float>(), oh,
                       log_alpha.data_ptr<float>(), out.data_ptr<float>(),
                       (int)B, (int)T, (int)oh_stride, (int)use_w);
  }
  return {out};
}

static std::vector<torch::Tensor> critic_loss_bwd(
    torch::Tensor q1, torch::Tensor q2, torch::Tensor y,
    torch::Tensor states, torch::Tensor log_alpha, torch::Tensor saved,
    torch::Tensor gscale, long T, long use_w) {
  CHECK_IN(q1);
  const long B = q1.size(0);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto dq1 = torch::empty_like(q1);
  auto dq2 = torch::empty_like(q2);
  hipLaunchKernelGGL(k_critic_loss_bwd, dim3((B + 255) / 256), dim3(256), 0,
                     cur_stream(), q1.data_ptr<float>(), q2.data_ptr<float>(),
                     y.data_ptr<float>(), oh, log_alpha.data_ptr<float>(),
                     saved.data_ptr<float>(), gscale.data_ptr<float>(),
                     dq1.data_ptr<float>(), dq2.data_ptr<float>(), (int)B,
                     (int)T, (int)oh_stride, (int)use_w);
  return {dq1, dq2};
}

static torch::Tensor actor_alpha_loss_fwd(
    torch::Tensor aq1, torch::Tensor aq2, torch::Tensor lp, torch::Tensor ls,
    torch::Tensor states, torch::Tensor log_alpha, long T, long use_w,
    double H_bar,
    c10::optional<torch::Tensor> dla = c10::nullopt,
    c10::optional<torch::Tensor> ws = c10::nullopt) {
  CHECK_IN(aq1); CHECK_IN(states); CHECK_IN(log_alpha);
  const long B = aq1.size(0);
  const long A = ls.size(1);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto out = torch::empty({8}, aq1.options());
  float* dla_p = nullptr;
  int dla_n = 0;
  if (dla.has_value() && dla->numel() > 0) {
    CHECK_IN(*dla);
    dla_p = dla->data_ptr<float>();
    dla_n = (int)dla->numel();
  }
  if (ws.has_value() && ws->numel() >= 1 + 32 * 4) {
    const int nblk = (int)std::min<long>((B + 255) / 256, 32);
    hipLaunchKernelGGL(k_actor_alpha_loss_fwd_mb, dim3(nblk), dim3(256), 0,
                       cur_stream(), aq1.data_ptr<float>(),
                       aq2.data_ptr<float>(), lp.data_ptr<float>(),
                       ls.data_ptr<float>(), oh, log_alpha.data_ptr<float>(),
                       out.data_ptr<float>(), ws->data_ptr<float>(), dla_p,
                       dla_n, (int)B, (int)T, (int)A, (int)oh_stride,
                       (int)use_w, (float)H_bar);
  } else {
    hipLaunchKernelGGL(k_actor_alpha_loss_fwd_1wg, dim3(1), dim3(256), 0,
                       cur_stream(), aq1.data_ptr<float>(),
                       aq2.data_ptr<float>(), lp.data_ptr<float>(),
                       ls.data_ptr<float>(), oh, log_alpha.data_ptr<float>(),
                       out.data_ptr<float>(), dla_p, dla_n, (int)B, (int)T,
                       (int)A, (int)oh_stride, (int)use_w, (float)H_bar);
  }
  return out;
}

static std::vector<torch::Tensor> actor_alpha_loss_bwd(
    torch::Tensor aq1, torch::Tensor aq2, torch::Tensor lp,
    torch::Tensor states, torch::Tensor log_alpha, torch::Tensor saved,
    torch::Tensor gscale, long T, long use_w, double H_bar) {
  CHECK_IN(aq1);
  const long B = aq1.size(0);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto daq1 = torch::empty_like(aq1);
  auto daq2 = torch::empty_like(aq2);
  auto dlp = torch::empty_like(lp);
  auto dla = torch::zeros_like(log_alpha);
  hipLaunchKernelGGL(k_actor_alpha_loss_bwd, dim3((B + 255) / 256), dim3(256),
                     0, cur_stream(), aq1.data_ptr<float>(),
                     aq2.data_ptr<float>(), lp.data_ptr<float>(), oh,
                     log_alpha.data_ptr<float>(), saved.data_ptr<float>(),
                     gscale.data_ptr<float>(), daq1.data_ptr<float>(),
                     daq2.data_ptr<float>(), dlp.data_ptr<float>(),
                     dla.data_ptr<float>(), (int)B, (int)T, (int)oh_stride,
                     (int)use_w, (float)H_bar);
  return {daq1, daq2, dlp, dla};
}

static torch::Tensor critic_loss_bwd2(
    torch::Tensor q1, torch::Tensor q2, torch::Tensor y,
    torch::Tensor states, torch::Tensor log_alpha, torch::Tensor saved,
    long T, long use_w) {
  CHECK_IN(q1);
  const long B = q1.size(0);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto dq = torch::empty({2, B, 1}, q1.options().dtype(torch::kBFloat16));
  hipLaunchKernelGGL(k_critic_loss_bwd2, dim3((B + 255) / 256), dim3(256), 0,
                     cur_stream(), q1.data_ptr<float>(), q2.data_ptr<float>(),
                     y.data_ptr<float>(), oh, log_alpha.data_ptr<float>(),
                     saved.data_ptr<float>(),
                     (unsigned short*)dq.data_ptr(), (int)B, (int)T,
                     (int)oh_stride, (int)use_w);
  return dq;
}

static std::vector<torch::Tensor> actor_alpha_loss_bwd2(
    torch::Tensor aq1, torch::Tensor aq2, torch::Tensor lp,
    torch::Tensor states, torch::Tensor log_alpha, torch::Tensor saved,
    torch::Tensor dla_out, long T, long use_w, double H_bar) {
  CHECK_IN(aq1); CHECK_IN(dla_out);
  const long B = aq1.size(0);
  const long oh_stride = states.size(1);
  const float* oh = states.data_ptr<float>() + (oh_stride - T);
  auto daq = torch::empty({2, B, 1}, aq1.options().dtype(torch::kBFloat16));
  auto dlp = torch::empty_like(lp);
  hipLaunchKernelGGL(k_actor_alpha_loss_bwd2, dim3((B + 255) / 256),
                     dim3(256), 0, cur_stream(), aq1.data_ptr<float>(),
                     aq2.data_ptr<float>(), lp.data_ptr<float>(), oh,
                     log_alpha.data_ptr<float>(), saved.data_ptr<float>(),
                     (unsigned short*)daq.data_ptr(), dlp.data_ptr<float>(),
                     dla_out.data_ptr<float>(), (int)B, (int)T,
                     (int)oh_stride, (int)use_w, (float)H_bar);
  return {daq, dlp};
}

static torch::Tensor squashed_gaussian_bwd2(
    torch::Tensor ga, torch::Tensor gl, torch::Tensor lsr, torch::Tensor ls,
    torch::Tensor eps, torch::Tensor tanh_u, double k) {
  CHECK_IN(ga);
  auto gac = ga.contiguous(); auto glc = gl.contiguous();
  const bool lsr_ok = lsr.dim() == 2 && lsr.stride(1) == 1;
  auto lsrc = lsr_ok ? lsr : lsr.contiguous();
  auto lsc = ls.contiguous();
  auto ec = eps.contiguous(); auto tc = tanh_u.contiguous();
  const int ga_twin = (gac.dim() == 3 && gac.size(0) == 2) ? 1 : 0;
  const long B = ga_twin ? gac.size(1) : gac.size(0);
  const long A = ga_twin ? gac.size(2) : gac.size(1);
  const long lsr_ld = lsr_ok ? lsr.stride(0) : A;
  auto dhead = torch::empty({B, 2 * A},
                            gac.options().dtype(torch::kBFloat16));
  hipLaunchKernelGGL(k_squash_bwd2, dim3((B + 255) / 256), dim3(256), 0,
                     cur_stream(), gac.data_ptr<float>(),
                     glc.data_ptr<float>(), lsrc.data_ptr<float>(),
                     lsc.data_ptr<float>(), ec.data_ptr<float>(),
                     tc.data_ptr<float>(),
                     (unsigned short*)dhead.data_ptr(), lsr_ld, ga_twin,
                     (int)B, (int)A, (float)k);
  return dhead;
}

static void adam_step_(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                       torch::Tensor v, long step, double lr, double b1,
                       double b2, double eps) {
  CHECK_IN(p);
  const long n = p.numel();
  const double bc1 = 1.0 - std::pow(b1, (double)step);
  const double bc2 = 1.0 - std::pow(b2, (double)step);
  const float step_size = (float)(lr / bc1);
  const float inv_sqrt_bc2 = (float)(1.0 / std::sqrt(bc2));
  hipLaunchKernelGGL(k_adam, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n,
                     (float)b1, (float)b2, step_size, inv_sqrt_bc2,
                     (float)eps);
}

static void adam_step_dev_(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                           torch::Tensor v, torch::Tensor state, double lr,
                           double b1, double b2, double eps,
                           c10::optional<torch::Tensor> mir_opt,
                           long skip_prolog = 0) {
  CHECK_IN(p); CHECK_IN(state);
  TORCH_CHECK(state.numel() >= 3, "state = {step, step_size, inv_sqrt_bc2}");
  const long n = p.numel();
  unsigned short* mp = nullptr;
  if (mir_opt.has_value() && mir_opt->defined() && mir_opt->numel() > 0) {
    TORCH_CHECK(mir_opt->numel() == n
                && mir_opt->scalar_type() == torch::kBFloat16);
    mp = (unsigned short*)mir_opt->data_ptr();
  }
  if (!skip_prolog)
    hipLaunchKernelGGL(k_adam_prolog, dim3(1), dim3(1), 0, cur_stream(),
                       state.data_ptr<float>(), (float)lr, (float)b1,
                       (float)b2);
  hipLaunchKernelGGL(k_adam_dev, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     state.data_ptr<float>(), n, (float)b1, (float)b2,
                     (float)eps, mp);
}

static void adam_prolog_many(std::vector<torch::Tensor> states,
                             std::vector<double> lrs, double b1, double b2,
                             c10::optional<torch::Tensor> rng) {
  const size_t G = states.size();
  TORCH_CHECK(G >= 1 && G <= 4 && lrs.size() == G, "1..4 states");
  float* S[4] = {nullptr, nullptr, nullptr, nullptr};
  double L[4] = {0, 0, 0, 0};
  for (size_t g = 0; g < G; ++g) {
    CHECK_IN(states[g]);
    S[g] = states[g].data_ptr<float>();
    L[g] = lrs[g];
  }
  long long* rng_p = (rng.has_value() && rng->numel() > 0)
      ? (long long*)rng->data_ptr<long>() : nullptr;
  hipLaunchKernelGGL(k_adam_prolog_many, dim3(1), dim3(4), 0, cur_stream(),
                     rng_p, S[0], S[1], S[2], S[3], (float)L[0],
                     (float)L[1], (float)L[2], (float)L[3], (float)b1,
                     (float)b2);
}

static void adam_step_multi_(std::vector<torch::Tensor> ps,
                             std::vector<torch::Tensor> gs,
                             std::vector<torch::Tensor> ms,
                             std::vector<torch::Tensor> vs,
                             std::vector<torch::Tensor> states,
                             std::vector<double> lrs, double b1, double b2,
                             double eps, std::vector<torch::Tensor> mirs,
                             c10::optional<torch::Tensor> rng = c10::nullopt,
                             long skip_prolog = 0) {
  const size_t G = ps.size();
  TORCH_CHECK(G >= 1 && G <= 3, "1..3 groups");
  float* P[3] = {nullptr, nullptr, nullptr};
  const float* Gr[3] = {nullptr, nullptr, nullptr};
  float* M[3] = {nullptr, nullptr, nullptr};
  float* V[3] = {nullptr, nullptr, nullptr};
  float* St[3] = {nullptr, nullptr, nullptr};
  long N[3] = {0, 0, 0};
  double LR[3] = {0, 0, 0};
  unsigned short* MR[3] = {nullptr, nullptr, nullptr};
  long total = 0;
  for (size_t g = 0; g < G; ++g) {
    CHECK_IN(ps[g]);
    P[g] = ps[g].data_ptr<float>();
    Gr[g] = gs[g].data_ptr<float>();
    M[g] = ms[g].data_ptr<float>();
    V[g] = vs[g].data_ptr<float>();
    St[g] = states[g].data_ptr<float>();
    N[g] = ps[g].numel();
    LR[g] = lrs[g];
    if (g < mirs.size() && mirs[g].defined() && mirs[g].numel() > 0) {
      TORCH_CHECK(mirs[g].numel() == N[g]
                  && mirs[g].scalar_type() == torch::kBFloat16);
      MR[g] = (unsigned short*)mirs[g].data_ptr();
    }
    total += N[g];
  }
  long long* rng_p = (rng.has_value() && rng->numel() > 0)
      ? (long long*)rng->data_ptr<long>() : nullptr;
  if (!skip_prolog)
    hipLaunchKernelGGL(k_adam_prolog3, dim3(1), dim3(3), 0, cur_stream(),
                       rng_p, St[0], St[1], St[2], (float)LR[0],
                       (float)LR[1], (float)LR[2], (float)b1, (float)b2);
  hipLaunchKernelGGL(k_adam_multi, dim3((total + 255) / 256), dim3(256), 0,
                     cur_stream(),
                     P[0], Gr[0], M[0], V[0], St[0], N[0],
                     P[1], Gr[1], M[1], V[1], St[1], N[1],
                     P[2], Gr[2], M[2], V[2], St[2], N[2],
                     (float)b1, (float)b2, (float)eps, MR[0], MR[1], MR[2]);
}

static void polyak_(torch::Tensor t, torch::Tensor s, double tau,
                    c10::optional<torch::Tensor> mir_opt) {
  CHECK_IN(t);
  const long n = t.numel();
  TORCH_CHECK(s.numel() == n, "polyak buffers must match");
  unsigned short* mp = nullptr;
  if (mir_opt.has_value() && mir_opt->defined() && mir_opt->numel() > 0) {
    TORCH_CHECK(mir_opt->numel() == n
                && mir_opt->scalar_type() == torch::kBFloat16);
    mp = (unsigned short*)mir_opt->data_ptr();
  }
  hipLaunchKernelGGL(k_polyak, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), t.data_ptr<float>(), s.data_ptr<float>(),
                     n, (float)tau, mp);
}

void register_shm_ring(pybind11::module_& m);
void register_bf16(pybind11::module_& m);
void register_chain(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  register_shm_ring(mod);
  register_bf16(mod);
  register_chain(mod);
  mod.def("linear_act_fwd", &linear_act_fwd, "fused GEMM+bias+act forward");
  mod.def("linear_act_fwd_g", &linear_act_fwd_g, "grouped (twin) variant");
  mod.def("linear_bwd_dx", &linear_bwd_dx, "GEMM backward dX (fused mask)");
  mod.def("linear_bwd_dx_g", &linear_bwd_dx_g, "grouped, sums over G");
  mod.def("linear_bwd_dwdb", &linear_bwd_dwdb, "split-K backward dW+db");
  mod.def("linear_bwd_dwdb_g", &linear_bwd_dwdb_g, "grouped split-K dW+db");
  mod.def("replay_sample", &replay_sample, "stratified replay gather",
          pybind11::arg("states"), pybind11::arg("actions"),
          pybind11::arg("rewards"), pybind11::arg("next_states"),
          pybind11::arg("dones"), pybind11::arg("sizes"),
          pybind11::arg("rnd"), pybind11::arg("B"),
          pybind11::arg("rng") = pybind11::none());
  mod.def("squashed_gaussian_fwd", &squashed_gaussian_fwd,
          pybind11::arg("mu"), pybind11::arg("lsr"),
          pybind11::arg("eps"), pybind11::arg("k"),
          pybind11::arg("rng") = pybind11::none());
  mod.def("squashed_gaussian_bwd", &squashed_gaussian_bwd);
  mod.def("td_target", &td_target);
  mod.def("td_target_mt", &td_target_mt);
  mod.def("critic_loss_fwd", &critic_loss_fwd,
          pybind11::arg("q1"), pybind11::arg("q2"),
          pybind11::arg("y"), pybind11::arg("states"),
          pybind11::arg("log_alpha"), pybind11::arg("T"),
          pybind11::arg("use_w"),
          pybind11::arg("ws") = pybind11::none());
  mod.def("critic_loss_bwd", &critic_loss_bwd);
  mod.def("actor_alpha_loss_fwd", &actor_alpha_loss_fwd,
          pybind11::arg("aq1"), pybind11::arg("aq2"),
          pybind11::arg("lp"), pybind11::arg("ls"),
          pybind11::arg("states"), pybind11::arg("log_alpha"),
          pybind11::arg("T"), pybind11::arg("use_w"),
          pybind11::arg("H_bar"),
          pybind11::arg("dla") = pybind11::none(),
          pybind11::arg("ws") = pybind11::none());
  mod.def("actor_alpha_loss_bwd", &actor_alpha_loss_bwd);
  mod.def("critic_loss_bwd2", &critic_loss_bwd2);
  mod.def("actor_alpha_loss_bwd2", &actor_alpha_loss_bwd2);
  mod.def("squashed_gaussian_bwd2", &squashed_gaussian_bwd2);
  mod.def("adam_step_", &adam_step_);
  mod.def("adam_step_dev_", &adam_step_dev_,
          pybind11::arg("p"), pybind11::arg("g"), pybind11::arg("m"),
          pybind11::arg("v"), pybind11::arg("state"), pybind11::arg("lr"),
          pybind11::arg("b1"), pybind11::arg("b2"), pybind11::arg("eps"),
          pybind11::arg("mir") = pybind11::none(),
          pybind11::arg("skip_prolog") = 0);
  mod.def("adam_step_multi_", &adam_step_multi_,
          pybind11::arg("ps"), pybind11::arg("gs"), pybind11::arg("ms"),
          pybind11::arg("vs"), pybind11::arg("states"), pybind11::arg("lrs"),
          pybind11::arg("b1"), pybind11::arg("b2"), pybind11::arg("eps"),
          pybind11::arg("mirs") = std::vector<torch::Tensor>(),
          pybind11::arg("rng") = pybind11::none(),
          pybind11::arg("skip_prolog") = 0);
  mod.def("adam_prolog_many", &adam_prolog_many,
          pybind11::arg("states"), pybind11::arg("lrs"),
          pybind11::arg("b1"), pybind11::arg("b2"),
          pybind11::arg("rng") = pybind11::none());
  mod.def("polyak_", &polyak_,
          pybind11::arg("t"), pybind11::arg("s"), pybind11::arg("tau"),
          pybind11::arg("mir") = pybind11::none());
}
