// bf16_gemm.hip — bf16 GEMM family for the SAC hot path (gfx950).
//
// Mixed precision: fp32 master weights (Adam) with bf16 mirrors; GEMMs run
// on v_mfma_f32_16x16x32_bf16 (K=32 per instruction — 8x the fp32 MFMA's
// K=4 at similar issue cost) with fp32 accumulation; weight grads (dW/db)
// accumulate and store fp32.  Halves LDS/HBM traffic vs the fp32 path.
//
// Fragment maps (verified by the identity-A/asymmetric-B GPU test):
//   A: lane l -> row i = l&15,  k = (l>>4)*8 .. +8   (8 bf16 = 4 VGPRs)
//   B: lane l -> col j = l&15,  k = (l>>4)*8 .. +8
//   C/D: col = l&15, row = (l>>4)*4 + reg (4 fp32) — same as fp32 MFMA.
//
// LDS images are arranged so every fragment read is ONE 16-byte short8
// read of 8 consecutive bf16: operands whose fragment k-run crosses the
// global minor axis are stored transposed (scatter ds_write, like the
// fp32 dW kernel's dy^T tile).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <vector>

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using u16 = unsigned short;

static constexpr int TBM = 64;   // rows per tile
static constexpr int TBN = 64;   // cols per tile
static constexpr int TBK = 64;   // reduction per tile (2 mfma k-steps)
// bf16 row pad: +8 elements (16B) keeps the 16-lane b128 groups spread
static constexpr int TPAD = TBK + 8;   // 72 bf16 = 144 B rows

#define CHECK_BF16(t) TORCH_CHECK((t).is_cuda() && (t).scalar_type() == torch::kBFloat16, \
                                  #t " must be a bf16 HIP tensor")
#define CHECK_F32(t) TORCH_CHECK((t).is_cuda() && (t).scalar_type() == torch::kFloat32, \
                                 #t " must be a fp32 HIP tensor")

static inline hipStream_t cur_stream2() {
  return c10::hip::getCurrentHIPStream().stream();
}

__device__ __forceinline__ u16 f32_to_bf16_rne(float f) {
  union { float f; unsigned u; } v{f};
  unsigned u = v.u;
  u += 0x7FFFu + ((u >> 16) & 1u);   // round-to-nearest-even
  return (u16)(u >> 16);
}

__global__ __launch_bounds__(256) void k_cast_f32_bf16(
    const float* __restrict__ src, u16* __restrict__ dst, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dst[i] = f32_to_bf16_rne(src[i]);
}

__global__ __launch_bounds__(256) void k_cast_bf16_f32(
    const u16* __restrict__ src, float* __restrict__ dst, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  union { float f; unsigned u; } v;
  v.u = ((unsigned)src[i]) << 16;
  dst[i] = v.f;
}

// ---------------------------------------------------------------------------
// fwd: y[g,M,N] = act(x[M,K]_bf16 @ w[g,N,K]_bf16^T + b[g,N]_f32)
// Both LDS tiles are k-minor (row-major [row][k]) — fragment k-runs are
// contiguous for A (x rows) and B (w rows).  grid (M/64, N/64, G).
// out_f32: 1 -> write fp32 y; 0 -> write bf16 y.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_bf16_fwd(
    const u16* __restrict__ x, const u16* __restrict__ w,
    const float* __restrict__ b, void* __restrict__ y,
    int M, int N, int K, int act, long xgs, int out_f32) {
  __shared__ u16 sx[2][TBM][TPAD];
  __shared__ u16 sw[2][TBN][TPAD];
  const long g = blockIdx.z;
  x += g * xgs;
  w += g * (long)N * K;
  b += g * (long)N;
  const int m0 = blockIdx.x * TBM, n0 = blockIdx.y * TBN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  // loader: 64x64 u16 tile = 4096 elems / 256 thr = 16 per thread.
  // Fast path: two 16-byte vector loads + stores when the 16-element run
  // is fully in range and 16B-aligned (K % 8 == 0 rows) — the scalar
  // fallback was the dominant stall in the first bf16 cut (PMC: 80%
  // SQ_WAIT_ANY; profiles/r05_NOTES.md).
  const int lr = tid >> 2, lc0 = (tid & 3) * 16;
  const bool krows_aligned = (K % 8) == 0;
#define LOAD16(dst_row, src, rowlim, gk0)                                   \
  {                                                                         \
    const int r_ = lr;                                                      \
    if (krows_aligned && r_ < (rowlim) && (gk0) + lc0 + 16 <= K) {          \
      const uint4* sp = (const uint4*)&src[(long)r_ * K + (gk0) + lc0];     \
      uint4* dp = (uint4*)&dst_row[lc0];                                    \
      dp[0] = sp[0];                                                        \
      dp[1] = sp[1];                                                        \
    } else {                                                                \
      _Pragma("unroll") for (int j = 0; j < 16; ++j) {                      \
        const int gk = (gk0) + lc0 + j;                                     \
        dst_row[lc0 + j] = (r_ < (rowlim) && gk < K)                        \
            ? src[(long)r_ * K + gk] : (u16)0;                              \
      }                                                                     \
    }                                                                       \
  }

  for (int k0 = 0, buf = 0; k0 < K; k0 += TBK, buf ^= 1) {
    {
      const u16* xs = x + (long)m0 * K;
      const u16* wsrc = w + (long)n0 * K;
      LOAD16(sx[buf][lr], xs, (M - m0 < TBM ? M - m0 : TBM), k0)
      LOAD16(sw[buf][lr], wsrc, (N - n0 < TBN ? N - n0 : TBN), k0)
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < TBK; kk += 32) {
      const bf16x8 a0 = *(const bf16x8*)&sx[buf][wr + fi][kk + fk * 8];
      const bf16x8 a1 = *(const bf16x8*)&sx[buf][wr + 16 + fi][kk + fk * 8];
      const bf16x8 b0 = *(const bf16x8*)&sw[buf][wc + fi][kk + fk * 8];
      const bf16x8 b1 = *(const bf16x8*)&sw[buf][wc + 16 + fi][kk + fk * 8];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }
  const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
  float* yf = (float*)y + g * (long)M * N;
  u16* yh = (u16*)y + g * (long)M * N;
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
          if (out_f32) yf[(long)row * N + col] = v;
          else yh[(long)row * N + col] = f32_to_bf16_rne(v);
        }
      }
    }
}

// ---------------------------------------------------------------------------
// dx: dx[M,K] = sum_g (dy*mask)[g,M,N] @ w[g,N,K]
// A = dy (m-major, n-minor: k-run over n contiguous) — natural layout.
// B = w: fragment k-run is over n at fixed k column -> store w TRANSPOSED
// [k][n] (scatter write).  grid (M/64, K/64, Gz); per-group via grid.z
// (sum_over_g=0) or summed inner loop (=1).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_bf16_dx(
    const u16* __restrict__ dy, const u16* __restrict__ w,
    const u16* __restrict__ yout, u16* __restrict__ dx,
    int M, int N, int K, int act, int G) {
  __shared__ u16 sdy[2][TBM][TPAD];     // [m][n-run]
  __shared__ u16 swT[2][TBN][TPAD];     // [k][n-run]  (transposed w)
  const long z = blockIdx.z;
  dy += z * (long)M * N;
  w += z * (long)N * K;
  yout += z * (long)M * N;
  dx += z * (long)M * K;
  const int m0 = blockIdx.x * TBM, c0 = blockIdx.y * TBN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  const int lr = tid >> 2, lc0 = (tid & 3) * 16;
  for (int g = 0; g < G; ++g) {
    const u16* dyg = dy + (long)g * M * N;
    const u16* wg = w + (long)g * N * K;
    const u16* yg = yout + (long)g * M * N;
    for (int n0 = 0, buf = 0; n0 < N; n0 += TBK, buf ^= 1) {
      // dy tile [64 m][64 n] with fused relu mask
      {
        const int gm = m0 + lr;
        if ((N % 8) == 0 && gm < M && n0 + lc0 + 16 <= N) {
          const uint4* dp_ = (const uint4*)&dyg[(long)gm * N + n0 + lc0];
          uint4 v0 = dp_[0], v1 = dp_[1];
          if (act == 1) {
            const uint4* yp = (const uint4*)&yg[(long)gm * N + n0 + lc0];
            uint4 m0v = yp[0], m1v = yp[1];
            // zero dy lanes where the (bf16) activation is exactly 0
            const u16* ym0 = (const u16*)&m0v;
            const u16* ym1 = (const u16*)&m1v;
            u16* d0 = (u16*)&v0;
            u16* d1 = (u16*)&v1;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              if (ym0[j] == 0) d0[j] = 0;
              if (ym1[j] == 0) d1[j] = 0;
            }
          }
          ((uint4*)&sdy[buf][lr][lc0])[0] = v0;
          ((uint4*)&sdy[buf][lr][lc0])[1] = v1;
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) {
            const int gn = n0 + lc0 + j;
            u16 v = 0;
            if (gm < M && gn < N) {
              v = dyg[(long)gm * N + gn];
              if (act == 1 && yg[(long)gm * N + gn] == 0) v = 0;
            }
            sdy[buf][lr][lc0 + j] = v;
          }
        }
      }
      // w tile transposed: read w[n0+lr][c0+lc0+j] -> swT[lc0+j][lr]
      {
        const int gn = n0 + lr;
        if ((K % 8) == 0 && gn < N && c0 + lc0 + 16 <= K) {
          const uint4* sp = (const uint4*)&wg[(long)gn * K + c0 + lc0];
          uint4 v0 = sp[0], v1 = sp[1];
          const u16* e0 = (const u16*)&v0;
          const u16* e1 = (const u16*)&v1;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            swT[buf][lc0 + j][lr] = e0[j];
            swT[buf][lc0 + 8 + j][lr] = e1[j];
          }
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) {
            const int gk = c0 + lc0 + j;
            swT[buf][lc0 + j][lr] =
                (gn < N && gk < K) ? wg[(long)gn * K + gk] : (u16)0;
          }
        }
      }
      __syncthreads();
#pragma unroll
      for (int kk = 0; kk < TBK; kk += 32) {
        const bf16x8 a0 = *(const bf16x8*)&sdy[buf][wr + fi][kk + fk * 8];
        const bf16x8 a1 = *(const bf16x8*)&sdy[buf][wr + 16 + fi][kk + fk * 8];
        const bf16x8 b0 = *(const bf16x8*)&swT[buf][wc + fi][kk + fk * 8];
        const bf16x8 b1 = *(const bf16x8*)&swT[buf][wc + 16 + fi][kk + fk * 8];
        acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
        acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
        acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
      }
      __syncthreads();
    }
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
        if (row < M && col < K) dx[(long)row * K + col] = f32_to_bf16_rne(a[r]);
      }
    }
}

// ---------------------------------------------------------------------------
// dwdb: dw[g,N,K]_f32 = (dy*mask)^T @ x; db fused; split-K over the batch
// into fp32 partials (reuses the fp32 k_reduce_partials for the fold).
// A = dy^T: [n][m-run] (transposed store); B = x: fragment k-run over m at
// fixed k -> x transposed [k][m-run].  grid (N/64, K/64, G*S).
// ---------------------------------------------------------------------------
__device__ __forceinline__ void dwdb_body(
    u16 (*sa)[TBN][TPAD], u16 (*sbT)[TBN][TPAD],
    const u16* __restrict__ dyg, const u16* __restrict__ xg,
    const u16* __restrict__ yg, float* __restrict__ wsp,
    float* __restrict__ dbp, int N, int K, int act, int m_lo, int m_hi,
    int n0, int c0, bool do_db, int transpose_w) {
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  float db_acc = 0.f;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  const int lr = tid >> 2, lc0 = (tid & 3) * 16;
  for (int m0 = m_lo, buf = 0; m0 < m_hi; m0 += TBK, buf ^= 1) {
    {
      const int gm = m0 + lr;   // this thread's batch row
      // dy block [64 m][64 n] -> sa[n][m] (scatter), masked
      if ((N % 8) == 0 && gm < m_hi && n0 + lc0 + 16 <= N) {
        const uint4* dp_ = (const uint4*)&dyg[(long)gm * N + n0 + lc0];
        uint4 v0 = dp_[0], v1 = dp_[1];
        u16* d0 = (u16*)&v0;
        u16* d1 = (u16*)&v1;
        if (act == 1) {
          const uint4* yp = (const uint4*)&yg[(long)gm * N + n0 + lc0];
          uint4 y0 = yp[0], y1 = yp[1];
          const u16* m0_ = (const u16*)&y0;
          const u16* m1_ = (const u16*)&y1;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (m0_[j] == 0) d0[j] = 0;
            if (m1_[j] == 0) d1[j] = 0;
          }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          sa[buf][lc0 + j][lr] = d0[j];
          sa[buf][lc0 + 8 + j][lr] = d1[j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int gn = n0 + lc0 + j;
          u16 v = 0;
          if (gm < m_hi && gn < N) {
            v = dyg[(long)gm * N + gn];
            if (act == 1 && yg[(long)gm * N + gn] == 0) v = 0;
          }
          sa[buf][lc0 + j][lr] = v;
        }
      }
      // x block [64 m][64 k] -> sbT[k][m] (scatter)
      if ((K % 8) == 0 && gm < m_hi && c0 + lc0 + 16 <= K) {
        const uint4* sp = (const uint4*)&xg[(long)gm * K + c0 + lc0];
        uint4 v0 = sp[0], v1 = sp[1];
        const u16* e0 = (const u16*)&v0;
        const u16* e1 = (const u16*)&v1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          sbT[buf][lc0 + j][lr] = e0[j];
          sbT[buf][lc0 + 8 + j][lr] = e1[j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int gk = c0 + lc0 + j;
          sbT[buf][lc0 + j][lr] =
              (gm < m_hi && gk < K) ? xg[(long)gm * K + gk] : (u16)0;
        }
      }
    }
    __syncthreads();
    if (do_db && tid < TBN) {
#pragma unroll
      for (int m = 0; m < TBK; ++m) {
        union { float f; unsigned u; } v;
        v.u = ((unsigned)sa[buf][tid][m]) << 16;
        db_acc += v.f;
      }
    }
#pragma unroll
    for (int kk = 0; kk < TBK; kk += 32) {
      const bf16x8 a0 = *(const bf16x8*)&sa[buf][wr + fi][kk + fk * 8];
      const bf16x8 a1 = *(const bf16x8*)&sa[buf][wr + 16 + fi][kk + fk * 8];
      const bf16x8 b0 = *(const bf16x8*)&sbT[buf][wc + fi][kk + fk * 8];
      const bf16x8 b1 = *(const bf16x8*)&sbT[buf][wc + 16 + fi][kk + fk * 8];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }
  if (do_db && tid < TBN && n0 + tid < N) dbp[n0 + tid] = db_acc;
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
        // transpose_w: destination is the (G,K,N) master layout (the
        // mixture Linear holds W as (k,in,out)) — store dW^T directly so
        // no permute kernel is needed on the manual path.
        if (row < N && col < K)
          wsp[transpose_w ? (long)col * N + row : (long)row * K + col] = a[r];
      }
    }
}

__global__ __launch_bounds__(256) void k_bf16_dwdb_splitk(
    const u16* __restrict__ dy, const u16* __restrict__ x,
    const u16* __restrict__ yout, float* __restrict__ ws,
    float* __restrict__ ws_db, int M, int N, int K, int act, int S,
    int chunk, long xgs, int transpose_w, long s_stride, long w_off,
    long b_off) {
  __shared__ u16 sa[2][TBN][TPAD];    // dy^T: [n][m-run]
  __shared__ u16 sbT[2][TBN][TPAD];   // x^T:  [k][m-run]
  const int gs = blockIdx.z;
  const int g = gs / S, s = gs % S;
  const u16* dyg = dy + (long)g * M * N;
  const u16* yg = yout + (long)g * M * N;
  const u16* xg = x + (long)g * xgs;
  // s_stride>0: phase-arena mode — partials for split s of EVERY layer
  // land in one [S, group_numel] arena laid out in flat-gradient order
  // (w_off/b_off = the layer's offsets in the flat buffer), so ONE
  // reduce per phase folds the whole gradient.
  float* wsp = s_stride > 0
      ? ws + (long)s * s_stride + w_off + (long)g * N * K
      : ws + (long)gs * N * K;
  float* dbp = s_stride > 0
      ? ws + (long)s * s_stride + b_off + (long)g * N
      : ws_db + (long)gs * N;
  dwdb_body(sa, sbT, dyg, xg, yg, wsp, dbp, N, K, act, s * chunk,
            min(M, s * chunk + chunk), blockIdx.x * TBM,
            blockIdx.y * TBN, blockIdx.y == 0, transpose_w);
}

// ---------------------------------------------------------------------------
// Grouped dwdb: ONE launch computes the dW/db split-K partials of up to
// six layers (a whole chain's weight gradients) straight into the phase
// arena.  The per-layer dy inputs are the PRE-MASKED dys the fused
// dx-chain kernel saved (act=0 here).  grid (sum_l nbx_l*nby_l, 1, G*S).
// ---------------------------------------------------------------------------
struct DwGroupDesc {
  const u16* dy[6];
  const u16* x[6];
  long xgs[6];
  long w_off[6], b_off[6];
  int N[6], K[6], nbx[6];
  int cum[7];
  int L, S, chunk, M;
};

__global__ __launch_bounds__(256) void k_bf16_dwdb_grouped(
    DwGroupDesc d, float* __restrict__ arena, long s_stride) {
  __shared__ u16 sa[2][TBN][TPAD];
  __shared__ u16 sbT[2][TBN][TPAD];
  const int bx = blockIdx.x;
  int l = 0;
  while (l + 1 < d.L && bx >= d.cum[l + 1]) ++l;
  const int local = bx - d.cum[l];
  const int n0 = (local % d.nbx[l]) * TBM;
  const int c0 = (local / d.nbx[l]) * TBN;
  const int gs = blockIdx.z;
  const int g = gs / d.S, s = gs % d.S;
  const int N = d.N[l], K = d.K[l];
  const u16* dyg = d.dy[l] + (long)g * d.M * N;
  const u16* xg = d.x[l] + (long)g * d.xgs[l];
  float* wsp = arena + (long)s * s_stride + d.w_off[l] + (long)g * N * K;
  float* dbp = arena + (long)s * s_stride + d.b_off[l] + (long)g * N;
  dwdb_body(sa, sbT, dyg, xg, dyg, wsp, dbp, N, K, /*act=*/0,
            s * d.chunk, min(d.M, s * d.chunk + d.chunk), n0, c0,
            c0 == 0, /*transpose_w=*/0);
}

// ---------------------------------------------------------------------------
// Fused narrow MLP chain (CARE state-encoder trunks): every layer is at
// most 64 wide, so ONE workgroup can push a 64-row tile of the batch
// through the WHOLE chain, staging each layer's activation in LDS — one
// kernel launch instead of one per layer.  Layer 0 streams x/W0 through
// the usual double-buffered k-loop (K0 arbitrary, e.g. the 768-d RoBERTa
// embedding); layers >=1 read the staged activation (k = prev width <= 64,
// a single MFMA k-tile) against an LDS-resident weight.  When `acts` slots
// are set, intermediate activations are also written to global memory for
// the manual backward (same compact [G,M,N] layout the per-layer kernels
// produce).  grid (ceil(M/64), 1, G).
// ---------------------------------------------------------------------------
struct NarrowDesc {
  const u16* w[6];
  const float* b[6];
  u16* acts[6];       // saved post-act outputs for layers 0..L-2 (or null)
  int N[6];
  int L;
};

__device__ __forceinline__ void narrow_body(
    u16 (*sx)[TBM][TPAD], u16 (*sw)[TBM][TPAD], u16 (*sy)[TPAD],
    const u16* __restrict__ x, const NarrowDesc& d, void* __restrict__ y,
    int M, int K0, long xgs, int act_last, int out_f32, int m0, long g) {
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  const int lr = tid >> 2, lc0 = (tid & 3) * 16;
  const u16* xg = x + g * xgs;
  const int rowlim = (M - m0 < TBM ? M - m0 : TBM);

  f32x4 acc00{}, acc01{}, acc10{}, acc11{};
  {  // ---- layer 0 ----
    const u16* w0 = d.w[0] + g * (long)d.N[0] * K0;
    const bool ka = (K0 % 8) == 0;
    for (int k0 = 0, buf = 0; k0 < K0; k0 += TBK, buf ^= 1) {
      if (ka && lr < rowlim && k0 + lc0 + 16 <= K0) {
        const uint4* sp = (const uint4*)&xg[((long)(m0 + lr)) * K0 + k0 + lc0];
        uint4* dp = (uint4*)&sx[buf][lr][lc0];
        dp[0] = sp[0]; dp[1] = sp[1];
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int gk = k0 + lc0 + j;
          sx[buf][lr][lc0 + j] = (lr < rowlim && gk < K0)
              ? xg[((long)(m0 + lr)) * K0 + gk] : (u16)0;
        }
      }
      if (ka && lr < d.N[0] && k0 + lc0 + 16 <= K0) {
        const uint4* sp = (const uint4*)&w0[(long)lr * K0 + k0 + lc0];
        uint4* dp = (uint4*)&sw[buf][lr][lc0];
        dp[0] = sp[0]; dp[1] = sp[1];
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int gk = k0 + lc0 + j;
          sw[buf][lr][lc0 + j] = (lr < d.N[0] && gk < K0)
              ? w0[(long)lr * K0 + gk] : (u16)0;
        }
      }
      __syncthreads();
#pragma unroll
      for (int kk = 0; kk < TBK; kk += 32) {
        const bf16x8 a0 = *(const bf16x8*)&sx[buf][wr + fi][kk + fk * 8];
        const bf16x8 a1 = *(const bf16x8*)&sx[buf][wr + 16 + fi][kk + fk * 8];
        const bf16x8 b0 = *(const bf16x8*)&sw[buf][wc + fi][kk + fk * 8];
        const bf16x8 b1 = *(const bf16x8*)&sw[buf][wc + 16 + fi][kk + fk * 8];
        acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
        acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
        acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
      }
      __syncthreads();
    }
  }

  for (int li = 0;; ++li) {
    const bool last = li == d.L - 1;
    const int N = d.N[li];
    const float* bb = d.b[li] + g * (long)N;
    const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
    if (last) {
      float* yf = (float*)y + g * (long)M * N;
      u16* yh = (u16*)y + g * (long)M * N;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const f32x4 a = *accs[mi][ni];
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = m0 + wr + mi * 16 + fk * 4 + r;
            const int col = wc + ni * 16 + fi;
            if (row < M && col < N) {
              float v = a[r] + bb[col];
              if (act_last == 1) v = fmaxf(v, 0.f);
              if (out_f32) yf[(long)row * N + col] = v;
              else yh[(long)row * N + col] = f32_to_bf16_rne(v);
            }
          }
        }
      return;
    }
    __syncthreads();   // sy free from previous layer's reads
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const f32x4 a = *accs[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = wr + mi * 16 + fk * 4 + r;
          const int col = wc + ni * 16 + fi;
          float v = col < N ? fmaxf(a[r] + bb[col], 0.f) : 0.f;
          sy[row][col] = f32_to_bf16_rne(v);
        }
      }
    __syncthreads();
    if (d.acts[li] != nullptr) {
      u16* ag = d.acts[li] + g * (long)M * N;
      if (lr < rowlim) {
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const int col = lc0 + j;
          if (col < N) ag[((long)(m0 + lr)) * N + col] = sy[lr][col];
        }
      }
    }
    // next layer's weight tile: [Nn, K=N<=64] into sw[0]
    const int Nn = d.N[li + 1];
    const u16* wn = d.w[li + 1] + g * (long)Nn * N;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int gk = lc0 + j;
      sw[0][lr][lc0 + j] = (lr < Nn && gk < N) ? wn[(long)lr * N + gk] : (u16)0;
    }
    __syncthreads();
    acc00 = f32x4{}; acc01 = f32x4{}; acc10 = f32x4{}; acc11 = f32x4{};
#pragma unroll
    for (int kk = 0; kk < TBK; kk += 32) {
      const bf16x8 a0 = *(const bf16x8*)&sy[wr + fi][kk + fk * 8];
      const bf16x8 a1 = *(const bf16x8*)&sy[wr + 16 + fi][kk + fk * 8];
      const bf16x8 b0 = *(const bf16x8*)&sw[0][wc + fi][kk + fk * 8];
      const bf16x8 b1 = *(const bf16x8*)&sw[0][wc + 16 + fi][kk + fk * 8];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void k_bf16_mlp_narrow(
    const u16* __restrict__ x, NarrowDesc d, void* __restrict__ y,
    int M, int K0, long xgs, int act_last, int out_f32) {
  __shared__ u16 sx[2][TBM][TPAD];
  __shared__ u16 sw[2][TBM][TPAD];
  __shared__ u16 sy[TBM][TPAD];
  narrow_body(sx, sw, sy, x, d, y, M, K0, xgs, act_last, out_f32,
              blockIdx.x * TBM, blockIdx.z);
}

// ---------------------------------------------------------------------------
// Multi-chain narrow forward (round 2): ONE launch runs up to 4
// independent narrow chains (CARE state-encoder = mixture G=k + trunk +
// mlp_context per forward) — block ranges partition the grid by chain.
// Each per-chain block count = ceil(M_c/64) * G_c.
// ---------------------------------------------------------------------------
struct NarrowMultiDesc {
  NarrowDesc d[4];
  const u16* x[4];
  void* y[4];
  int M[4], K0[4], act_last[4], out_f32[4], nbm[4];
  long xgs[4];
  int cum[5];
  int C;
};

__global__ __launch_bounds__(256) void k_bf16_mlp_narrow_multi(
    NarrowMultiDesc nm) {
  __shared__ u16 sx[2][TBM][TPAD];
  __shared__ u16 sw[2][TBM][TPAD];
  __shared__ u16 sy[TBM][TPAD];
  const int bx = blockIdx.x;
  int c = 0;
  while (c + 1 < nm.C && bx >= nm.cum[c + 1]) ++c;
  const int local = bx - nm.cum[c];
  const int g = local / nm.nbm[c];
  const int m0 = (local % nm.nbm[c]) * TBM;
  narrow_body(sx, sw, sy, nm.x[c], nm.d[c], nm.y[c], nm.M[c], nm.K0[c],
              nm.xgs[c], nm.act_last[c], nm.out_f32[c], m0, g);
}

// ---------------------------------------------------------------------------
// Fused narrow MLP chain BACKWARD (round-2 staging — compiled and bound
// but NOT called by default: no GPU budget remained this round to
// validate it; enable via CAREEngine env gate DSAC_NARROW_BWD=1 after
// the bitwise check against the per-layer path).
//
// One workgroup owns 64 batch rows of group g and walks the whole chain
// backward: per layer, (a) mask dy by the saved post-ReLU activation,
// (b) dW/db partial = dy^T @ a_i via MFMA into the caller's arena at
// arena[s=blockIdx.x][w_off_i] (flat-gradient layout, one reduce for the
// whole chain set), (c) dy_{i-1} = (dy_i @ W_i) * relu'(a_i).  The chain
// input is detached in every use (reference gradient-flow rules), so no
// dx0 is produced.  All layers <= 64 wide; L <= 6.
// grid (ceil(M/64), 1, G); arena S = gridDim.x.
// ---------------------------------------------------------------------------
struct NarrowBwdDesc {
  const u16* w[6];      // kernel-layout weights [G,N_i,K_i] (bf16 mirrors)
  const u16* acts[6];   // acts[0]=chain input x; acts[i]=post-act out of i-1
  long a_gs[6];         // per-group stride of acts[i] (0 = shared across
                        // groups, e.g. the mixture's raw-state input)
  long w_off[6];        // flat-gradient offsets for dW_i
  long b_off[6];        //   "                 "     db_i
  int N[6];             // layer widths; K_i = N_{i-1} (K_0 = Kin)
  int L;
  int transpose_w;      // store dW^T (the mixture's (k,in,out) masters)
};

__device__ __forceinline__ void narrow_bwd_body(
    u16 (*s_dy)[TPAD], u16 (*s_dyT)[TPAD], u16 (*s_a)[TPAD],
    u16 (*s_aT)[TPAD], u16 (*s_wT)[TPAD],
    const u16* __restrict__ dy_last, const NarrowBwdDesc& d,
    float* __restrict__ arena, long s_stride, int M, int Kin,
    int s, long g) {
  const int m0 = s * TBM;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  const int lr = tid >> 2, lc0 = (tid & 3) * 16;
  const int rowlim = (M - m0 < TBM ? M - m0 : TBM);

  // ---- stage dy_{L-1} (no activation on the last layer) --------------
  {
    const int N = d.N[d.L - 1];
    const u16* dyg = dy_last + g * (long)M * N;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int col = lc0 + j;
      const u16 v = (lr < rowlim && col < N)
          ? dyg[((long)(m0 + lr)) * N + col] : (u16)0;
      s_dy[lr][col] = v;
      s_dyT[col][lr] = v;
    }
  }

  for (int i = d.L - 1; i >= 0; --i) {
    const int N = d.N[i];
    const int K = i == 0 ? Kin : d.N[i - 1];
    const u16* ag = d.acts[i] + g * d.a_gs[i];
    float* wsp = arena + (long)s * s_stride + d.w_off[i] + g * (long)N * K;
    // ---- dW_i = dy^T @ a_i, looping a_i's columns in 64-wide tiles
    // (layer 0 of the trunk/mlp_context chains has K = 768) -----------
    for (int c0 = 0; c0 < K; c0 += TBN) {
      __syncthreads();   // prior readers of s_a/s_aT done
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int col = c0 + lc0 + j;
        const u16 v = (lr < rowlim && col < K)
            ? ag[((long)(m0 + lr)) * K + col] : (u16)0;
        s_aT[lc0 + j][lr] = v;
        // natural image only needed for the dx relu mask (i>0, where
        // K <= 64 so this is the one and only tile)
        s_a[lr][lc0 + j] = v;
      }
      __syncthreads();
      f32x4 acc00{}, acc01{}, acc10{}, acc11{};
#pragma unroll
      for (int kk = 0; kk < TBK; kk += 32) {
        const bf16x8 a0 = *(const bf16x8*)&s_dyT[wr + fi][kk + fk * 8];
        const bf16x8 a1 = *(const bf16x8*)&s_dyT[wr + 16 + fi][kk + fk * 8];
        const bf16x8 b0 = *(const bf16x8*)&s_aT[wc + fi][kk + fk * 8];
        const bf16x8 b1 = *(const bf16x8*)&s_aT[wc + 16 + fi][kk + fk * 8];
        acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
        acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
        acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
      }
      const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const f32x4 a = *accs[mi][ni];
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = wr + mi * 16 + fk * 4 + r;   // n
            const int col = c0 + wc + ni * 16 + fi;      // k
            if (row < N && col < K)
              wsp[d.transpose_w ? (long)col * N + row
                                : (long)row * K + col] = a[r];
          }
        }
    }
    if (tid < TBN && tid < N) {   // db_i = col-sums of dy
      float db = 0.f;
#pragma unroll
      for (int m = 0; m < TBM; ++m) {
        union { float f; unsigned u; } v;
        v.u = ((unsigned)s_dyT[tid][m]) << 16;
        db += v.f;
      }
      arena[(long)s * s_stride + d.b_off[i] + g * (long)N + tid] = db;
    }
    if (i == 0) return;
    // ---- dy_{i-1} = (dy_i @ W_i) * relu-mask(a_i) --------------------
    {
      const u16* wg = d.w[i] + g * (long)N * K;
      __syncthreads();
#pragma unroll
      for (int j = 0; j < 16; ++j) {           // W^T: [k][n]
        const int gk = lc0 + j;
        s_wT[gk][lr] = (lr < N && gk < K)
            ? wg[(long)lr * K + gk] : (u16)0;
      }
      __syncthreads();
      f32x4 acc00{}, acc01{}, acc10{}, acc11{};
#pragma unroll
      for (int kk = 0; kk < TBK; kk += 32) {
        const bf16x8 a0 = *(const bf16x8*)&s_dy[wr + fi][kk + fk * 8];
        const bf16x8 a1 = *(const bf16x8*)&s_dy[wr + 16 + fi][kk + fk * 8];
        const bf16x8 b0 = *(const bf16x8*)&s_wT[wc + fi][kk + fk * 8];
        const bf16x8 b1 = *(const bf16x8*)&s_wT[wc + 16 + fi][kk + fk * 8];
        acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc00, 0, 0, 0);
        acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc01, 0, 0, 0);
        acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc10, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc11, 0, 0, 0);
      }
      __syncthreads();   // s_dy free for the next layer's image
      const f32x4* accs[2][2] = {{&acc00, &acc01}, {&acc10, &acc11}};
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const f32x4 a = *accs[mi][ni];
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = wr + mi * 16 + fk * 4 + r;   // m
            const int col = wc + ni * 16 + fi;           // k (< 64)
            u16 v = 0;
            if (s_a[row][col] != 0) v = f32_to_bf16_rne(a[r]);
            s_dy[row][col] = v;
            s_dyT[col][row] = v;
          }
        }
    }
  }
}

// ---------------------------------------------------------------------------
// CARE attention pool, fused (reference state_encoder.py:85-94).
// fwd: alpha = softmax(logits[M,E]); z_enc[M,D] = sum_e alpha_e * z_encs[e,M,D]
//      (the reference's divide by alpha.sum() is a softmax no-op, skipped).
// bwd: given dz (grad wrt z_enc, a strided bf16 row-slice of the head-input
//      gradient), emit  d_z_encs[e] = alpha_e * dz   (bf16, feeds the grouped
//      mixture backward) and  dlogits = alpha*(s - <alpha,s>), s_e = <dz,z_e>
//      (bf16, feeds the trunk backward) — one launch instead of the ~8
//      mul/sum/softmax-backward kernels autograd records.
// One wavefront per batch row; requires D <= 64 and E <= 16 (cfg: 50, 6).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_bf16_mlp_narrow_bwd(
    const u16* __restrict__ dy_last, NarrowBwdDesc d,
    float* __restrict__ arena, long s_stride, int M, int Kin) {
  __shared__ u16 s_dy[TBM][TPAD];    // dy_i   [m][n]
  __shared__ u16 s_dyT[TBM][TPAD];   // dy_i^T [n][m]
  __shared__ u16 s_a[TBM][TPAD];     // a_i    [m][k]
  __shared__ u16 s_aT[TBM][TPAD];    // a_i^T  [k][m]
  __shared__ u16 s_wT[TBM][TPAD];    // W_i^T  [k][n]
  narrow_bwd_body(s_dy, s_dyT, s_a, s_aT, s_wT, dy_last, d, arena,
                  s_stride, M, Kin, blockIdx.x, blockIdx.z);
}

// Multi-chain narrow backward (round 2): the SE backward's three chains
// (mlp_context, trunk, mixture) in ONE launch, all writing the same
// phase arena.  grid (sum_c S_c * G_c).
struct NarrowBwdMultiDesc {
  NarrowBwdDesc d[4];
  const u16* dy[4];
  int M[4], Kin[4], S[4];
  int cum[5];
  int C;
};

__global__ __launch_bounds__(256) void k_bf16_mlp_narrow_bwd_multi(
    NarrowBwdMultiDesc nm, float* __restrict__ arena, long s_stride) {
  __shared__ u16 s_dy[TBM][TPAD];
  __shared__ u16 s_dyT[TBM][TPAD];
  __shared__ u16 s_a[TBM][TPAD];
  __shared__ u16 s_aT[TBM][TPAD];
  __shared__ u16 s_wT[TBM][TPAD];
  const int bx = blockIdx.x;
  int c = 0;
  while (c + 1 < nm.C && bx >= nm.cum[c + 1]) ++c;
  const int local = bx - nm.cum[c];
  const long g = local / nm.S[c];
  const int sblk = local % nm.S[c];
  narrow_bwd_body(s_dy, s_dyT, s_a, s_aT, s_wT, nm.dy[c], nm.d[c], arena,
                  s_stride, nm.M[c], nm.Kin[c], sblk, g);
}

// enc_out mode (round 2): logits/alpha/hc have Mz = M/rep rows (the
// trunk and context projections run ONCE on deduplicated z_context rows
// — the batched [next|current] SE forward duplicated them); the pooled
// output row r uses alpha[r % Mz] and writes the CONCATENATED head input
// enc[r] = [hc[r % Mz] | pool(r)] directly (the torch.cat launch is
// gone).  Legacy mode (enc_out == null): rep == 1, emit z_enc only.
__global__ __launch_bounds__(256) void k_attn_pool_fwd(
    const float* __restrict__ logits, const float* __restrict__ z_encs,
    float* __restrict__ alpha_out, u16* __restrict__ z_enc_out,
    int M, int E, int D, int rep, const u16* __restrict__ hc, int zc,
    u16* __restrict__ enc_out) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (row >= M) return;
  const int Mz = M / rep;
  const int zrow = row % Mz;
  float a[16];
  float mx = -1e30f;
  for (int e = 0; e < E; ++e) { a[e] = logits[(long)zrow * E + e]; mx = fmaxf(mx, a[e]); }
  float den = 0.f;
  for (int e = 0; e < E; ++e) { a[e] = __expf(a[e] - mx); den += a[e]; }
  const float inv = 1.f / den;
  for (int e = 0; e < E; ++e) a[e] *= inv;
  if (lane < E && row == zrow) alpha_out[(long)zrow * E + lane] = a[lane];
  if (enc_out != nullptr) {
    const int W = zc + D;
    for (int c = lane; c < W; c += 64) {
      if (c < zc) {
        enc_out[(long)row * W + c] = hc[(long)zrow * zc + c];
      } else {
        const int d = c - zc;
        float acc = 0.f;
        for (int e = 0; e < E; ++e)
          acc += a[e] * z_encs[((long)e * M + row) * D + d];
        enc_out[(long)row * W + c] = f32_to_bf16_rne(acc);
      }
    }
    return;
  }
  for (int d = lane; d < D; d += 64) {
    float acc = 0.f;
    for (int e = 0; e < E; ++e)
      acc += a[e] * z_encs[((long)e * M + row) * D + d];
    z_enc_out[(long)row * D + d] = f32_to_bf16_rne(acc);
  }
}

__device__ __forceinline__ float wave_sum64(float v) {
  for (int o = 32; o; o >>= 1) v += __shfl_down(v, o, 64);
  return __shfl(v, 0, 64);
}

__global__ __launch_bounds__(256) void k_attn_pool_bwd(
    const float* __restrict__ z_encs, const float* __restrict__ alpha,
    const u16* __restrict__ dz, long dz_ld, long dz_off,
    u16* __restrict__ d_zencs, u16* __restrict__ dlogits,
    int M, int E, int D) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (row >= M) return;
  float dzv = 0.f;
  if (lane < D) {
    union { float f; unsigned u; } v;
    v.u = ((unsigned)dz[(long)row * dz_ld + dz_off + lane]) << 16;
    dzv = v.f;
  }
  float s[16];
  float t = 0.f;
  for (int e = 0; e < E; ++e) {
    const float ae = alpha[(long)row * E + e];
    const float ze = lane < D ? z_encs[((long)e * M + row) * D + lane] : 0.f;
    s[e] = wave_sum64(ze * dzv);
    t += ae * s[e];
    if (lane < D)
      d_zencs[((long)e * M + row) * D + lane] = f32_to_bf16_rne(ae * dzv);
  }
  if (lane < E) {
    const float ae = alpha[(long)row * E + lane];
    dlogits[(long)row * E + lane] = f32_to_bf16_rne(ae * (s[lane] - t));
  }
}

// fp32 partial fold (duplicate of the fp32 TU's reducer; no RDC linking)
__global__ __launch_bounds__(256) void k_reduce_partials2(
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

// one launch folds BOTH the dw slabs and the db slabs (flat index over
// G*(N*K) then G*N)
__global__ __launch_bounds__(256) void k_reduce_dwdb(
    const float* __restrict__ ws, const float* __restrict__ ws_db,
    float* __restrict__ dw, float* __restrict__ db, long nw_per_g,
    long nb_per_g, int S, int G) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long tot_w = (long)G * nw_per_g;
  if (i < tot_w) {
    const int g = (int)(i / nw_per_g);
    const long e = i % nw_per_g;
    const float* base = ws + ((long)g * S) * nw_per_g + e;
    float acc = 0.f;
    for (int s = 0; s < S; ++s) acc += base[(long)s * nw_per_g];
    dw[i] = acc;
    return;
  }
  i -= tot_w;
  if (i < (long)G * nb_per_g) {
    const int g = (int)(i / nb_per_g);
    const long e = i % nb_per_g;
    const float* base = ws_db + ((long)g * S) * nb_per_g + e;
    float acc = 0.f;
    for (int s = 0; s < S; ++s) acc += base[(long)s * nb_per_g];
    db[i] = acc;
  }
}

__global__ __launch_bounds__(256) void k_reduce_arena(
    const float* __restrict__ arena, float* __restrict__ out, long stride,
    long n, int S, long lo) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float acc = 0.f;
  for (int s = 0; s < S; ++s) acc += arena[(long)s * stride + lo + i];
  out[lo + i] = acc;
}

// ===========================================================================
// Host wrappers
// ===========================================================================

static void f32_to_bf16_(torch::Tensor src, torch::Tensor dst) {
  CHECK_F32(src); CHECK_BF16(dst);
  const long n = src.numel();
  TORCH_CHECK(dst.numel() == n);
  hipLaunchKernelGGL(k_cast_f32_bf16, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream2(), src.data_ptr<float>(),
                     (u16*)dst.data_ptr(), n);
}

static torch::Tensor linear_act_fwd_bf16(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, long act, long G,
                                         long out_f32) {
  CHECK_BF16(x); CHECK_BF16(w); CHECK_F32(b);
  auto xc = x.contiguous(); auto wc = w.contiguous(); auto bc = b.contiguous();
  const bool per_group_x = xc.dim() == 3;
  const long M = per_group_x ? xc.size(1) : xc.size(0);
  const long K = xc.size(-1);
  const long N = wc.numel() / (G * K);
  const long xgs = per_group_x ? M * K : 0;
  auto opts = xc.options().dtype(out_f32 ? torch::kFloat32 : torch::kBFloat16);
  auto y = G == 1 ? torch::empty({M, N}, opts)
                  : torch::empty({G, M, N}, opts);
  dim3 grid((M + TBM - 1) / TBM, (N + TBN - 1) / TBN, G);
  hipLaunchKernelGGL(k_bf16_fwd, grid, dim3(256), 0, cur_stream2(),
                     (const u16*)xc.data_ptr(), (const u16*)wc.data_ptr(),
                     bc.data_ptr<float>(), y.data_ptr(),
                     (int)M, (int)N, (int)K, (int)act, xgs, (int)out_f32);
  return y;
}

static torch::Tensor linear_bwd_dx_bf16(torch::Tensor dy, torch::Tensor w,
                                        torch::Tensor yout, long act,
                                        long G, long sum_over_g) {
  CHECK_BF16(dy); CHECK_BF16(w); CHECK_BF16(yout);
  auto dyc = dy.contiguous(); auto wc = w.contiguous();
  auto yc = yout.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  const long N = G == 1 ? dyc.size(1) : dyc.size(2);
  const long K = wc.size(-1);
  const long Gz = (G == 1 || sum_over_g) ? 1 : G;
  const int Gin = (int)((G > 1 && sum_over_g) ? G : 1);
  auto dx = Gz == 1 ? torch::empty({M, K}, dyc.options())
                    : torch::empty({Gz, M, K}, dyc.options());
  dim3 grid((M + TBM - 1) / TBM, (K + TBN - 1) / TBN, Gz);
  hipLaunchKernelGGL(k_bf16_dx, grid, dim3(256), 0, cur_stream2(),
                     (const u16*)dyc.data_ptr(), (const u16*)wc.data_ptr(),
                     (const u16*)yc.data_ptr(), (u16*)dx.data_ptr(),
                     (int)M, (int)N, (int)K, (int)act, Gin);
  return dx;
}

static std::vector<torch::Tensor> linear_bwd_dwdb_bf16(
    torch::Tensor dy, torch::Tensor x, torch::Tensor yout, long act,
    long G) {
  CHECK_BF16(dy); CHECK_BF16(x); CHECK_BF16(yout);
  auto dyc = dy.contiguous(); auto xc = x.contiguous();
  auto yc = yout.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  const long N = G == 1 ? dyc.size(1) : dyc.size(2);
  const long K = xc.size(-1);
  const long xgs = xc.dim() == 3 ? M * K : 0;
  const long nbx = (N + TBM - 1) / TBM, nby = (K + TBN - 1) / TBN;
  long S = std::max<long>(1, 512 / std::max<long>(1, nbx * nby * G));
  S = std::min<long>(S, (M + TBK - 1) / TBK);
  const int chunk = (int)(((M + S - 1) / S + TBK - 1) / TBK * TBK);
  S = (M + chunk - 1) / chunk;
  auto fopts = dyc.options().dtype(torch::kFloat32);
  auto ws = torch::empty({G * S, N, K}, fopts);
  auto ws_db = torch::empty({G * S, N}, fopts);
  dim3 grid(nbx, nby, G * S);
  hipLaunchKernelGGL(k_bf16_dwdb_splitk, grid, dim3(256), 0, cur_stream2(),
                     (const u16*)dyc.data_ptr(), (const u16*)xc.data_ptr(),
                     (const u16*)yc.data_ptr(), ws.data_ptr<float>(),
                     ws_db.data_ptr<float>(), (int)M, (int)N, (int)K,
                     (int)act, (int)S, chunk, xgs, 0, 0, 0, 0);
  auto dw = G == 1 ? torch::empty({N, K}, fopts)
                   : torch::empty({G, N, K}, fopts);
  auto db = G == 1 ? torch::empty({N}, fopts)
                   : torch::empty({G, N}, fopts);
  if (S == 1) {
    dw.copy_(ws.view_as(dw));
    db.copy_(ws_db.view_as(db));
  } else {
    const long tot = G * (N * K + N);
    hipLaunchKernelGGL(k_reduce_dwdb, dim3((tot + 255) / 256), dim3(256), 0,
                       cur_stream2(), ws.data_ptr<float>(),
                       ws_db.data_ptr<float>(), dw.data_ptr<float>(),
                       db.data_ptr<float>(), N * K, N, (int)S, (int)G);
  }
  return {dw, db};
}

static std::vector<torch::Tensor> attn_pool_fwd(torch::Tensor logits,
                                                torch::Tensor z_encs) {
  CHECK_F32(logits); CHECK_F32(z_encs);
  auto lc = logits.contiguous(); auto zc = z_encs.contiguous();
  const long M = lc.size(0), E = lc.size(1), D = zc.size(2);
  TORCH_CHECK(zc.size(0) == E && zc.size(1) == M && E <= 16 && D <= 64,
              "attn_pool_fwd: need z_encs (E,M,D), E<=16, D<=64");
  auto alpha = torch::empty({M, E}, lc.options());
  auto z_enc = torch::empty({M, D}, lc.options().dtype(torch::kBFloat16));
  hipLaunchKernelGGL(k_attn_pool_fwd, dim3((M + 3) / 4), dim3(256), 0,
                     cur_stream2(), lc.data_ptr<float>(), zc.data_ptr<float>(),
                     alpha.data_ptr<float>(), (u16*)z_enc.data_ptr(),
                     (int)M, (int)E, (int)D, 1, nullptr, 0, nullptr);
  return {alpha, z_enc};
}

// pool + concat fused, with the trunk logits / context projection hc
// computed ONCE on deduplicated rows (rep = M / logits_rows):
//   enc[r] = [ hc[r % Mz] | sum_e alpha[r % Mz][e] * z_encs[e][r] ]
static std::vector<torch::Tensor> attn_pool_fwd_enc(torch::Tensor logits,
                                                    torch::Tensor z_encs,
                                                    torch::Tensor hc,
                                                    long rep) {
  CHECK_F32(logits); CHECK_F32(z_encs); CHECK_BF16(hc);
  auto lc = logits.contiguous();
  auto zz = z_encs.contiguous();
  auto hcc = hc.contiguous();
  const long Mz = lc.size(0), E = lc.size(1), D = zz.size(2);
  const long M = Mz * rep;
  const long zcw = hcc.size(1);
  TORCH_CHECK(zz.size(0) == E && zz.size(1) == M && E <= 16 && D <= 64,
              "attn_pool_fwd_enc: need z_encs (E, Mz*rep, D), E<=16, D<=64");
  TORCH_CHECK(hcc.size(0) == Mz);
  auto alpha = torch::empty({Mz, E}, lc.options());
  auto enc = torch::empty({M, zcw + D},
                          lc.options().dtype(torch::kBFloat16));
  hipLaunchKernelGGL(k_attn_pool_fwd, dim3((M + 3) / 4), dim3(256), 0,
                     cur_stream2(), lc.data_ptr<float>(),
                     zz.data_ptr<float>(), alpha.data_ptr<float>(),
                     nullptr, (int)M, (int)E, (int)D, (int)rep,
                     (const u16*)hcc.data_ptr(), (int)zcw,
                     (u16*)enc.data_ptr());
  return {alpha, enc};
}

static std::vector<torch::Tensor> attn_pool_bwd(torch::Tensor z_encs,
                                                torch::Tensor alpha,
                                                torch::Tensor dz, long dz_ld,
                                                long dz_off) {
  CHECK_F32(z_encs); CHECK_F32(alpha); CHECK_BF16(dz);
  TORCH_CHECK(dz.is_contiguous(), "attn_pool_bwd: dz base must be contiguous");
  auto ac = alpha.contiguous(); auto zc = z_encs.contiguous();
  const long E = zc.size(0), M = zc.size(1), D = zc.size(2);
  TORCH_CHECK(E <= 16 && D <= 64);
  auto dzencs = torch::empty({E, M, D}, dz.options());
  auto dlogits = torch::empty({M, E}, dz.options());
  hipLaunchKernelGGL(k_attn_pool_bwd, dim3((M + 3) / 4), dim3(256), 0,
                     cur_stream2(), zc.data_ptr<float>(), ac.data_ptr<float>(),
                     (const u16*)dz.data_ptr(), dz_ld, dz_off,
                     (u16*)dzencs.data_ptr(), (u16*)dlogits.data_ptr(),
                     (int)M, (int)E, (int)D);
  return {dzencs, dlogits};
}

static std::vector<torch::Tensor> mlp_narrow_fwd_bf16(
    torch::Tensor x, std::vector<torch::Tensor> ws,
    std::vector<torch::Tensor> bs, long G, long act_last, long out_f32,
    long save) {
  CHECK_BF16(x);
  const int L = (int)ws.size();
  TORCH_CHECK(L >= 2 && L <= 6 && bs.size() == ws.size());
  auto xc = x.contiguous();
  const bool per_group_x = xc.dim() == 3;
  const long M = per_group_x ? xc.size(1) : xc.size(0);
  const long K0 = xc.size(-1);
  const long xgs = per_group_x ? M * K0 : 0;
  NarrowDesc d{};
  d.L = L;
  std::vector<torch::Tensor> keep;   // contiguity holders
  long K = K0;
  std::vector<torch::Tensor> out;
  out.resize(1);
  for (int i = 0; i < L; ++i) {
    CHECK_BF16(ws[i]); CHECK_F32(bs[i]);
    auto wc = ws[i].contiguous(); auto bc = bs[i].contiguous();
    keep.push_back(wc); keep.push_back(bc);
    const long N = wc.numel() / (G * K);
    TORCH_CHECK(N <= 64, "narrow MLP: layer width must be <= 64");
    TORCH_CHECK(i == 0 || K <= 64);
    d.w[i] = (const u16*)wc.data_ptr();
    d.b[i] = bc.data_ptr<float>();
    d.N[i] = (int)N;
    d.acts[i] = nullptr;
    if (save && i < L - 1) {
      auto a = G == 1 ? torch::empty({M, N}, xc.options())
                      : torch::empty({G, M, N}, xc.options());
      d.acts[i] = (u16*)a.data_ptr();
      out.push_back(a);
    }
    K = N;
  }
  auto yopts = xc.options().dtype(out_f32 ? torch::kFloat32
                                          : torch::kBFloat16);
  auto y = G == 1 ? torch::empty({M, (long)d.N[L - 1]}, yopts)
                  : torch::empty({G, M, (long)d.N[L - 1]}, yopts);
  out[0] = y;
  dim3 grid((M + TBM - 1) / TBM, 1, G);
  hipLaunchKernelGGL(k_bf16_mlp_narrow, grid, dim3(256), 0, cur_stream2(),
                     (const u16*)xc.data_ptr(), d, y.data_ptr(), (int)M,
                     (int)K0, xgs, (int)act_last, (int)out_f32);
  return out;
}

// Multi-chain narrow forward: per chain returns [y, act_0, ...] like
// mlp_narrow_fwd_bf16, all chains in ONE kernel launch.
static std::vector<std::vector<torch::Tensor>> mlp_narrow_fwd_multi(
    std::vector<torch::Tensor> xs,
    std::vector<std::vector<torch::Tensor>> wss,
    std::vector<std::vector<torch::Tensor>> bss,
    std::vector<long> Gs, std::vector<long> act_lasts,
    std::vector<long> out_f32s, long save) {
  const int C = (int)xs.size();
  TORCH_CHECK(C >= 1 && C <= 4 && (int)wss.size() == C
              && (int)bss.size() == C && (int)Gs.size() == C
              && (int)act_lasts.size() == C && (int)out_f32s.size() == C);
  NarrowMultiDesc nm{};
  nm.C = C;
  std::vector<torch::Tensor> keep;
  std::vector<std::vector<torch::Tensor>> outs(C);
  int cum = 0;
  for (int ci = 0; ci < C; ++ci) {
    auto& ws = wss[ci];
    auto& bs = bss[ci];
    const long G = Gs[ci];
    const int L = (int)ws.size();
    TORCH_CHECK(L >= 1 && L <= 6 && (int)bs.size() == L);
    auto xc = xs[ci].contiguous();
    CHECK_BF16(xc);
    keep.push_back(xc);
    const bool per_group_x = xc.dim() == 3;
    const long M = per_group_x ? xc.size(1) : xc.size(0);
    const long K0 = xc.size(-1);
    NarrowDesc d{};
    d.L = L;
    long K = K0;
    outs[ci].resize(1);
    for (int i = 0; i < L; ++i) {
      CHECK_BF16(ws[i]);
      CHECK_F32(bs[i]);
      auto wc = ws[i].contiguous();
      auto bc = bs[i].contiguous();
      keep.push_back(wc);
      keep.push_back(bc);
      const long N = wc.numel() / (G * K);
      TORCH_CHECK(N <= 64, "narrow multi: layer width must be <= 64");
      TORCH_CHECK(i == 0 || K <= 64);
      d.w[i] = (const u16*)wc.data_ptr();
      d.b[i] = bc.data_ptr<float>();
      d.N[i] = (int)N;
      d.acts[i] = nullptr;
      if (save && i < L - 1) {
        auto a = G == 1 ? torch::empty({M, N}, xc.options())
                        : torch::empty({G, M, N}, xc.options());
        d.acts[i] = (u16*)a.data_ptr();
        outs[ci].push_back(a);
      }
      K = N;
    }
    auto yopts = xc.options().dtype(out_f32s[ci] ? torch::kFloat32
                                                 : torch::kBFloat16);
    auto y = G == 1 ? torch::empty({M, (long)d.N[L - 1]}, yopts)
                    : torch::empty({G, M, (long)d.N[L - 1]}, yopts);
    outs[ci][0] = y;
    nm.d[ci] = d;
    nm.x[ci] = (const u16*)xc.data_ptr();
    nm.y[ci] = y.data_ptr();
    nm.M[ci] = (int)M;
    nm.K0[ci] = (int)K0;
    nm.act_last[ci] = (int)act_lasts[ci];
    nm.out_f32[ci] = (int)out_f32s[ci];
    nm.xgs[ci] = per_group_x ? M * K0 : 0;
    nm.nbm[ci] = (int)((M + TBM - 1) / TBM);
    nm.cum[ci] = cum;
    cum += nm.nbm[ci] * (int)G;
  }
  nm.cum[C] = cum;
  hipLaunchKernelGGL(k_bf16_mlp_narrow_multi, dim3(cum), dim3(256), 0,
                     cur_stream2(), nm);
  return outs;
}

static void linear_bwd_dwdb_arena(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor yout, long act, long G,
                                  torch::Tensor arena, long w_off,
                                  long b_off, long S, long chunk,
                                  long transpose_w) {
  CHECK_BF16(dy); CHECK_BF16(x); CHECK_BF16(yout); CHECK_F32(arena);
  auto dyc = dy.contiguous(); auto xc = x.contiguous();
  auto yc = yout.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  const long N = G == 1 ? dyc.size(1) : dyc.size(2);
  const long K = xc.size(-1);
  const long xgs = xc.dim() == 3 ? M * K : 0;
  const long stride = arena.size(1);
  TORCH_CHECK(arena.dim() == 2 && arena.size(0) >= S && arena.is_contiguous());
  TORCH_CHECK(w_off + G * N * K <= stride && b_off + G * N <= stride);
  const long nbx = (N + TBM - 1) / TBM, nby = (K + TBN - 1) / TBN;
  dim3 grid(nbx, nby, G * S);
  hipLaunchKernelGGL(k_bf16_dwdb_splitk, grid, dim3(256), 0, cur_stream2(),
                     (const u16*)dyc.data_ptr(), (const u16*)xc.data_ptr(),
                     (const u16*)yc.data_ptr(), arena.data_ptr<float>(),
                     nullptr, (int)M, (int)N, (int)K, (int)act, (int)S,
                     (int)chunk, xgs, (int)transpose_w, stride, w_off,
                     b_off);
}

static void reduce_arena(torch::Tensor arena, torch::Tensor out, long S,
                         long lo, long hi) {
  CHECK_F32(arena); CHECK_F32(out);
  TORCH_CHECK(arena.dim() == 2 && arena.is_contiguous()
              && out.is_contiguous() && arena.size(1) == out.numel());
  if (hi < 0) hi = out.numel();
  const long n = hi - lo;
  TORCH_CHECK(lo >= 0 && n >= 0 && hi <= out.numel());
  if (n == 0) return;
  hipLaunchKernelGGL(k_reduce_arena, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream2(), arena.data_ptr<float>(),
                     out.data_ptr<float>(), arena.size(1), n, (int)S, lo);
}

static void mlp_narrow_bwd_bf16(torch::Tensor dy_last,
                                std::vector<torch::Tensor> acts,
                                std::vector<torch::Tensor> ws,
                                torch::Tensor arena,
                                std::vector<long> w_offs,
                                std::vector<long> b_offs, long G,
                                long transpose_w) {
  CHECK_BF16(dy_last); CHECK_F32(arena);
  const int L = (int)ws.size();
  TORCH_CHECK(L >= 2 && L <= 6 && (int)acts.size() == L
              && (int)w_offs.size() == L && (int)b_offs.size() == L);
  TORCH_CHECK(arena.dim() == 2 && arena.is_contiguous());
  auto dyc = dy_last.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  NarrowBwdDesc d{};
  d.L = L;
  d.transpose_w = (int)transpose_w;
  std::vector<torch::Tensor> keep;
  long Kin = acts[0].size(-1);
  long K = Kin;
  for (int i = 0; i < L; ++i) {
    CHECK_BF16(ws[i]); CHECK_BF16(acts[i]);
    auto wc = ws[i].contiguous(); auto ac = acts[i].contiguous();
    keep.push_back(wc); keep.push_back(ac);
    const long N = wc.numel() / (G * K);
    TORCH_CHECK(N <= 64 && (i == 0 || K <= 64));
    d.w[i] = (const u16*)wc.data_ptr();
    d.acts[i] = (const u16*)ac.data_ptr();
    d.a_gs[i] = ac.dim() == 3 ? M * K : 0;   // 2-D act = shared across G
    TORCH_CHECK(ac.numel() == (ac.dim() == 3 ? G : 1) * M * K);
    d.w_off[i] = w_offs[i];
    d.b_off[i] = b_offs[i];
    d.N[i] = (int)N;
    TORCH_CHECK(w_offs[i] + G * N * K <= arena.size(1)
                && b_offs[i] + G * N <= arena.size(1));
    K = N;
  }
  TORCH_CHECK(dyc.numel() == G * M * d.N[L - 1]);
  const long S = (M + TBM - 1) / TBM;
  TORCH_CHECK(arena.size(0) >= S, "arena rows must cover ceil(M/64)");
  dim3 grid(S, 1, G);
  hipLaunchKernelGGL(k_bf16_mlp_narrow_bwd, grid, dim3(256), 0,
                     cur_stream2(), (const u16*)dyc.data_ptr(), d,
                     arena.data_ptr<float>(), arena.size(1), (int)M,
                     (int)Kin);
}

static void dwdb_grouped_arena(std::vector<torch::Tensor> dys,
                               std::vector<torch::Tensor> xs,
                               torch::Tensor arena,
                               std::vector<long> w_offs,
                               std::vector<long> b_offs, long G, long S,
                               long chunk) {
  CHECK_F32(arena);
  const int L = (int)dys.size();
  TORCH_CHECK(L >= 1 && L <= 6 && (int)xs.size() == L
              && (int)w_offs.size() == L && (int)b_offs.size() == L);
  TORCH_CHECK(arena.dim() == 2 && arena.is_contiguous()
              && arena.size(0) >= S);
  const long stride = arena.size(1);
  DwGroupDesc d{};
  d.L = L;
  d.S = (int)S;
  d.chunk = (int)chunk;
  std::vector<torch::Tensor> keep;
  long M = -1;
  int cum = 0;
  for (int i = 0; i < L; ++i) {
    CHECK_BF16(dys[i]);
    CHECK_BF16(xs[i]);
    auto dc = dys[i].contiguous();
    auto xc = xs[i].contiguous();
    keep.push_back(dc);
    keep.push_back(xc);
    TORCH_CHECK(dc.dim() == 3 && dc.size(0) == G);
    const long Mi = dc.size(1), N = dc.size(2);
    const long K = xc.size(-1);
    if (M < 0) M = Mi;
    TORCH_CHECK(Mi == M, "all layers must share the batch dim");
    TORCH_CHECK(xc.numel() == (xc.dim() == 3 ? G : 1) * M * K);
    d.dy[i] = (const u16*)dc.data_ptr();
    d.x[i] = (const u16*)xc.data_ptr();
    d.xgs[i] = xc.dim() == 3 ? M * K : 0;
    d.w_off[i] = w_offs[i];
    d.b_off[i] = b_offs[i];
    TORCH_CHECK(w_offs[i] + G * N * K <= stride
                && b_offs[i] + G * N <= stride);
    d.N[i] = (int)N;
    d.K[i] = (int)K;
    d.nbx[i] = (int)((N + TBM - 1) / TBM);
    d.cum[i] = cum;
    cum += d.nbx[i] * (int)((K + TBN - 1) / TBN);
  }
  d.cum[L] = cum;
  d.M = (int)M;
  dim3 grid(cum, 1, (unsigned)(G * S));
  hipLaunchKernelGGL(k_bf16_dwdb_grouped, grid, dim3(256), 0,
                     cur_stream2(), d, arena.data_ptr<float>(), stride);
}

static void mlp_narrow_bwd_multi(
    std::vector<torch::Tensor> dy_lasts,
    std::vector<std::vector<torch::Tensor>> actss,
    std::vector<std::vector<torch::Tensor>> wss,
    torch::Tensor arena,
    std::vector<std::vector<long>> w_offss,
    std::vector<std::vector<long>> b_offss,
    std::vector<long> Gs, std::vector<long> transpose_ws) {
  CHECK_F32(arena);
  const int C = (int)dy_lasts.size();
  TORCH_CHECK(C >= 1 && C <= 4 && (int)actss.size() == C
              && (int)wss.size() == C && (int)w_offss.size() == C
              && (int)b_offss.size() == C && (int)Gs.size() == C
              && (int)transpose_ws.size() == C);
  TORCH_CHECK(arena.dim() == 2 && arena.is_contiguous());
  NarrowBwdMultiDesc nm{};
  nm.C = C;
  std::vector<torch::Tensor> keep;
  int cum = 0;
  for (int ci = 0; ci < C; ++ci) {
    auto dyc = dy_lasts[ci].contiguous();
    CHECK_BF16(dyc);
    keep.push_back(dyc);
    const long G = Gs[ci];
    auto& ws = wss[ci];
    auto& acts = actss[ci];
    const int L = (int)ws.size();
    TORCH_CHECK(L >= 2 && L <= 6 && (int)acts.size() == L
                && (int)w_offss[ci].size() == L
                && (int)b_offss[ci].size() == L);
    const long M = G == 1 ? dyc.size(0) : dyc.size(1);
    NarrowBwdDesc d{};
    d.L = L;
    d.transpose_w = (int)transpose_ws[ci];
    long Kin = acts[0].size(-1);
    long K = Kin;
    for (int i = 0; i < L; ++i) {
      CHECK_BF16(ws[i]);
      CHECK_BF16(acts[i]);
      auto wc = ws[i].contiguous();
      auto ac = acts[i].contiguous();
      keep.push_back(wc);
      keep.push_back(ac);
      const long N = wc.numel() / (G * K);
      TORCH_CHECK(N <= 64 && (i == 0 || K <= 64));
      d.w[i] = (const u16*)wc.data_ptr();
      d.acts[i] = (const u16*)ac.data_ptr();
      d.a_gs[i] = ac.dim() == 3 ? M * K : 0;
      TORCH_CHECK(ac.numel() == (ac.dim() == 3 ? G : 1) * M * K);
      d.w_off[i] = w_offss[ci][i];
      d.b_off[i] = b_offss[ci][i];
      d.N[i] = (int)N;
      TORCH_CHECK(w_offss[ci][i] + G * N * K <= arena.size(1)
                  && b_offss[ci][i] + G * N <= arena.size(1));
      K = N;
    }
    TORCH_CHECK(dyc.numel() == G * M * d.N[L - 1]);
    const long S = (M + TBM - 1) / TBM;
    TORCH_CHECK(arena.size(0) >= S, "arena rows must cover ceil(M/64)");
    nm.d[ci] = d;
    nm.dy[ci] = (const u16*)dyc.data_ptr();
    nm.M[ci] = (int)M;
    nm.Kin[ci] = (int)Kin;
    nm.S[ci] = (int)S;
    nm.cum[ci] = cum;
    cum += (int)(S * G);
  }
  nm.cum[C] = cum;
  hipLaunchKernelGGL(k_bf16_mlp_narrow_bwd_multi, dim3(cum), dim3(256), 0,
                     cur_stream2(), nm, arena.data_ptr<float>(),
                     arena.size(1));
}

void register_bf16(pybind11::module_& m) {
  m.def("mlp_narrow_bwd_multi", &mlp_narrow_bwd_multi);
  m.def("dwdb_grouped_arena", &dwdb_grouped_arena);
  m.def("f32_to_bf16_", &f32_to_bf16_);
  m.def("attn_pool_fwd", &attn_pool_fwd);
  m.def("attn_pool_fwd_enc", &attn_pool_fwd_enc);
  m.def("mlp_narrow_fwd_bf16", &mlp_narrow_fwd_bf16);
  m.def("mlp_narrow_fwd_multi", &mlp_narrow_fwd_multi);
  m.def("linear_bwd_dwdb_arena", &linear_bwd_dwdb_arena);
  m.def("reduce_arena", &reduce_arena, pybind11::arg("arena"),
        pybind11::arg("out"), pybind11::arg("S"),
        pybind11::arg("lo") = 0, pybind11::arg("hi") = -1);
  m.def("mlp_narrow_bwd_bf16", &mlp_narrow_bwd_bf16);
  m.def("attn_pool_bwd", &attn_pool_bwd);
  m.def("linear_act_fwd_bf16", &linear_act_fwd_bf16);
  m.def("linear_bwd_dx_bf16", &linear_bwd_dx_bf16);
  m.def("linear_bwd_dwdb_bf16", &linear_bwd_dwdb_bf16);
}
