// dsac_kernels.hip — CDNA4 (gfx950) kernels for the SAC hot path.
//
// Replaces the PyTorch op-sequences inventoried in SURVEY.md §2.6:
//   K1  linear_act_fwd       — GEMM + bias + ReLU (MFMA f32 16x16x4)
//   K8  linear_bwd_dx/dwdb   — backward GEMMs with fused ReLU masking +
//                              fused bias-grad column reduction
//   K4  squashed_gaussian_*  — fused clamp/exp/rsample/tanh/log-prob (+bwd)
//   K5  td_target            — Bellman backup elementwise
//   K9  adam_step_           — fused Adam over one flat parameter buffer
//   K10 polyak_              — fused soft target update over flat buffers
//
// Design notes (see /opt/skills guides):
// - fp32 end-to-end like the reference (no mixed precision); GEMMs use the
//   exact f32-input MFMA v_mfma_f32_16x16x4_f32 (155 TF on MI355X — far
//   above what these latency-bound tiny GEMMs need, at fp32-exact numerics).
// - tiles are 64x64x32 with LDS staging; row pads chosen so the MFMA
//   fragment gathers are LDS-bank-conflict-free (+2 on 32-wide rows:
//   bank = (34*r + k) % 32 = (2r + k) % 32 distinct for r<16, k<2;
//   +16 on 64-wide rows: (80*m + j) % 32 = (16m + j) % 32 distinct).
// - one workgroup = 4 waves, each wave owns a 32x32 output sub-tile as a
//   2x2 grid of 16x16 MFMA fragments (wave64 per-wave MFMA, not warp32).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <cmath>
#include <vector>

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;

#define DEV_INLINE __device__ __forceinline__

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
// K1: y[M,N] = act(x[M,K] @ w[N,K]^T + b[N])      act: 0=none, 1=relu
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_act_fwd(
    const float* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ b, float* __restrict__ y,
    int M, int N, int K, int act) {
  __shared__ float sx[BM][PAD_K];
  __shared__ float sw[BN][PAD_K];
  const int m0 = blockIdx.x * BM, n0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;  // fragment row / k index
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  for (int k0 = 0; k0 < K; k0 += BK) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {  // 2048 elements / 256 threads
      const int idx = tid * 8 + j;
      const int r = idx >> 5, c = idx & 31;
      const int gk = k0 + c;
      sx[r][c] = (m0 + r < M && gk < K) ? x[(long)(m0 + r) * K + gk] : 0.f;
      sw[r][c] = (n0 + r < N && gk < K) ? w[(long)(n0 + r) * K + gk] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sx[wr + fi][kk + fk];
      const float a1 = sx[wr + 16 + fi][kk + fk];
      const float b0 = sw[wc + fi][kk + fk];
      const float b1 = sw[wc + 16 + fi][kk + fk];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
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
// K8a: dx[M,K] = (dy * mask)[M,N] @ w[N,K]   (mask = yout > 0 when act==1)
// grid: (ceil(M/64), ceil(K/64))
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_bwd_dx(
    const float* __restrict__ dy, const float* __restrict__ w,
    const float* __restrict__ yout, float* __restrict__ dx,
    int M, int N, int K, int act) {
  __shared__ float sdy[BM][PAD_K];   // [m][n-slice]
  __shared__ float sw[BK][PAD_N];    // [n-slice][k]
  const int m0 = blockIdx.x * BM, c0 = blockIdx.y * BN;  // c over K
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  for (int n0 = 0; n0 < N; n0 += BK) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {  // dy tile: 64x32
      const int idx = tid * 8 + j;
      const int r = idx >> 5, c = idx & 31;
      const int gm = m0 + r, gn = n0 + c;
      float v = 0.f;
      if (gm < M && gn < N) {
        v = dy[(long)gm * N + gn];
        if (act == 1 && yout[(long)gm * N + gn] <= 0.f) v = 0.f;
      }
      sdy[r][c] = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {  // w tile: 32x64, k fast
      const int idx = tid * 8 + j;
      const int r = idx >> 6, c = idx & 63;
      const int gn = n0 + r, gk = c0 + c;
      sw[r][c] = (gn < N && gk < K) ? w[(long)gn * K + gk] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sdy[wr + fi][kk + fk];
      const float a1 = sdy[wr + 16 + fi][kk + fk];
      const float b0 = sw[kk + fk][wc + fi];
      const float b1 = sw[kk + fk][wc + 16 + fi];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
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
// K8b: dw[N,K] = (dy*mask)^T[N,M] @ x[M,K];  db[N] = sum_m (dy*mask)[m][n]
// grid: (ceil(N/64), ceil(K/64)); blocks with blockIdx.y==0 also produce db.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_linear_bwd_dwdb(
    const float* __restrict__ dy, const float* __restrict__ x,
    const float* __restrict__ yout, float* __restrict__ dw,
    float* __restrict__ db, int M, int N, int K, int act) {
  __shared__ float sa[BN][PAD_K];   // dy^T tile: [n][m-slice]
  __shared__ float sb[BK][PAD_N];   // x tile:    [m-slice][k]
  const int n0 = blockIdx.x * BM, c0 = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int wr = (wid >> 1) * 32, wc = (wid & 1) * 32;
  const int fi = lane & 15, fk = lane >> 4;
  const bool do_db = (blockIdx.y == 0);
  float db_acc = 0.f;
  f32x4 acc00{}, acc01{}, acc10{}, acc11{};

  for (int m0 = 0; m0 < M; m0 += BK) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {  // dy block 32(m) x 64(n), transposed store
      const int idx = tid * 8 + j;
      const int n = idx & 63, m = idx >> 6;
      const int gm = m0 + m, gn = n0 + n;
      float v = 0.f;
      if (gm < M && gn < N) {
        v = dy[(long)gm * N + gn];
        if (act == 1 && yout[(long)gm * N + gn] <= 0.f) v = 0.f;
      }
      sa[n][m] = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {  // x tile 32(m) x 64(k)
      const int idx = tid * 8 + j;
      const int m = idx >> 6, c = idx & 63;
      const int gm = m0 + m, gk = c0 + c;
      sb[m][c] = (gm < M && gk < K) ? x[(long)gm * K + gk] : 0.f;
    }
    __syncthreads();
    if (do_db && tid < BN) {
#pragma unroll
      for (int m = 0; m < BK; ++m) db_acc += sa[tid][m];
    }
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const float a0 = sa[wr + fi][kk + fk];
      const float a1 = sa[wr + 16 + fi][kk + fk];
      const float b0 = sb[kk + fk][wc + fi];
      const float b1 = sb[kk + fk][wc + 16 + fi];
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }
  if (do_db && tid < BN && n0 + tid < N) db[n0 + tid] = db_acc;
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
        if (row < N && col < K) dw[(long)row * K + col] = a[r];
      }
    }
}

// ---------------------------------------------------------------------------
// K4: fused tanh-squashed Gaussian sample + log-prob (fwd + bwd).
// One thread per batch row; A = action_dim <= 32.
//   ls = clamp(lsr, -20, 2); s = exp(ls); u = mu + s*eps; t = tanh(u)
//   a = k * t
//   logp = sum_i [ -0.5 eps_i^2 - ls_i - 0.5 log(2pi)
//                  - log(k (1 - t_i^2 + 1e-6)) ]
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_squash_fwd(
    const float* __restrict__ mu, const float* __restrict__ lsr,
    const float* __restrict__ eps, float* __restrict__ act,
    float* __restrict__ logp, float* __restrict__ tanh_u,
    float* __restrict__ ls_out, int B, int A, float k) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  constexpr float C = 0.9189385332046727f;  // 0.5*log(2*pi)
  float lp = 0.f;
  for (int a = 0; a < A; ++a) {
    const long idx = (long)i * A + a;
    const float ls = fminf(fmaxf(lsr[idx], -20.f), 2.f);
    const float s = __expf(ls);
    const float e = eps[idx];
    const float u = mu[idx] + s * e;
    const float t = tanhf(u);
    act[idx] = k * t;
    tanh_u[idx] = t;
    ls_out[idx] = ls;
    lp += -0.5f * e * e - ls - C - __logf(k * (1.f - t * t + 1e-6f));
  }
  logp[i] = lp;
}

// bwd: dmu_i = ga_i*k*(1-t^2) + gl*2t(1-t^2)/(1-t^2+1e-6)
//      dlsr_i = mask * [ ga_i*k*(1-t^2)*e*s
//                        + gl*(-1 + 2t(1-t^2)/(1-t^2+1e-6)*e*s) ]
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
    const float gai = ga[idx];
    const float du = gai * k * omt2 + g * dlp_du;   // dL/du
    dmu[idx] = du;
    const float raw = lsr[idx];
    const float mask = (raw >= -20.f && raw <= 2.f) ? 1.f : 0.f;
    dlsr[idx] = mask * (du * e * s - g);
  }
}

// ---------------------------------------------------------------------------
// K5: y = rs*r + gamma*(1-d)*(min(q1,q2) - alpha*lp)
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
// k_adam_prolog: step += 1; coeffs = {step_size, inv_sqrt_bc2}.
__global__ void k_adam_prolog(float* __restrict__ state, float lr, float b1,
                              float b2) {
  const float step = state[0] + 1.f;
  state[0] = step;
  state[1] = lr / (1.f - __powf(b1, step));          // step_size
  state[2] = 1.f / sqrtf(1.f - __powf(b2, step));    // inv_sqrt_bc2
}

__global__ __launch_bounds__(256) void k_adam_dev(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ state, long n, float b1, float b2, float eps) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float step_size = state[1], inv_sqrt_bc2 = state[2];
  const float gi = g[i];
  const float mi = b1 * m[i] + (1.f - b1) * gi;
  const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
  m[i] = mi;
  v[i] = vi;
  p[i] -= step_size * mi / (sqrtf(vi) * inv_sqrt_bc2 + eps);
}

// ---------------------------------------------------------------------------
// K10: t = (1-tau)*t + tau*s over flat buffers.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_polyak(
    float* __restrict__ t, const float* __restrict__ s, long n, float tau) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  t[i] = (1.f - tau) * t[i] + tau * s[i];
}

// ===========================================================================
// Host wrappers
// ===========================================================================

static torch::Tensor linear_act_fwd(torch::Tensor x, torch::Tensor w,
                                    torch::Tensor b, long act) {
  CHECK_IN(x); CHECK_IN(w); CHECK_IN(b);
  auto xc = x.contiguous(); auto wc = w.contiguous(); auto bc = b.contiguous();
  const long M = xc.size(0), K = xc.size(1), N = wc.size(0);
  TORCH_CHECK(wc.size(1) == K, "weight shape mismatch");
  auto y = torch::empty({M, N}, xc.options());
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  hipLaunchKernelGGL(k_linear_act_fwd, grid, dim3(256), 0, cur_stream(),
                     xc.data_ptr<float>(), wc.data_ptr<float>(),
                     bc.data_ptr<float>(), y.data_ptr<float>(),
                     (int)M, (int)N, (int)K, (int)act);
  return y;
}

static torch::Tensor linear_bwd_dx(torch::Tensor dy, torch::Tensor w,
                                   torch::Tensor yout, long act) {
  CHECK_IN(dy); CHECK_IN(w); CHECK_IN(yout);
  auto dyc = dy.contiguous(); auto wc = w.contiguous();
  auto yc = yout.contiguous();
  const long M = dyc.size(0), N = dyc.size(1), K = wc.size(1);
  TORCH_CHECK(wc.size(0) == N, "weight shape mismatch");
  auto dx = torch::empty({M, K}, dyc.options());
  dim3 grid((M + BM - 1) / BM, (K + BN - 1) / BN);
  hipLaunchKernelGGL(k_linear_bwd_dx, grid, dim3(256), 0, cur_stream(),
                     dyc.data_ptr<float>(), wc.data_ptr<float>(),
                     yc.data_ptr<float>(), dx.data_ptr<float>(),
                     (int)M, (int)N, (int)K, (int)act);
  return dx;
}

static std::vector<torch::Tensor> linear_bwd_dwdb(torch::Tensor dy,
                                                  torch::Tensor x,
                                                  torch::Tensor yout,
                                                  long act) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(yout);
  auto dyc = dy.contiguous(); auto xc = x.contiguous();
  auto yc = yout.contiguous();
  const long M = dyc.size(0), N = dyc.size(1), K = xc.size(1);
  auto dw = torch::empty({N, K}, dyc.options());
  auto db = torch::empty({N}, dyc.options());
  dim3 grid((N + BM - 1) / BM, (K + BN - 1) / BN);
  hipLaunchKernelGGL(k_linear_bwd_dwdb, grid, dim3(256), 0, cur_stream(),
                     dyc.data_ptr<float>(), xc.data_ptr<float>(),
                     yc.data_ptr<float>(), dw.data_ptr<float>(),
                     db.data_ptr<float>(), (int)M, (int)N, (int)K, (int)act);
  return {dw, db};
}

static std::vector<torch::Tensor> squashed_gaussian_fwd(torch::Tensor mu,
                                                        torch::Tensor lsr,
                                                        torch::Tensor eps,
                                                        double k) {
  CHECK_IN(mu); CHECK_IN(lsr); CHECK_IN(eps);
  auto muc = mu.contiguous(); auto lc = lsr.contiguous();
  auto ec = eps.contiguous();
  const long B = muc.size(0), A = muc.size(1);
  TORCH_CHECK(A <= 32, "action_dim too large for fused kernel");
  auto act = torch::empty_like(muc);
  auto logp = torch::empty({B, 1}, muc.options());
  auto tanh_u = torch::empty_like(muc);
  auto ls_out = torch::empty_like(muc);
  const int grid = (B + 255) / 256;
  hipLaunchKernelGGL(k_squash_fwd, dim3(grid), dim3(256), 0, cur_stream(),
                     muc.data_ptr<float>(), lc.data_ptr<float>(),
                     ec.data_ptr<float>(), act.data_ptr<float>(),
                     logp.data_ptr<float>(), tanh_u.data_ptr<float>(),
                     ls_out.data_ptr<float>(), (int)B, (int)A, (float)k);
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
                           double b1, double b2, double eps) {
  CHECK_IN(p); CHECK_IN(state);
  TORCH_CHECK(state.numel() >= 3, "state = {step, step_size, inv_sqrt_bc2}");
  const long n = p.numel();
  hipLaunchKernelGGL(k_adam_prolog, dim3(1), dim3(1), 0, cur_stream(),
                     state.data_ptr<float>(), (float)lr, (float)b1,
                     (float)b2);
  hipLaunchKernelGGL(k_adam_dev, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     state.data_ptr<float>(), n, (float)b1, (float)b2,
                     (float)eps);
}

static void polyak_(torch::Tensor t, torch::Tensor s, double tau) {
  CHECK_IN(t);
  const long n = t.numel();
  TORCH_CHECK(s.numel() == n, "polyak buffers must match");
  hipLaunchKernelGGL(k_polyak, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), t.data_ptr<float>(), s.data_ptr<float>(),
                     n, (float)tau);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("linear_act_fwd", &linear_act_fwd, "fused GEMM+bias+act forward");
  mod.def("linear_bwd_dx", &linear_bwd_dx, "GEMM backward dX (fused mask)");
  mod.def("linear_bwd_dwdb", &linear_bwd_dwdb, "GEMM backward dW+db");
  mod.def("squashed_gaussian_fwd", &squashed_gaussian_fwd);
  mod.def("squashed_gaussian_bwd", &squashed_gaussian_bwd);
  mod.def("td_target", &td_target);
  mod.def("adam_step_", &adam_step_);
  mod.def("adam_step_dev_", &adam_step_dev_);
  mod.def("polyak_", &polyak_);
}
