// chain_gemm.hip — fused wide MLP-chain kernels for the SAC hot path
// (gfx950).  Round-2 addition.
//
// Motivation (profiles/r16_round2_call1-3.md): every per-layer GEMM
// launch costs ~10-17 us at these tiny shapes REGARDLESS of FLOPs — the
// whole family is launch/latency-bound (e.g. k_bf16_fwd on the twin head
// M1280 N1 G2 runs 0.1 GFLOP/s).  An MLP forward is row-parallel, so one
// workgroup can own a batch row-slice and run the ENTIRE chain:
// activations stay in LDS, weights stream from L2 (every WG reads the
// same ~320 KB per 400x400 layer), and the 4-6 launches per chain
// collapse to ONE.  The first layer reads the raw fp32 inputs (optionally
// two tensors column-concatenated, e.g. [states || actions]) and converts
// to bf16 in the loader — this also eliminates the separate torch::cat +
// bf16-cast launches (VERDICT round-1 item 5).
//
// Geometry: 512 threads = 8 waves (2/SIMD so loads of one wave overlap
// MFMAs of the other); each wave computes 16x16 MFMA output tiles
// (v_mfma_f32_16x16x32_bf16) in QUADS of column tiles x RM row-fragments
// — 4*RM independent accumulator chains keep 4 B-fragment loads in
// flight per k-step, and the k-loop is a manual unroll-2 software
// pipeline (set P computes while set Q loads) so the loads span a full
// MFMA step.  RM=2 (32 rows/WG) halves the total L2 weight re-read
// (traffic = M/TM * layer bytes) at the cost of half the workgroups —
// the host picks RM by M.  A-fragments come from the LDS activation
// buffer: row stride 776 u16 = 388 dwords = 4 banks mod 64, so the
// 16-lane b128 read groups touch all 64 banks exactly once —
// conflict-free without padding tricks (see cdna_hip_programming.md §2).
//
// grid (ceil(M/TM), 1, G).  G>1 = twin critics: shared x, per-group
// weights/biases/outputs (same convention as k_bf16_fwd).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace chain {

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using u16 = unsigned short;

static constexpr int CMAX = 768;  // max layer width / input width
                                  // (768 = the RoBERTa context width)
static constexpr int CPAD = 776;  // LDS row stride (u16): 388 dwords ≡ 4 mod 64
                                  // -> 16-lane b128 A-fragment groups hit
                                  // all 64 banks exactly once
static constexpr int NTHR = 512;  // 8 waves

#define CHAIN_CHECK_BF16(t) \
  TORCH_CHECK((t).is_cuda() && (t).scalar_type() == torch::kBFloat16, \
              #t " must be a bf16 HIP tensor")
#define CHAIN_CHECK_F32(t) \
  TORCH_CHECK((t).is_cuda() && (t).scalar_type() == torch::kFloat32, \
              #t " must be a fp32 HIP tensor")

static inline hipStream_t cur_stream3() {
  return c10::hip::getCurrentHIPStream().stream();
}

__device__ __forceinline__ u16 f32_to_bf16_rne3(float f) {
  union { float f; unsigned u; } v{f};
  unsigned u = v.u;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return (u16)(u >> 16);
}

struct ChainFwdDesc {
  const u16* w[6];    // [G, N_l, K_l] bf16 (k-minor)
  const u16* wp[6];   // fragment-PACKED mirror (k_bf16_pack_frag) or null
  const float* b[6];  // [G, N_l]
  u16* acts[6];       // post-act output of layer l (null = don't save)
  int N[6];
  int act[6];         // 1 = ReLU
  int L;
};

// B-operand fragment: 8 consecutive bf16 of one weight row from GLOBAL.
__device__ __forceinline__ bf16x8 load_bfrag(const u16* __restrict__ w,
                                             int row, int N, int K,
                                             int k8) {
  if (row < N) {
    const long base = (long)row * K + k8;
    if (k8 + 8 <= K && (base & 7) == 0)
      return *(const bf16x8*)&w[base];
    bf16x8 v{};
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (k8 + j < K) v[j] = ((const __bf16*)w)[base + j];
    return v;
  }
  return bf16x8{};
}

template <int RM>
__global__ __launch_bounds__(NTHR) void k_bf16_chain_fwd(
    const void* __restrict__ x1, const void* __restrict__ x2,
    int C1, int C2, int x1f, int x2f, int rowcat, int M1,
    u16* __restrict__ xsave,
    ChainFwdDesc d, void* __restrict__ y, int M, int out_f32) {
  constexpr int TMv = 16 * RM;
  __shared__ u16 sa[2][TMv][CPAD];
  const int g = blockIdx.z;
  const int m0 = blockIdx.x * TMv;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int fi = lane & 15, fk = lane >> 4;
  const int rowlim = (M - m0 < TMv ? M - m0 : TMv);
  const int K0 = rowcat ? C1 : C1 + C2;

  // zero both activation buffers (pads must read 0 in the MFMA A-frags)
  {
    unsigned* p = (unsigned*)sa;
    for (int i = tid; i < 2 * TMv * CPAD / 2; i += NTHR) p[i] = 0u;
  }
  __syncthreads();

  // ---- input tile -> sa[0] (converted to bf16) --------------------
  // rowcat: x = [x1 ; x2] stacked over rows (equal widths C1);
  // else:   x = [x1 || x2] concatenated over columns (C1 + C2).
  {
    const int r = tid >> 5;      // 32 threads per row covers 16 rows/pass
    const int lc = tid & 31;
    for (int rr = r; rr < rowlim; rr += NTHR / 32) {
      const long row = m0 + rr;
      for (int c = lc; c < K0; c += 32) {
        float v;
        if (rowcat) {
          const bool top = row < M1;
          const long rr2 = top ? row : row - M1;
          const int f_ = top ? x1f : x2f;
          v = f_ ? ((const float*)(top ? x1 : x2))[rr2 * C1 + c]
                 : (float)((const __bf16*)(top ? x1 : x2))[rr2 * C1 + c];
        } else if (c < C1) {
          v = x1f ? ((const float*)x1)[row * C1 + c]
                  : (float)((const __bf16*)x1)[row * C1 + c];
        } else {
          v = x2f ? ((const float*)x2)[row * C2 + (c - C1)]
                  : (float)((const __bf16*)x2)[row * C2 + (c - C1)];
        }
        sa[0][rr][c] = f32_to_bf16_rne3(v);
      }
    }
  }
  __syncthreads();
  // coalesced bf16 input save (backward consumes it): 16-B chunks from
  // LDS instead of the loader's scalar stores (TA relief)
  if (xsave != nullptr && g == 0) {
    const int r = tid >> 5;
    const int lc = tid & 31;
    for (int rr = r; rr < rowlim; rr += NTHR / 32) {
      u16* dst = xsave + (long)(m0 + rr) * K0;
      const u16* src = &sa[0][rr][0];
      for (int c = lc * 8; c + 8 <= K0; c += (NTHR / 32) * 8)
        *(uint4*)&dst[c] = *(const uint4*)&src[c];
    }
    if ((K0 & 7) && tid < TMv && tid < rowlim) {
      for (int c = K0 & ~7; c < K0; ++c)
        xsave[(long)(m0 + tid) * K0 + c] = sa[0][tid][c];
    }
  }

  int cur = 0;
  int prevw0 = K0, prevw1 = CMAX;  // written widths of buffers 0/1
  int K = K0;
  for (int li = 0; li < d.L; ++li) {
    const int N = d.N[li];
    const bool last = li == d.L - 1;
    const int relu = d.act[li];
    const u16* wg = d.w[li] + (long)g * N * K;
    const float* bb = d.b[li] + (long)g * N;
    u16* ag = (!last && d.acts[li] != nullptr)
                  ? d.acts[li] + (long)g * M * N : nullptr;
    float* yf = (float*)y + (long)g * M * N;
    u16* yh = (u16*)y + (long)g * M * N;
    const int ntiles = (N + 15) >> 4;
    const int nb = cur ^ 1;

    // K-body where every lane's 8-run is in-range AND 16B-aligned:
    // requires K % 8 == 0 for row alignment; the tail (K % 32) and
    // odd-K layers (layer 0's 49/53-d inputs) take the guarded path.
    const bool k_aligned = (K % 8) == 0;
    const int kbody = k_aligned ? (K & ~31) : 0;

    for (int t0 = wid * 4; t0 < ntiles; t0 += 4 * (NTHR / 64)) {
      const int nq = (ntiles - t0 < 4) ? (ntiles - t0) : 4;
      f32x4 acc[RM][4];
#pragma unroll
      for (int m = 0; m < RM; ++m)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[m][j] = f32x4{};
      // Per-lane CLAMPED row bases: surplus quad slots (j >= nq) and
      // out-of-range rows of a partial-N tile load a VALID row instead
      // of branching — their MFMA lanes produce garbage that the
      // epilogue's col<N / q<nq masks never store, and out-of-range K
      // contributions are killed by the zero-padded LDS A operand.
      // This keeps EVERY K%8==0 layer (including the N=1/N=8 heads and
      // the 25th tile of a 400-wide layer) on the branch-free pipelined
      // path — the guarded loop (a per-load branch + vmcnt drain) made
      // one straggler wave serialize the whole chain at the layer
      // barrier.
      long rb[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = (t0 + (j < nq ? j : nq - 1)) * 16 + fi;
        if (row >= N) row = N - 1;
        rb[j] = (long)row * K;
      }

#define LOADQ(dst, kk)                                                    \
      _Pragma("unroll") for (int j = 0; j < 4; ++j)                       \
        dst[j] = *(const bf16x8*)&wg[rb[j] + fk * 8 + (kk)];
#define MF4(bset, kk)                                                     \
      {                                                                   \
        _Pragma("unroll") for (int m = 0; m < RM; ++m) {                  \
          const bf16x8 a_ =                                               \
              *(const bf16x8*)&sa[cur][m * 16 + fi][(kk) + fk * 8];       \
          _Pragma("unroll") for (int j = 0; j < 4; ++j)                   \
            acc[m][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(          \
                a_, bset[j], acc[m][j], 0, 0, 0);                         \
        }                                                                 \
      }

      if (d.wp[li] != nullptr) {
        // FRAGMENT-PACKED path: every load is wp + base + ks*512 +
        // lane*8 — one contiguous 1 KiB line per wave (TA cost 1/instr
        // instead of 16), zero-padding baked in (no guards, covers odd
        // K and partial-N tiles uniformly).
        const int KS = (K + 31) >> 5;
        const int NTp = (N + 15) >> 4;
        const u16* wpl = d.wp[li];
        long bp[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int tt = t0 + (j < (nq > 0 ? nq : 1) ? j
                         : (nq > 0 ? nq - 1 : 0));
          if (tt >= NTp) tt = NTp - 1;
          if (tt < 0) tt = 0;
          bp[j] = (((long)g * NTp + tt) * KS) * 512 + lane * 8;
        }
#define LOADP(dst, kk)                                                    \
        _Pragma("unroll") for (int j = 0; j < 4; ++j)                     \
          dst[j] = *(const bf16x8*)&wpl[bp[j] + (long)(kk) * 16];
        // (kk is the k element offset; one 32-k step = 512 u16 = kk*16
        //  since kk advances by 32)
        {
          bf16x8 S0[4], S1[4], S2[4];
          const int pbody = KS * 32;
          LOADP(S0, 0)
          if (32 < pbody) LOADP(S1, 32)
          int k = 0;
          while (k + 96 <= pbody) {
            LOADP(S2, k + 64)
            MF4(S0, k)
            if (k + 96 < pbody) {
              LOADP(S0, k + 96)
            }
            MF4(S1, k + 32)
            if (k + 128 < pbody) {
              LOADP(S1, k + 128)
            }
            MF4(S2, k + 64)
            k += 96;
          }
          if (pbody - k == 32) {
            MF4(S0, k)
          } else if (pbody - k == 64) {
            MF4(S0, k)
            MF4(S1, k + 32)
          }
        }
#undef LOADP
      } else if (k_aligned) {
        // unconditional loads, 3-set rotating software pipeline: loads
        // run TWO k-steps ahead of their MFMAs (12 B-fragments in
        // flight), hiding cold-L2/L3 weight-miss latency that a 1-deep
        // pipeline left exposed.
        if (kbody >= 32) {
          bf16x8 S0[4], S1[4], S2[4];
          LOADQ(S0, 0)
          if (32 < kbody) LOADQ(S1, 32)
          int k = 0;
          while (k + 96 <= kbody) {
            LOADQ(S2, k + 64)
            MF4(S0, k)
            if (k + 96 < kbody) {
              LOADQ(S0, k + 96)
            }
            MF4(S1, k + 32)
            if (k + 128 < kbody) {
              LOADQ(S1, k + 128)
            }
            MF4(S2, k + 64)
            k += 96;
          }
          if (kbody - k == 32) {
            MF4(S0, k)
          } else if (kbody - k == 64) {
            MF4(S0, k)
            MF4(S1, k + 32)
          }
        }
        if (kbody < K) {
          // K%32 tail: clamp each lane's k-run into range — lanes whose
          // true k-run starts past K read duplicate (valid) bytes that
          // multiply the zero LDS pad.
          const int k8 = kbody + fk * 8;
          const int k8c = (k8 + 8 <= K) ? k8 : (K - 8);
          bf16x8 B[4];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            B[j] = *(const bf16x8*)&wg[rb[j] + k8c];
          MF4(B, kbody)
        }
      } else {
        // odd-K layer (the 49/53-d layer-0 inputs): guarded loads
        for (int k = 0; k < K; k += 32) {
          const int k8 = k + fk * 8;
          bf16x8 B[4];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            B[j] = (j < nq) ? load_bfrag(wg, t0 * 16 + j * 16 + fi, N, K, k8)
                            : bf16x8{};
          MF4(B, k)
        }
      }
#undef LOADQ
#undef MF4

      // epilogue: C/D map col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
      for (int m = 0; m < RM; ++m) {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          if (q >= nq) break;
          const f32x4 a = acc[m][q];
          const int col = (t0 + q) * 16 + fi;
          if (col < N) {
            const float bias = bb[col];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int row = m * 16 + fk * 4 + r;
              float v = a[r] + bias;
              if (relu) v = fmaxf(v, 0.f);
              if (last) {
                if (row < rowlim) {
                  if (out_f32) yf[(long)(m0 + row) * N + col] = v;
                  else yh[(long)(m0 + row) * N + col] = f32_to_bf16_rne3(v);
                }
              } else {
                const u16 h = f32_to_bf16_rne3(v);
                sa[nb][row][col] = h;
                if (ag != nullptr && row < rowlim)
                  ag[(long)(m0 + row) * N + col] = h;
              }
            }
          }
        }
      }
    }
    if (!last) {
      // zero the stale tail of the next buffer: cols [N, prevw(nb))
      const int pw = nb ? prevw1 : prevw0;
      for (int i = tid; i < TMv * (pw > N ? pw - N : 0); i += NTHR) {
        const int r = i / (pw - N), c = N + i % (pw - N);
        sa[nb][r][c] = 0;
      }
      if (nb) prevw1 = N; else prevw0 = N;
      __syncthreads();
      // coalesced activation save from LDS (the epilogue only wrote
      // LDS; a 16-B-chunk copy replaces 4x as many scalar 2-B stores)
      if (ag != nullptr) {
        const int r2 = tid >> 5;
        const int lc2 = tid & 31;
        for (int rr = r2; rr < rowlim; rr += NTHR / 32) {
          u16* dst = ag + (long)(m0 + rr) * N;
          const u16* src = &sa[nb][rr][0];
          for (int c = lc2 * 8; c + 8 <= N; c += (NTHR / 32) * 8)
            *(uint4*)&dst[c] = *(const uint4*)&src[c];
        }
        if ((N & 7) && tid < TMv && tid < rowlim) {
          for (int c = N & ~7; c < N; ++c)
            ag[(long)(m0 + tid) * N + c] = sa[nb][tid][c];
        }
      }
      cur = nb;
      K = N;
    }
  }
}

// ---------------------------------------------------------------------------
// FRAGMENT-PACKED weight mirrors (round 2, the TA fix): the chain
// kernels' B-operand loads are fragment-shaped — lane l of a wave reads
// 16 B of row tile*16+(l&15) at k-offset (l>>4)*8, i.e. 16 scattered
// cache lines per instruction, which kept GRBM_TA_BUSY at 66% of the
// kernel.  This kernel re-packs a weight matrix ONCE per Adam step into
// exactly that per-lane layout:
//     fwd:  P[((g*NT + t)*KS + ks)*64 + l][0..8) = W[g][t*16 + (l&15)]
//                                           [ks*32 + (l>>4)*8 + j]
//     dx:   same with W TRANSPOSED (rows = k, cols = n)
// (zero-filled out of range) so every chain load is base + lane*16 —
// one contiguous 1 KiB line per wave, and the odd-K / partial-tile
// guard paths disappear.  One launch packs up to 24 matrices.
// ---------------------------------------------------------------------------
struct PackDesc {
  const u16* w[32];
  u16* p[32];
  int N[32], K[32], G[32], dx[32];
  int cum[33];   // cumulative output chunks (G*NT*KS*64 per matrix)
  int C;
};

__global__ __launch_bounds__(256) void k_bf16_pack_frag(PackDesc d) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  int c = 0;
  while (c + 1 < d.C && i >= d.cum[c + 1]) ++c;
  if (i >= d.cum[d.C]) return;
  long local = i - d.cum[c];
  const int lane = (int)(local & 63);
  local >>= 6;
  const int N = d.N[c], K = d.K[c];
  const bool dx = d.dx[c] != 0;
  // logical tile dims of the PACKED operand
  const int rows = dx ? K : N;      // fragment row dimension
  const int cols = dx ? N : K;      // fragment k dimension
  const int KS = (cols + 31) >> 5;
  const int NT = (rows + 15) >> 4;
  const int ks = (int)(local % KS);
  local /= KS;
  const int t = (int)(local % NT);
  const long g = local / NT;
  const int row = t * 16 + (lane & 15);
  const int k0 = ks * 32 + (lane >> 4) * 8;
  const u16* w = d.w[c] + g * (long)N * K;
  u16* out = d.p[c] + (i - d.cum[c]) * 8;
  u16 v[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int kk = k0 + j;
    if (row < rows && kk < cols)
      v[j] = dx ? w[(long)kk * K + row]    // W^T: [k-row=row][n=kk] -> W[kk][row]
                : w[(long)row * K + kk];
    else
      v[j] = 0;
  }
  *(uint4*)out = *(uint4*)v;
}

static void pack_weights_frag(std::vector<torch::Tensor> ws,
                              std::vector<torch::Tensor> ps,
                              std::vector<long> Gs,
                              std::vector<long> dxs) {
  const int C = (int)ws.size();
  TORCH_CHECK(C >= 1 && C <= 32 && (int)ps.size() == C
              && (int)Gs.size() == C && (int)dxs.size() == C);
  PackDesc d{};
  d.C = C;
  std::vector<torch::Tensor> keep;
  long cum = 0;
  for (int i = 0; i < C; ++i) {
    CHAIN_CHECK_BF16(ws[i]);
    CHAIN_CHECK_BF16(ps[i]);
    auto wc = ws[i].contiguous();
    keep.push_back(wc);
    TORCH_CHECK(ps[i].is_contiguous());
    const long G = Gs[i];
    const long K = wc.size(-1);
    const long N = wc.numel() / (G * K);
    const long rows = dxs[i] ? K : N;
    const long cols = dxs[i] ? N : K;
    const long KS = (cols + 31) / 32, NT = (rows + 15) / 16;
    TORCH_CHECK(ps[i].numel() == G * NT * KS * 512,
                "packed buffer size mismatch at ", i);
    d.w[i] = (const u16*)wc.data_ptr();
    d.p[i] = (u16*)ps[i].data_ptr();
    d.N[i] = (int)N;
    d.K[i] = (int)K;
    d.G[i] = (int)G;
    d.dx[i] = (int)dxs[i];
    d.cum[i] = (int)cum;
    cum += G * NT * KS * 64;
  }
  d.cum[C] = (int)cum;
  hipLaunchKernelGGL(k_bf16_pack_frag, dim3((cum + 255) / 256), dim3(256),
                     0, cur_stream3(), d);
}

// ---------------------------------------------------------------------------
// Multi-tensor 64x64 tiled weight transpose: wt[g][k][n] = w[g][n][k]
// (bf16).  ONE launch re-materializes every transposed weight mirror the
// fused dx-chain consumes — the mirrors change every Adam step, and the
// dx GEMM's B-fragment needs n-minor (k-run over n) layout; reading W^T
// from global with the fwd chain's fast contiguous loader beats per-tile
// LDS transpose staging inside the chain (which re-stages W per WG).
// ---------------------------------------------------------------------------
struct TransDesc {
  const u16* w[8];
  u16* wt[8];
  int N[8], K[8];
  int cum[9];   // cumulative tiles per layer (tiles = G*ceil(N/64)*ceil(K/64))
  int ntx[8];   // ceil(N/64)
  int nty[8];   // ceil(K/64)
  int Gl[8];    // per-tensor group count
  int L;
};

__global__ __launch_bounds__(256) void k_bf16_transpose_multi(TransDesc d) {
  __shared__ u16 t[64][72];   // +8 u16 pad: conflict-free column reads
  int bx = blockIdx.x;
  int l = 0;
  while (l + 1 < d.L && bx >= d.cum[l + 1]) ++l;
  int local = bx - d.cum[l];
  const int per_g = d.ntx[l] * d.nty[l];
  const int g = local / per_g;
  local -= g * per_g;
  (void)d.Gl;
  const int n0 = (local % d.ntx[l]) * 64;
  const int k0 = (local / d.ntx[l]) * 64;
  const int N = d.N[l], K = d.K[l];
  const u16* w = d.w[l] + (long)g * N * K;
  u16* wt = d.wt[l] + (long)g * N * K;
  const int tid = threadIdx.x;
  const int r = tid >> 3, c0 = (tid & 7) * 8;   // 32 rows/pass, 8 cols/thread
#pragma unroll
  for (int rr = 0; rr < 64; rr += 32) {
    const int row = n0 + r + rr;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = k0 + c0 + j;
      t[r + rr][c0 + j] = (row < N && col < K)
          ? w[(long)row * K + col] : (u16)0;
    }
  }
  __syncthreads();
#pragma unroll
  for (int rr = 0; rr < 64; rr += 32) {
    const int row = k0 + r + rr;        // wt row = k
    if (row < K) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int col = n0 + c0 + j;    // wt col = n
        if (col < N) wt[(long)row * N + col] = t[c0 + j][r + rr];
      }
    }
  }
}

static void transpose_weights_bf16(std::vector<torch::Tensor> ws,
                                   std::vector<torch::Tensor> wts,
                                   std::vector<long> Gs) {
  const int L = (int)ws.size();
  TORCH_CHECK(L >= 1 && L <= 8 && (int)wts.size() == L
              && (int)Gs.size() == L);
  TransDesc d{};
  d.L = L;
  int cum = 0;
  std::vector<torch::Tensor> keep;
  for (int i = 0; i < L; ++i) {
    CHAIN_CHECK_BF16(ws[i]);
    CHAIN_CHECK_BF16(wts[i]);
    auto wc = ws[i].contiguous();
    keep.push_back(wc);
    TORCH_CHECK(wts[i].is_contiguous());
    const long G = Gs[i];
    const long NK = wc.numel() / G;
    const long K = wc.size(-1);
    const long N = NK / K;
    TORCH_CHECK(wts[i].numel() == wc.numel());
    d.w[i] = (const u16*)wc.data_ptr();
    d.wt[i] = (u16*)wts[i].data_ptr();
    d.N[i] = (int)N;
    d.K[i] = (int)K;
    d.Gl[i] = (int)G;
    d.ntx[i] = (int)((N + 63) / 64);
    d.nty[i] = (int)((K + 63) / 64);
    d.cum[i] = cum;
    cum += (int)G * d.ntx[i] * d.nty[i];
  }
  d.cum[L] = cum;
  hipLaunchKernelGGL(k_bf16_transpose_multi, dim3(cum), dim3(256), 0,
                     cur_stream3(), d);
}

// ---------------------------------------------------------------------------
// Fused MLP-chain BACKWARD-dx: walk the chain top-down, keeping dy in
// LDS; per layer (a) mask dy by the saved post-ReLU activation and write
// the masked dy to global (the grouped dwdb launch consumes it with no
// re-masking), (b) dy_{l-1} = dy_masked @ W_l via the TRANSPOSED weight
// mirror (same branch-free quad/pipelined loader as the forward chain).
// Optionally emits the chain-input gradient restricted to columns
// >= dx0_lo (the SAC actor step needs only the ACTION columns of
// d(critic input)) as fp32 per group.  grid (ceil(M/16), 1, G).
// ---------------------------------------------------------------------------
struct ChainDxDesc {
  const u16* wp[6];     // fragment-PACKED W^T mirror (pack dx mode) or null
  const u16* wt[6];     // [G, K_l, N_l] n-minor transposed weights
  const u16* yout[6];   // post-act output of layer l (mask); null if act=0
  long yo_gs[6];        // per-group element stride of yout
  u16* dysave[6];       // masked dy_l [G, M, N_l] (null = skip)
  int N[6];
  int act[6];
  int L;
  int K0;
  int dx0_lo;           // -1 = no input gradient
};

__global__ __launch_bounds__(NTHR) void k_bf16_chain_dx(
    const u16* __restrict__ dy_last, ChainDxDesc d,
    float* __restrict__ dx0, int M) {
  constexpr int TMv = 16;
  __shared__ u16 sd[2][TMv][CPAD];
  const int g = blockIdx.z;
  const int m0 = blockIdx.x * TMv;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wid = tid >> 6;
  const int fi = lane & 15, fk = lane >> 4;
  const int rowlim = (M - m0 < TMv ? M - m0 : TMv);

  {
    unsigned* p = (unsigned*)sd;
    for (int i = tid; i < 2 * TMv * CPAD / 2; i += NTHR) p[i] = 0u;
  }
  __syncthreads();

  // load dy_last tile
  const int NL = d.N[d.L - 1];
  {
    const int r = tid >> 5;
    const int lc = tid & 31;
    const u16* src = dy_last + (long)g * M * NL;
    for (int rr = r; rr < rowlim; rr += NTHR / 32)
      for (int c = lc; c < NL; c += 32)
        sd[0][rr][c] = src[(long)(m0 + rr) * NL + c];
  }
  __syncthreads();

  int cur = 0;
  int prevw0 = NL, prevw1 = CMAX;
  for (int li = d.L - 1; li >= 0; --li) {
    const int N = d.N[li];
    const int K = li > 0 ? d.N[li - 1] : d.K0;

    // ---- mask by relu' + save masked dy --------------------------------
    // only the ENTRY layer needs this standalone pass: deeper layers'
    // masks are fused into the producing GEMM's epilogue below
    if (li == d.L - 1) {
      const int r = tid >> 5;
      const int lc = tid & 31;
      const u16* yo = d.act[li] && d.yout[li] != nullptr
          ? d.yout[li] + (long)g * d.yo_gs[li] : nullptr;
      u16* sv = d.dysave[li] != nullptr
          ? d.dysave[li] + (long)g * M * N : nullptr;
      for (int rr = r; rr < rowlim; rr += NTHR / 32) {
        const long row = m0 + rr;
        for (int c = lc; c < N; c += 32) {
          u16 v = sd[cur][rr][c];
          if (yo != nullptr && yo[row * N + c] == 0) v = 0;
          sd[cur][rr][c] = v;
          if (sv != nullptr) sv[row * N + c] = v;
        }
      }
      __syncthreads();
    }

    const bool want_dx0 = li == 0 && d.dx0_lo >= 0;
    if (li == 0 && !want_dx0) break;

    // ---- GEMM: out[16, K] = dy_pre[16, N] @ Wt[K, N] -------------------
    const u16* wtg = d.wt[li] + (long)g * N * K;   // [K][N]
    const int ln = li - 1;
    const u16* yo_n = (!want_dx0 && ln >= 0 && d.act[ln]
                       && d.yout[ln] != nullptr)
        ? d.yout[ln] + (long)g * d.yo_gs[ln] : nullptr;
    u16* sv_n = (!want_dx0 && ln >= 0 && d.dysave[ln] != nullptr)
        ? d.dysave[ln] + (long)g * M * K : nullptr;
    const int col_lo = want_dx0 ? d.dx0_lo : 0;
    const int t_lo = col_lo >> 4;
    const int ntiles = (K + 15) >> 4;
    const int nb = cur ^ 1;
    const bool n_aligned = (N % 8) == 0;
    const int nbody = n_aligned ? (N & ~31) : 0;
    float* dx0g = want_dx0
        ? dx0 + (long)g * M * (K - col_lo) : nullptr;

    for (int t0 = t_lo + wid * 4; t0 < ntiles; t0 += 4 * (NTHR / 64)) {
      const int nq = (ntiles - t0 < 4) ? (ntiles - t0) : 4;
      f32x4 acc[1][4];
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[0][j] = f32x4{};
      long rb[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = (t0 + (j < nq ? j : nq - 1)) * 16 + fi;   // k index
        if (row >= K) row = K - 1;
        rb[j] = (long)row * N;
      }

#define LOADQ(dst, kk)                                                    \
      _Pragma("unroll") for (int j = 0; j < 4; ++j)                       \
        dst[j] = *(const bf16x8*)&wtg[rb[j] + fk * 8 + (kk)];
#define MF4(bset, kk)                                                     \
      {                                                                   \
        const bf16x8 a_ =                                                 \
            *(const bf16x8*)&sd[cur][fi][(kk) + fk * 8];                  \
        _Pragma("unroll") for (int j = 0; j < 4; ++j)                     \
          acc[0][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(            \
              a_, bset[j], acc[0][j], 0, 0, 0);                           \
      }

      if (d.wp[li] != nullptr) {
        // fragment-packed W^T: unit-stride loads, guards baked in
        const int KSp = (N + 31) >> 5;
        const int NTp = (K + 15) >> 4;
        const u16* wpl = d.wp[li];
        long bp[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int tt = t0 + (j < (nq > 0 ? nq : 1) ? j
                         : (nq > 0 ? nq - 1 : 0));
          if (tt >= NTp) tt = NTp - 1;
          if (tt < 0) tt = 0;
          bp[j] = (((long)g * NTp + tt) * KSp) * 512 + lane * 8;
        }
#define LOADP(dst, kk)                                                    \
        _Pragma("unroll") for (int j = 0; j < 4; ++j)                     \
          dst[j] = *(const bf16x8*)&wpl[bp[j] + (long)(kk) * 16];
        {
          bf16x8 S0[4], S1[4], S2[4];
          const int pbody = KSp * 32;
          LOADP(S0, 0)
          if (32 < pbody) LOADP(S1, 32)
          int k = 0;
          while (k + 96 <= pbody) {
            LOADP(S2, k + 64)
            MF4(S0, k)
            if (k + 96 < pbody) {
              LOADP(S0, k + 96)
            }
            MF4(S1, k + 32)
            if (k + 128 < pbody) {
              LOADP(S1, k + 128)
            }
            MF4(S2, k + 64)
            k += 96;
          }
          if (pbody - k == 32) {
            MF4(S0, k)
          } else if (pbody - k == 64) {
            MF4(S0, k)
            MF4(S1, k + 32)
          }
        }
#undef LOADP
      } else if (n_aligned) {
        if (nbody >= 32) {
          bf16x8 S0[4], S1[4], S2[4];
          LOADQ(S0, 0)
          if (32 < nbody) LOADQ(S1, 32)
          int k = 0;
          while (k + 96 <= nbody) {
            LOADQ(S2, k + 64)
            MF4(S0, k)
            if (k + 96 < nbody) {
              LOADQ(S0, k + 96)
            }
            MF4(S1, k + 32)
            if (k + 128 < nbody) {
              LOADQ(S1, k + 128)
            }
            MF4(S2, k + 64)
            k += 96;
          }
          if (nbody - k == 32) {
            MF4(S0, k)
          } else if (nbody - k == 64) {
            MF4(S0, k)
            MF4(S1, k + 32)
          }
        }
        if (nbody < N) {
          const int k8 = nbody + fk * 8;
          const int k8c = (k8 + 8 <= N) ? k8 : (N - 8);
          bf16x8 B[4];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            B[j] = *(const bf16x8*)&wtg[rb[j] + k8c];
          MF4(B, nbody)
        }
      } else {
        // tiny-N head layers (N = 1 or 8): guarded single pass
        for (int k = 0; k < N; k += 32) {
          const int k8 = k + fk * 8;
          bf16x8 B[4];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            B[j] = (j < nq)
                ? load_bfrag(wtg, t0 * 16 + j * 16 + fi, K, N, k8)
                : bf16x8{};
          MF4(B, k)
        }
      }
#undef LOADQ
#undef MF4

      // epilogue: when producing dy_{li-1}, apply ITS relu' mask here
      // (its dysave is written by the coalesced copy after the barrier)
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q >= nq) break;
        const f32x4 a = acc[0][q];
        const int col = (t0 + q) * 16 + fi;   // k index
        if (col < K) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = fk * 4 + r;
            if (want_dx0) {
              if (row < rowlim && col >= col_lo)
                dx0g[(long)(m0 + row) * (K - col_lo) + (col - col_lo)] =
                    a[r];
            } else {
              u16 v = f32_to_bf16_rne3(a[r]);
              if (yo_n != nullptr && row < rowlim
                  && yo_n[(long)(m0 + row) * K + col] == 0)
                v = 0;
              sd[nb][row][col] = v;
            }
          }
        }
      }
    }
    if (want_dx0) break;
    {
      const int pw = nb ? prevw1 : prevw0;
      for (int i = tid; i < TMv * (pw > K ? pw - K : 0); i += NTHR) {
        const int r = i / (pw - K), c = K + i % (pw - K);
        sd[nb][r][c] = 0;
      }
      if (nb) prevw1 = K; else prevw0 = K;
    }
    __syncthreads();
    // coalesced masked-dy save from LDS for the layer just produced
    if (sv_n != nullptr) {
      const int r2 = tid >> 5;
      const int lc2 = tid & 31;
      for (int rr = r2; rr < rowlim; rr += NTHR / 32) {
        u16* dst = sv_n + (long)(m0 + rr) * K;
        const u16* src = &sd[nb][rr][0];
        for (int c = lc2 * 8; c + 8 <= K; c += (NTHR / 32) * 8)
          *(uint4*)&dst[c] = *(const uint4*)&src[c];
      }
      if ((K & 7) && tid < TMv && tid < rowlim) {
        for (int c = K & ~7; c < K; ++c)
          sv_n[(long)(m0 + tid) * K + c] = sd[nb][tid][c];
      }
    }
    cur = nb;
  }
}

// Host binding: [dy_0 .. dy_{L-1} [, dx0]] = mlp_chain_dx_bf16(
//     dy_last, wts, youts, K0, acts_flags, G, save_dys, dx0_lo)
// wts[i]: TRANSPOSED bf16 weights [G*K_i, N_i]; youts[i]: mask tensors
// (post-act outputs, same [G?,M,N_i] layout) or an empty tensor where
// act=0.  Returns the per-layer MASKED dy tensors (empty placeholders
// when save_dys=0) and, when dx0_lo >= 0, the fp32 input gradient
// restricted to columns >= dx0_lo, per group [G, M, K0-dx0_lo].
static std::vector<torch::Tensor> mlp_chain_dx_bf16(
    torch::Tensor dy_last, std::vector<torch::Tensor> wts,
    std::vector<torch::Tensor> youts, long K0,
    std::vector<long> acts, long G, long save_dys, long dx0_lo,
    std::vector<torch::Tensor> wps) {
  CHAIN_CHECK_BF16(dy_last);
  const int L = (int)wts.size();
  TORCH_CHECK(L >= 1 && L <= 6 && (int)youts.size() == L
              && (int)acts.size() == L);
  TORCH_CHECK(wps.empty() || (int)wps.size() == L);
  auto dyc = dy_last.contiguous();
  const long M = G == 1 ? dyc.size(0) : dyc.size(1);
  ChainDxDesc d{};
  d.L = L;
  d.K0 = (int)K0;
  d.dx0_lo = (int)dx0_lo;
  std::vector<torch::Tensor> keep, out;
  auto bopts = dyc.options();
  long K = K0;
  for (int i = 0; i < L; ++i) {
    CHAIN_CHECK_BF16(wts[i]);
    auto wc = wts[i].contiguous();
    keep.push_back(wc);
    const long N = wc.numel() / (G * K);
    TORCH_CHECK(N * G * K == wc.numel(), "wt shape mismatch at layer ", i);
    TORCH_CHECK(N <= CMAX && K <= CMAX);
    d.wt[i] = (const u16*)wc.data_ptr();
    d.wp[i] = nullptr;
    if (!wps.empty() && wps[i].numel() > 0) {
      CHAIN_CHECK_BF16(wps[i]);
      TORCH_CHECK(wps[i].is_contiguous()
                  && wps[i].numel() ==
                         G * ((K + 15) / 16) * ((N + 31) / 32) * 512);
      d.wp[i] = (const u16*)wps[i].data_ptr();
      keep.push_back(wps[i]);
    }
    d.N[i] = (int)N;
    d.act[i] = (int)acts[i];
    d.yout[i] = nullptr;
    d.yo_gs[i] = 0;
    if (acts[i] && youts[i].numel() > 0) {
      CHAIN_CHECK_BF16(youts[i]);
      auto yc = youts[i].contiguous();
      keep.push_back(yc);
      TORCH_CHECK(yc.numel() == (yc.dim() == 3 ? G : 1) * M * N);
      d.yout[i] = (const u16*)yc.data_ptr();
      d.yo_gs[i] = yc.dim() == 3 ? M * N : 0;
    }
    d.dysave[i] = nullptr;
    if (save_dys) {
      auto t = torch::empty({G, M, N}, bopts);
      d.dysave[i] = (u16*)t.data_ptr();
      out.push_back(t);
    } else {
      out.push_back(torch::Tensor());
    }
    K = N;
  }
  TORCH_CHECK(dyc.numel() == G * M * d.N[L - 1]);
  torch::Tensor dx0;
  if (dx0_lo >= 0) {
    TORCH_CHECK(dx0_lo < K0);
    dx0 = torch::empty({G, M, K0 - dx0_lo},
                       bopts.dtype(torch::kFloat32));
    out.push_back(dx0);
  }
  dim3 grid((M + 15) / 16, 1, G);
  hipLaunchKernelGGL(k_bf16_chain_dx, grid, dim3(NTHR), 0, cur_stream3(),
                     (const u16*)dyc.data_ptr(), d,
                     dx0_lo >= 0 ? dx0.data_ptr<float>() : nullptr,
                     (int)M);
  return out;
}

// ---------------------------------------------------------------------------
// Host binding: [y, x_bf16, act_0, ..., act_{L-2}] = mlp_chain_fwd_bf16(
//     x1, x2_or_empty, ws, bs, act_last, G, out_f32, rm=0, rowcat=0,
//     save_acts=1)
// ws[i]: bf16 [G*N_i, K_i] (or [N,K] / [G,N,K]); bs[i]: f32 [G*N_i].
// Inputs fp32 or bf16, 2-D [M, C]; x2 may be an empty tensor.  x_bf16 is
// the converted (concatenated) input, so the backward consumes the same
// activation list the per-layer path produced.  rm: 0 = auto (by M),
// 1/2 = force 16/32 rows per workgroup.  rowcat=1 stacks x1/x2 over ROWS
// (equal widths; M = M1+M2) instead of columns.  save_acts=0 skips the
// intermediate-activation writes (no-grad forwards, e.g. TD target).
// ---------------------------------------------------------------------------
static std::vector<torch::Tensor> mlp_chain_fwd_bf16(
    torch::Tensor x1, torch::Tensor x2, std::vector<torch::Tensor> ws,
    std::vector<torch::Tensor> bs, long act_last, long G, long out_f32,
    long rm, long rowcat, long save_acts,
    std::vector<torch::Tensor> wps) {
  const int L = (int)ws.size();
  TORCH_CHECK(L >= 1 && L <= 6 && (int)bs.size() == L);
  TORCH_CHECK(wps.empty() || (int)wps.size() == L);
  TORCH_CHECK(x1.is_cuda() && x1.dim() == 2);
  const bool has2 = x2.numel() > 0;
  const bool x1f = x1.scalar_type() == torch::kFloat32;
  TORCH_CHECK(x1.scalar_type() == torch::kFloat32
              || x1.scalar_type() == torch::kBFloat16);
  auto x1c = x1.contiguous();
  torch::Tensor x2c = x2;
  bool x2f = false;
  if (has2) {
    x2f = x2.scalar_type() == torch::kFloat32;
    TORCH_CHECK(x2.is_cuda() && x2.dim() == 2
                && (x2.scalar_type() == torch::kFloat32
                    || x2.scalar_type() == torch::kBFloat16));
    if (rowcat) {
      TORCH_CHECK(x2.size(1) == x1.size(1), "rowcat needs equal widths");
    } else {
      TORCH_CHECK(x2.size(0) == x1.size(0));
    }
    x2c = x2.contiguous();
  }
  TORCH_CHECK(!rowcat || has2, "rowcat needs two inputs");
  const long M1 = x1c.size(0);
  const long M = rowcat ? M1 + x2c.size(0) : M1;
  const long C1 = x1c.size(1), C2 = has2 ? x2c.size(1) : 0;
  const long K0 = rowcat ? C1 : C1 + C2;
  TORCH_CHECK(K0 <= CMAX, "chain fwd: input width must be <= ", CMAX);

  ChainFwdDesc d{};
  d.L = L;
  std::vector<torch::Tensor> keep, out;
  auto bopts = x1c.options().dtype(torch::kBFloat16);
  auto xsave = torch::empty({M, K0}, bopts);
  out.resize(2);
  out[1] = xsave;
  long K = K0;
  for (int i = 0; i < L; ++i) {
    CHAIN_CHECK_BF16(ws[i]);
    CHAIN_CHECK_F32(bs[i]);
    auto wc = ws[i].contiguous();
    auto bc = bs[i].contiguous();
    keep.push_back(wc);
    keep.push_back(bc);
    const long N = wc.numel() / (G * K);
    TORCH_CHECK(N * G * K == wc.numel(), "weight shape mismatch at layer ",
                i, " (K=", K, ")");
    TORCH_CHECK(N <= CMAX, "chain fwd: layer width must be <= ", CMAX);
    d.w[i] = (const u16*)wc.data_ptr();
    d.wp[i] = nullptr;
    if (!wps.empty() && wps[i].numel() > 0) {
      CHAIN_CHECK_BF16(wps[i]);
      TORCH_CHECK(wps[i].is_contiguous()
                  && wps[i].numel() ==
                         G * ((N + 15) / 16) * ((K + 31) / 32) * 512);
      d.wp[i] = (const u16*)wps[i].data_ptr();
      keep.push_back(wps[i]);
    }
    d.b[i] = bc.data_ptr<float>();
    d.N[i] = (int)N;
    d.act[i] = (i == L - 1) ? (int)act_last : 1;
    d.acts[i] = nullptr;
    if (save_acts && i < L - 1) {
      auto a = G == 1 ? torch::empty({M, N}, bopts)
                      : torch::empty({G, M, N}, bopts);
      d.acts[i] = (u16*)a.data_ptr();
      out.push_back(a);
    }
    K = N;
  }
  auto yopts = x1c.options().dtype(out_f32 ? torch::kFloat32
                                           : torch::kBFloat16);
  auto y = G == 1 ? torch::empty({M, (long)d.N[L - 1]}, yopts)
                  : torch::empty({G, M, (long)d.N[L - 1]}, yopts);
  out[0] = y;
  // RM=1 measured faster at every bench shape (RM=2 halves the L2
  // weight re-read but also halves the workgroups, and these chains are
  // occupancy/latency-bound, not L2-bound — tools/validate_chain_fwd.py
  // A/B and the in-graph profile both agree).  rm=2 stays available for
  // experiments.
  int use_rm = (int)rm;
  if (use_rm == 0) use_rm = 1;
  const int TMv = 16 * use_rm;
  dim3 grid((M + TMv - 1) / TMv, 1, G);
  if (use_rm == 2)
    hipLaunchKernelGGL(k_bf16_chain_fwd<2>, grid, dim3(NTHR), 0,
                       cur_stream3(), x1c.data_ptr(),
                       has2 ? x2c.data_ptr() : nullptr, (int)C1, (int)C2,
                       (int)(x1f ? 1 : 0), (int)(x2f ? 1 : 0),
                       (int)rowcat, (int)M1,
                       (u16*)xsave.data_ptr(), d,
                       y.data_ptr(), (int)M, (int)out_f32);
  else
    hipLaunchKernelGGL(k_bf16_chain_fwd<1>, grid, dim3(NTHR), 0,
                       cur_stream3(), x1c.data_ptr(),
                       has2 ? x2c.data_ptr() : nullptr, (int)C1, (int)C2,
                       (int)(x1f ? 1 : 0), (int)(x2f ? 1 : 0),
                       (int)rowcat, (int)M1,
                       (u16*)xsave.data_ptr(), d,
                       y.data_ptr(), (int)M, (int)out_f32);
  return out;
}

}  // namespace chain

void register_chain(pybind11::module_& m) {
  m.def("transpose_weights_bf16", &chain::transpose_weights_bf16);
  m.def("pack_weights_frag", &chain::pack_weights_frag);
  m.def("mlp_chain_dx_bf16", &chain::mlp_chain_dx_bf16,
        pybind11::arg("dy_last"), pybind11::arg("wts"),
        pybind11::arg("youts"), pybind11::arg("K0"), pybind11::arg("acts"),
        pybind11::arg("G"), pybind11::arg("save_dys"),
        pybind11::arg("dx0_lo"),
        pybind11::arg("wps") = std::vector<torch::Tensor>{});
  m.def("mlp_chain_fwd_bf16", &chain::mlp_chain_fwd_bf16,
        pybind11::arg("x1"), pybind11::arg("x2"), pybind11::arg("ws"),
        pybind11::arg("bs"), pybind11::arg("act_last"), pybind11::arg("G"),
        pybind11::arg("out_f32"), pybind11::arg("rm") = 0,
        pybind11::arg("rowcat") = 0, pybind11::arg("save_acts") = 1,
        pybind11::arg("wps") = std::vector<torch::Tensor>{});
}
