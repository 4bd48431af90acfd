// MFMA-tiled 1-D convolution family for the VITS graph (gfx950).
//
// The HiFi-GAN generator is ~all the FLOPs of Piper synthesis
// (SURVEY.md §2.2): Conv1d (k=3/7/11, dilated) and ConvTranspose1d
// (k=16/4, stride 8/2) plus many 1x1 projections.  All are expressed as
// one MFMA GEMM kernel over "taps":
//
//   Out[co][n] = act( bias[co] + sum_tap sum_ci
//                     W[tap][co][ci] * pre_act(x[ci][in_off(tap) + n]) )
//
//   Conv1d          : tap = kernel position j, in_off = n0 - pad + j*dil,
//                     out index = n, ntaps = k.
//   ConvTranspose1d : phase decomposition — for output phase r (t = s*v +
//                     r - pad), tap m uses weight W[ci][co][r + s*m] and
//                     input x[ci][v - m]; out index = s*v + r - pad.
//                     ntaps = ceil((k - r)/s).  This turns the transposed
//                     conv into s dense GEMMs with shifted input windows
//                     (no zero-stuffing, no atomics).
//
// Tiling: block = 4 waves (256 thr), BM x BN output tile, K staged in
// 32-deep slices of Cin through LDS.  A (weights) is read as contiguous
// bf16x8 fragments (ds_read_b128); X is staged [ci][t] row-major with
// +8 element row padding.  Weights are pre-permuted host-side (cached) to
// [tap][Cout][Cin] so every global load is coalesced.
//
// mfma_f32_16x16x32_bf16 operand maps (verified against rocm CK headers,
// ck_tile warp_gemm_attribute_mfma_impl.hpp: kABKLane=4, kABKPerLane=8):
//   a_vec[q] = A[lane&15][(lane>>4)*8 + q]
//   b_vec[q] = B[(lane>>4)*8 + q][lane&15]
//   d[reg]   = D[(lane>>4)*4 + reg][lane&15]
#include "common.h"

#define BK 32

// activation modes
#define ACT_NONE 0
#define ACT_LRELU 1
#define ACT_TANH 2

// Max halo: (ntaps-1)*|tap_stride| — VITS worst case is k=11, dil=5 -> 50.
#define HALO_MAX 64

template <int BM, int BN, int WGM, int WGN, int TC>
__global__ __launch_bounds__(512) void conv1d_mfma_kernel(
    const bf16* __restrict__ x,     // [B][Cin][Tin]
    const bf16* __restrict__ w,     // [ntaps][CoutP][CinP] pre-permuted
    const float* __restrict__ bias, // [Cout] or null
    bf16* __restrict__ out,         // [B][Cout][Tout]
    const bf16* __restrict__ resid, // optional residual, same shape as out
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int ntaps, int tap_in_stride,  // input offset step per tap (dil or -1)
    int in_off0,                   // input offset of n=0, tap=0
    long Nvirt,                    // GEMM N size (Tout or V)
    int out_stride, int out_off,   // out t = n*out_stride + out_off
    float pre_slope,               // <0: no pre-act; else LeakyReLU slope
    int act_mode, float post_slope,
    // transposed-conv mode: all phases in ONE launch (convt_s > 0).
    // grid.y = s * mtiles; per-block phase r is derived below and ntaps /
    // offsets / weight base recomputed from (convt_k, convt_s, convt_pad).
    int convt_s, int convt_k, int convt_pad, int kr_max) {
  // grid: (ceil(Nvirt/BN), ceil(Cout/BM) [* s], B); 512 threads = 8 waves.
  const int n_tile = blockIdx.x;
  int m_tile = blockIdx.y;
  const int b = blockIdx.z;

  if (convt_s > 0) {
    const int mtiles = gridDim.y / convt_s;
    const int r = m_tile / mtiles;
    m_tile = m_tile % mtiles;
    ntaps = (convt_k - r + convt_s - 1) / convt_s;
    int v_lo = (convt_pad - r + convt_s - 1) / convt_s;
    if (v_lo < 0) v_lo = 0;
    const long v_hi = (Tout - 1 + convt_pad - r) / convt_s;  // inclusive
    Nvirt = v_hi - v_lo + 1;
    if (Nvirt <= 0) return;
    in_off0 = v_lo;
    out_off = v_lo * convt_s + r - convt_pad;
    out_stride = convt_s;
    w += (long)r * kr_max * CoutP * CinP;
  }

  const long n0 = (long)n_tile * BN;
  const int m0 = m_tile * BM;

  // X window staged ONCE per K-slice: all taps read from it at their own
  // column offset (k-fold less LDS staging + barriers than per-tap tiles).
  __shared__ bf16 Ws[TC][BM][BK + 8];
  __shared__ bf16 Xs[BK][BN + HALO_MAX + 8];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;  // wave row (M)
  const int wc = wid % WGN;  // wave col (N)
  constexpr int WM = BM / WGM;  // per-wave M
  constexpr int WN = BN / WGN;  // per-wave N
  constexpr int MT = WM / 16;   // m fragments per wave
  constexpr int NT = WN / 16;   // n fragments per wave
  constexpr int XW = BN + HALO_MAX;  // staged X width

  f32x4 acc[MT][NT];
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Cin * Tin;
  const int kl = lane >> 4;  // k-lane group (0..3)
  const int il = lane & 15;  // row/col within fragment

  // window start: smallest input index any tap needs for col 0
  const int min_tap_off =
      tap_in_stride < 0 ? (ntaps - 1) * tap_in_stride : 0;
  const long w0 = n0 + in_off0 + min_tap_off;

  // thread -> (row, col-chunk) map for X staging, computed once: thread i
  // handles chunks e = i*8 + s*4096 -> row/col via constexpr div (no
  // per-iteration guards on the interior fast path).
  const bool x_interior = (w0 >= 0) && (w0 + XW <= Tin);

  for (int c0 = 0; c0 < CinP; c0 += BK) {
    // ---- stage X window: Xs[r][c] = pre(x[c0+r][w0+c]), c < XW -------
    const bool rows_ok = (c0 + BK) <= Cin;
    if (x_interior && rows_ok) {
      if (pre_slope < 0.f) {
#pragma unroll
        for (int e = tid * 8; e < BK * XW; e += 512 * 8) {
          const int r = e / XW, c = e % XW;
          *(ulonglong2*)&Xs[r][c] =
              *(const ulonglong2*)&xb[(long)(c0 + r) * Tin + w0 + c];
        }
      } else {
#pragma unroll
        for (int e = tid * 8; e < BK * XW; e += 512 * 8) {
          const int r = e / XW, c = e % XW;
          bf16 vals[8];
          *(ulonglong2*)vals =
              *(const ulonglong2*)&xb[(long)(c0 + r) * Tin + w0 + c];
#pragma unroll
          for (int q = 0; q < 8; ++q)
            vals[q] = f2bf(lrelu_(bf2f(vals[q]), pre_slope));
          *(ulonglong2*)&Xs[r][c] = *(ulonglong2*)vals;
        }
      }
    } else {
#pragma unroll 2
      for (int e = tid * 8; e < BK * XW; e += 512 * 8) {
        int r = e / XW, c = e % XW;
        int ci = c0 + r;
        bf16 vals[8];
        long p = w0 + c;
        if (ci < Cin && p >= 0 && p + 7 < Tin) {
          *(ulonglong2*)vals = *(const ulonglong2*)&xb[(long)ci * Tin + p];
          if (pre_slope >= 0.f) {
#pragma unroll
            for (int q = 0; q < 8; ++q)
              vals[q] = f2bf(lrelu_(bf2f(vals[q]), pre_slope));
          }
        } else {
#pragma unroll
          for (int q = 0; q < 8; ++q) {
            long pq = p + q;
            float v = (ci < Cin && pq >= 0 && pq < Tin)
                          ? bf2f(xb[(long)ci * Tin + pq])
                          : 0.f;
            if (pre_slope >= 0.f) v = lrelu_(v, pre_slope);
            vals[q] = f2bf(v);
          }
        }
        // c is a multiple of 8 and rows are 16B-aligned -> one b128 write
        *(ulonglong2*)&Xs[r][c] = *(ulonglong2*)vals;
      }
    }

    for (int tap0 = 0; tap0 < ntaps; tap0 += TC) {
      const int ntc = min(TC, ntaps - tap0);
      // ---- stage W chunk: Ws[tc][m][kk] = w[tap0+tc][m0+m][c0+kk] ----
      for (int tc = 0; tc < ntc; ++tc) {
        const long wbase = ((long)(tap0 + tc) * CoutP + m0) * CinP + c0;
#pragma unroll
        for (int e = tid * 8; e < BM * BK; e += 512 * 8) {
          int m = e / BK, kk = e % BK;
          *(ulonglong2*)&Ws[tc][m][kk] =
              *(const ulonglong2*)&w[wbase + (long)m * CinP + kk];
        }
      }
      __syncthreads();

      for (int tc = 0; tc < ntc; ++tc) {
        // column offset of this tap inside the staged window
        const int toff = (tap0 + tc) * tap_in_stride - min_tap_off;
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) {
          const int ncol = wc * WN + nj * 16 + il + toff;
#pragma unroll
          for (int q = 0; q < 8; ++q)
            b_frag[nj][q] = *(__bf16*)&Xs[kl * 8 + q][ncol];
        }
#pragma unroll
        for (int mi = 0; mi < MT; ++mi) {
          bf16x8 a_frag =
              *(const bf16x8*)&Ws[tc][wr * WM + mi * 16 + il][kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj) {
            acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
          }
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: bias + activation (+ residual) + store --------------
  bf16* ob = out + (long)b * Cout * Tout;
  const bf16* rb = resid ? resid + (long)b * Cout * Tout : nullptr;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
    const int row = m0 + wr * WM + mi * 16 + kl * 4;
#pragma unroll
    for (int nj = 0; nj < NT; ++nj) {
      const long col = n0 + wc * WN + nj * 16 + il;
      if (col >= Nvirt) continue;
      const long t = col * out_stride + out_off;
      if (t < 0 || t >= Tout) continue;
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const int co = row + rg;
        if (co >= Cout) continue;
        float v = acc[mi][nj][rg];
        if (bias) v += bias[co];
        if (act_mode == ACT_LRELU) v = lrelu_(v, post_slope);
        else if (act_mode == ACT_TANH) v = tanhf(v);
        if (rb) v += bf2f(rb[(long)co * Tout + t]);
        ob[(long)co * Tout + t] = f2bf(v);
      }
    }
  }
}

// --------------------------------------------------------------------------
// ConvTranspose1d, phase-merged: each block computes ALL s phases of a
// BM x BN(v) tile, so each lane's stores are CONTIGUOUS s-runs
// (t = v*s + r).  The strided per-phase epilogue of the generic kernel
// collapses write efficiency s-fold; this kernel fixes it.
// Weights: [s][KR][CoutP][CinP] flattened to taps tau = r*KR + m;
// tap tau reads x[v - m] and accumulates into acc[r].
// --------------------------------------------------------------------------
template <int BM, int WGM, int WGN, int S, int KR>
__global__ __launch_bounds__(512) void convt1d_merged_kernel(
    const bf16* __restrict__ x,   // [B][Cin][Tin]
    const bf16* __restrict__ w,   // [S*KR][CoutP][CinP]
    const float* __restrict__ bias,
    bf16* __restrict__ out,       // [B][Cout][Tout]
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int pad, long Vn,             // v in [0, Vn)
    float pre_slope) {
  constexpr int BN = 32;            // v columns per block
  constexpr int TAUC = 4;           // tap chunk (Ws LDS budget)
  constexpr int XW = BN + KR + 6;   // window + halo (align pad)
  const long v0 = (long)blockIdx.x * BN;
  const int m0 = blockIdx.y * BM;
  const int b = blockIdx.z;

  __shared__ bf16 Ws[TAUC][BM][BK + 8];
  __shared__ bf16 Xs[BK][XW + 8];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  constexpr int WM = BM / WGM;
  constexpr int WN = BN / WGN;   // 16 when WGN=2
  constexpr int MT = WM / 16;
  static_assert(WN == 16, "convT merged kernel assumes NT=1");
  const int kl = lane >> 4;
  const int il = lane & 15;

  f32x4 acc[S][MT];
#pragma unroll
  for (int r = 0; r < S; ++r)
#pragma unroll
    for (int i = 0; i < MT; ++i) acc[r][i] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Cin * Tin;
  const long w0 = v0 - (KR - 1);  // tap m reads x[v - m]
  const bool x_interior = (w0 >= 0) && (w0 + XW <= Tin);

  for (int c0 = 0; c0 < CinP; c0 += BK) {
    const bool rows_ok = (c0 + BK) <= Cin;
    // ---- stage X window (BK x XW), one b128 chunk per thread ---------
    {
      // BK*XW elements; XW+8 row pitch keeps 16B alignment (XW mult of 8
      // not guaranteed -> use element loop with 8-chunks over rows)
      for (int e = tid * 8; e < BK * 48; e += 512 * 8) {
        int r = e / 48, c = e % 48;  // 48 >= XW rounded to 8
        if (c >= XW) continue;
        int ci = c0 + r;
        bf16 vals[8];
        long p = w0 + c;
        if (rows_ok && x_interior && ci < Cin && p + 7 < Tin && p >= 0) {
          *(ulonglong2*)vals = *(const ulonglong2*)&xb[(long)ci * Tin + p];
          if (pre_slope >= 0.f) {
#pragma unroll
            for (int q = 0; q < 8; ++q)
              vals[q] = f2bf(lrelu_(bf2f(vals[q]), pre_slope));
          }
        } else {
#pragma unroll
          for (int q = 0; q < 8; ++q) {
            long pq = p + q;
            float v = (ci < Cin && pq >= 0 && pq < Tin)
                          ? bf2f(xb[(long)ci * Tin + pq]) : 0.f;
            if (pre_slope >= 0.f) v = lrelu_(v, pre_slope);
            vals[q] = f2bf(v);
          }
        }
        *(ulonglong2*)&Xs[r][c] = *(ulonglong2*)vals;
      }
    }

    constexpr int NTAU = S * KR;
    static_assert(NTAU % TAUC == 0, "tap chunking must divide evenly");
    static_assert(XW + 8 == 48, "staging loop assumes 48-elem row pitch");
#pragma unroll
    for (int cc = 0; cc < NTAU / TAUC; ++cc) {
#pragma unroll
      for (int tc = 0; tc < TAUC; ++tc) {
        const long wbase = ((long)(cc * TAUC + tc) * CoutP + m0) * CinP + c0;
#pragma unroll
        for (int e = tid * 8; e < BM * BK; e += 512 * 8) {
          int m = e / BK, kk = e % BK;
          *(ulonglong2*)&Ws[tc][m][kk] =
              *(const ulonglong2*)&w[wbase + (long)m * CinP + kk];
        }
      }
      __syncthreads();

#pragma unroll
      for (int tc = 0; tc < TAUC; ++tc) {
        constexpr int dummy = 0;
        const int tau = cc * TAUC + tc;  // compile-time (both loops unroll)
        const int r = tau / KR;          // phase -> static acc index
        const int m = tau % KR;          // tap
        const int toff = (KR - 1) - m;
        (void)dummy;
        bf16x8 b_frag;
        const int ncol = wc * WN + il + toff;
#pragma unroll
        for (int q = 0; q < 8; ++q)
          b_frag[q] = *(__bf16*)&Xs[kl * 8 + q][ncol];
#pragma unroll
        for (int mi = 0; mi < MT; ++mi) {
          bf16x8 a_frag =
              *(const bf16x8*)&Ws[tc][wr * WM + mi * 16 + il][kl * 8];
          acc[r][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[r][mi], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: per (row, v) store the S-run contiguously -----------
  bf16* ob = out + (long)b * Cout * Tout;
  const long v = v0 + wc * WN + il;
  if (v < Vn) {
#pragma unroll
    for (int mi = 0; mi < MT; ++mi) {
      const int row = m0 + wr * WM + mi * 16 + kl * 4;
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const int co = row + rg;
        if (co >= Cout) continue;
        const float bv = bias ? bias[co] : 0.f;
        bf16 run[S];
#pragma unroll
        for (int r = 0; r < S; ++r) run[r] = f2bf(acc[r][mi][rg] + bv);
        const long t0 = v * S - pad;
        if (t0 >= 0 && t0 + S <= Tout) {
          // t0 is even -> byte address 8B-aligned (pad is even for k=2s
          // upsamplers); use two b64 stores for S=8, scalars otherwise
          if constexpr (S == 8) {
            *(unsigned long long*)&ob[(long)co * Tout + t0] =
                ((unsigned long long*)run)[0];
            *(unsigned long long*)&ob[(long)co * Tout + t0 + 4] =
                ((unsigned long long*)run)[1];
          } else {
#pragma unroll
            for (int r = 0; r < S; ++r) ob[(long)co * Tout + t0 + r] = run[r];
          }
        } else {
#pragma unroll
          for (int r = 0; r < S; ++r) {
            const long t = t0 + r;
            if (t >= 0 && t < Tout) ob[(long)co * Tout + t] = run[r];
          }
        }
      }
    }
  }
}

// --------------------------------------------------------------------------
// Naive direct conv (grouped / f32 / stride>1 fallback + oracle).
// One thread per (b, co, t_out).
// --------------------------------------------------------------------------
template <typename T>
__global__ void conv1d_naive_kernel(
    const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ bias, T* __restrict__ out, int Cin, int Cout,
    long Tin, long Tout, int k, int stride, int pad, int dil, int groups,
    float pre_slope, int act_mode, float post_slope, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const long t = i % Tout;
  const int co = (i / Tout) % Cout;
  const long b = i / ((long)Tout * Cout);
  const int cpg_in = Cin / groups;
  const int cpg_out = Cout / groups;
  const int g = co / cpg_out;
  const T* xb = x + (b * Cin + (long)g * cpg_in) * Tin;
  const T* wc = w + (long)co * cpg_in * k;
  float accv = bias ? bias[co] : 0.f;
  for (int ci = 0; ci < cpg_in; ++ci) {
    for (int j = 0; j < k; ++j) {
      long p = t * stride + (long)j * dil - pad;
      if (p < 0 || p >= Tin) continue;
      float xv = ld_f(xb + (long)ci * Tin + p);
      if (pre_slope >= 0.f) xv = lrelu_(xv, pre_slope);
      accv += xv * ld_f(wc + ci * k + j);
    }
  }
  if (act_mode == ACT_LRELU) accv = lrelu_(accv, post_slope);
  else if (act_mode == ACT_TANH) accv = tanhf(accv);
  st_f(out + i, accv);
}

// naive transposed conv fallback: thread per (b, co, t_out)
template <typename T>
__global__ void convt1d_naive_kernel(
    const T* __restrict__ x, const T* __restrict__ w,  // [Cin][Cout][k]
    const float* __restrict__ bias, T* __restrict__ out, int Cin, int Cout,
    long Tin, long Tout, int k, int stride, int pad, float pre_slope,
    long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const long t = i % Tout;
  const int co = (i / Tout) % Cout;
  const long b = i / ((long)Tout * Cout);
  const T* xb = x + b * (long)Cin * Tin;
  float accv = bias ? bias[co] : 0.f;
  for (int j = 0; j < k; ++j) {
    long num = t + pad - j;
    if (num < 0 || num % stride) continue;
    long u = num / stride;
    if (u >= Tin) continue;
    for (int ci = 0; ci < Cin; ++ci) {
      float xv = ld_f(xb + (long)ci * Tin + u);
      if (pre_slope >= 0.f) xv = lrelu_(xv, pre_slope);
      accv += xv * ld_f(w + ((long)ci * Cout + co) * k + j);
    }
  }
  st_f(out + i, accv);
}

// ========================================================================
// host wrappers
// ========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static inline hipStream_t cur_stream2() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

static inline long conv_out_len(long Tin, int k, int stride, int pad,
                                int dil) {
  return (Tin + 2L * pad - (long)dil * (k - 1) - 1) / stride + 1;
}

// Launch helper: pick tile config by Cout and launch the MFMA kernel.
static void launch_conv_mfma(const bf16* x, const bf16* w, const float* bias,
                             bf16* out, const bf16* resid, int B, int Cin,
                             int CinP, int Cout, int CoutP, long Tin,
                             long Tout, int ntaps, int tap_in_stride,
                             int in_off0, long Nvirt, int out_stride,
                             int out_off, float pre_slope, int act_mode,
                             float post_slope, hipStream_t stream,
                             int convt_s = 0, int convt_k = 0,
                             int convt_pad = 0, int kr_max = 0) {
  const int phases = convt_s > 0 ? convt_s : 1;
#define LAUNCH(BM, BN, WGM, WGN, TC)                                        \
  do {                                                                      \
    dim3 grid(ceil_div(Nvirt, BN), ceil_div(Cout, BM) * phases, B);         \
    hipLaunchKernelGGL((conv1d_mfma_kernel<BM, BN, WGM, WGN, TC>), grid,    \
                       dim3(512), 0, stream, x, w, bias, out, resid, Cin,   \
                       CinP, Cout, CoutP, Tin, Tout, ntaps, tap_in_stride,  \
                       in_off0, Nvirt, out_stride, out_off, pre_slope,      \
                       act_mode, post_slope, convt_s, convt_k, convt_pad,   \
                       kr_max);                                             \
  } while (0)
  if (Cout >= 128) LAUNCH(128, 128, 4, 2, 2);
  else if (Cout >= 64) LAUNCH(64, 128, 2, 4, 4);
  else LAUNCH(32, 256, 1, 8, 8);
#undef LAUNCH
}

torch::Tensor conv1d_fused(torch::Tensor x, torch::Tensor w_perm,
                           c10::optional<torch::Tensor> bias, long Cout,
                           long k, long stride, long padding, long dilation,
                           long groups, double pre_lrelu, long act_mode,
                           double post_slope,
                           c10::optional<torch::Tensor> residual) {
  // w_perm: [k][CoutP][CinP] bf16 (pre-permuted+padded) for the MFMA path
  //         or the raw [Cout][Cin/g][k] tensor for the fallback path.
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const long Tout = conv_out_len(Tin, k, stride, padding, dilation);
  auto out = torch::empty({B, Cout, Tout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat
                 ? *bias
                 : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const bool mfma_ok = x.scalar_type() == at::kBFloat16 && groups == 1 &&
                       stride == 1 && w_perm.dim() == 3 &&
                       w_perm.size(0) == k &&
                       (k - 1) * dilation <= 64 /* HALO_MAX */;
  const bf16* res_p = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->sizes() == out.sizes() && residual->is_contiguous());
    res_p = (const bf16*)residual->data_ptr();
  }
  if (mfma_ok) {
    const int CoutP = w_perm.size(1), CinP = w_perm.size(2);
    launch_conv_mfma((const bf16*)x.data_ptr(),
                     (const bf16*)w_perm.data_ptr(), bias_p,
                     (bf16*)out.data_ptr(), res_p, B, Cin, CinP, Cout, CoutP,
                     Tin, Tout, k, dilation, -(int)padding, Tout, 1, 0,
                     (float)pre_lrelu, (int)act_mode, (float)post_slope,
                     cur_stream2());
  } else {
    // fallback: w_perm is the raw [Cout][Cin/g][k] weight
    const long n = B * Cout * Tout;
    const int threads = 256;
    DISPATCH_FT_CONV(x, {
      hipLaunchKernelGGL(conv1d_naive_kernel<scalar_t>,
                         dim3((n + threads - 1) / threads), dim3(threads), 0,
                         cur_stream2(), (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w_perm.data_ptr(), bias_p,
                         (scalar_t*)out.data_ptr(), Cin, Cout, Tin, Tout, k,
                         stride, padding, dilation, groups, (float)pre_lrelu,
                         (int)act_mode, (float)post_slope, n);
    });
    if (res_p) out.add_(*residual);
  }
  return out;
}

torch::Tensor convtranspose1d_fused(torch::Tensor x, torch::Tensor w_perm,
                                    c10::optional<torch::Tensor> bias,
                                    long Cout, long k, long stride,
                                    long padding, double pre_lrelu) {
  // w_perm (MFMA): [s][kr_max][CoutP][CinP]; fallback: raw [Cin][Cout][k].
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const long Tout = (Tin - 1) * stride - 2 * padding + k;
  auto out = torch::empty({B, Cout, Tout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat
                 ? *bias
                 : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const bool mfma_ok =
      x.scalar_type() == at::kBFloat16 && w_perm.dim() == 4;
  if (mfma_ok) {
    const int kr_max = w_perm.size(1);
    const int CoutP = w_perm.size(2), CinP = w_perm.size(3);
    const long Vn = (Tout - 1 + padding) / stride + 1;
    const float pre = (float)pre_lrelu;
    hipStream_t st = cur_stream2();
    const bf16* xp = (const bf16*)x.data_ptr();
    const bf16* wp = (const bf16*)w_perm.data_ptr();
    bf16* op = (bf16*)out.data_ptr();
    // merged kernel needs WM = BM/WGM >= 16 (MT >= 1); BM=32 with the
    // fixed 4x2 wave grid gives MT=0 -> Cout<64 uses the generic path.
    bool merged = (k == 2 * stride) && kr_max == 2 &&
                  (stride == 8 || stride == 2) && Cout >= 64;
#define LAUNCH_M(BM, WGM, WGN, S)                                           \
  hipLaunchKernelGGL((convt1d_merged_kernel<BM, WGM, WGN, S, 2>),           \
                     dim3(ceil_div(Vn, 32), ceil_div(Cout, BM), B),         \
                     dim3(512), 0, st, xp, wp, bias_p, op, Cin, CinP,       \
                     Cout, CoutP, Tin, Tout, (int)padding, Vn, pre)
    if (merged && stride == 8) {
      if (Cout >= 128) LAUNCH_M(128, 4, 2, 8);
      else if (Cout >= 64) LAUNCH_M(64, 4, 2, 8);
      else LAUNCH_M(64, 4, 2, 8); /* unreachable: Cout>=64 guard */
    } else if (merged && stride == 2) {
      if (Cout >= 128) LAUNCH_M(128, 4, 2, 2);
      else if (Cout >= 64) LAUNCH_M(64, 4, 2, 2);
      else LAUNCH_M(64, 4, 2, 2); /* unreachable: Cout>=64 guard */
    } else {
      // generic: all phase GEMMs in one launch (phase folded into grid.y)
      launch_conv_mfma(xp, wp, bias_p, op, nullptr, B, Cin, CinP, Cout,
                       CoutP, Tin, Tout, kr_max, -1, 0, Vn, stride, 0,
                       pre, ACT_NONE, 0.f, st,
                       (int)stride, (int)k, (int)padding, kr_max);
    }
#undef LAUNCH_M
  } else {
    const long n = B * Cout * Tout;
    const int threads = 256;
    DISPATCH_FT_CONV(x, {
      hipLaunchKernelGGL(convt1d_naive_kernel<scalar_t>,
                         dim3((n + threads - 1) / threads), dim3(threads), 0,
                         cur_stream2(), (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w_perm.data_ptr(), bias_p,
                         (scalar_t*)out.data_ptr(), Cin, Cout, Tin, Tout, k,
                         stride, padding, (float)pre_lrelu, n);
    });
  }
  return out;
}
