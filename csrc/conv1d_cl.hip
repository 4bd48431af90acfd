// Channel-last (NTC) MFMA 1-D convolutions for the HiFi-GAN decode path
// (gfx950).  v5 of the conv engine; replaces the channel-first kernels of
// conv1d.hip on the hot path.
//
// Why channel-last: mfma_f32_16x16x32_bf16 consumes 8 CONTIGUOUS k
// elements per lane for BOTH operands (a_vec[q]=A[m][k0+q],
// b_vec[q]=B[k0+q][n] read from a [n][k] image).  With activations
// stored [B][T][C] the reduction dim (Cin) is memory-contiguous, so
//   - LDS staging is lane-linear b128 writes (no transpose, no scatter),
//   - A (X window) and B (weights) fragments are single ds_read_b128s,
//   - with a 40-element (20-dword) row pitch all 16 il-lanes land on
//     distinct banks (20*il mod 64 cycles through 16 residues): zero
//     bank conflicts by construction.
// The previous channel-first kernel read B fragments as 8 scalar bf16
// LDS loads (column reads) — measured SQ_LDS_BANK_CONFLICT/IDX ≈ 50%,
// MFMA busy ≈ 2-4% (profiles/r01_bench_kernel_stats.txt, PMC run).
//
// GEMM orientation: D[t][co] = sum_{ci,tap} X[t + tap*dil - pad][ci] *
// W[tap][co][ci] — A = X window (M = time rows), B = W tap slice
// (N = Cout columns), K = Cin in 32-deep LDS slices.  Taps shift the A
// ROW offset (halo rows staged once per K-slice).  Weights keep the
// host-side [tap][CoutP][CinP] permutation of conv1d.hip.
//
// Fused epilogue: bias + LeakyReLU/tanh + residual add + ragged-batch
// row masking (out rows >= out_lens[b] store 0 — replaces the separate
// mask_tail_ pass of the channel-first path).
#include "common.h"

#define BK 32
#define BKP 40  // row pitch: 32 data + 8 pad -> 20 dwords, conflict-free
#define HALO_MAX 64

#define ACT_NONE 0
#define ACT_LRELU 1
#define ACT_TANH 2

// --------------------------------------------------------------------------
// stride-1 conv, channel-last.
// grid: (ceil(Tout/BM), ceil(CoutP/BN), B); 512 threads = 8 waves.
// --------------------------------------------------------------------------
template <int BM, int BN, int WGM, int WGN, int TC>
__global__ __launch_bounds__(512) void conv1d_cl_kernel(
    const bf16* __restrict__ x,     // [B][Tin][Cin]
    const bf16* __restrict__ w,     // [ntaps][CoutP][CinP]
    const float* __restrict__ bias, // [Cout] or null
    bf16* __restrict__ out,         // [B][Tout][Cout]
    const bf16* __restrict__ resid, // optional [B][Tout][Cout]
    const int* __restrict__ out_lens,  // optional valid rows per batch
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int ntaps, int dil, int pad, float pre_slope, int act_mode,
    float post_slope) {
  constexpr int ROWS = BM + HALO_MAX;
  const long t0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int b = blockIdx.z;

  __shared__ bf16 Xs[ROWS][BKP];
  __shared__ bf16 Ws[TC][BN][BKP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  constexpr int WM = BM / WGM;
  constexpr int WN = BN / WGN;
  constexpr int MT = WM / 16;
  constexpr int NT = WN / 16;
  const int kl = lane >> 4;
  const int il = lane & 15;
  // conflict-free staging lane map (see resblock_cl.hip: quads keep a
  // row's 4 chunks contiguous for 64B coalescing; phase-group quads
  // cover rows {0,4,8,12}+g so LDS bank slots (5r+c)%16 = 4q+c)
  const int sr_l = 4 * ((lane >> 2) & 3) + (lane >> 4);
  const int sr_c = (lane & 3) * 8;
  const int sr_base0 = wid * 16;

  f32x4 acc[MT][NT];
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Tin * Cin;
  const long row0 = t0 - pad;  // x row staged at Xs[0]
  const int halo = (ntaps - 1) * dil;
  const int rows_used = BM + halo;
  // interior fast path: all staged rows in [0, Tin), full K-slice in Cin
  const bool t_interior = (row0 >= 0) && (row0 + rows_used <= Tin);

  for (int c0 = 0; c0 < CinP; c0 += BK) {
    const bool c_interior = (c0 + BK) <= Cin;
    // ---- stage X rows: Xs[r][0..31] = pre(x[row0+r][c0..c0+31]) ------
    if (t_interior && c_interior) {
      for (int base = sr_base0; base < rows_used; base += 128) {
        const int r = base + sr_l;
        if (r >= rows_used) continue;
        const int ch = sr_c;
        bf16 v8[8];
        *(ulonglong2*)v8 =
            *(const ulonglong2*)&xb[(row0 + r) * Cin + c0 + ch];
        if (pre_slope >= 0.f) {
#pragma unroll
          for (int q = 0; q < 8; ++q)
            v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    } else {
      for (int base = sr_base0; base < rows_used; base += 128) {
        const int r = base + sr_l;
        if (r >= rows_used) continue;
        const int ch = sr_c;
        const long t = row0 + r;
        bf16 v8[8];
        if (t >= 0 && t < Tin && c_interior) {
          *(ulonglong2*)v8 = *(const ulonglong2*)&xb[t * Cin + c0 + ch];
          if (pre_slope >= 0.f) {
#pragma unroll
            for (int q = 0; q < 8; ++q)
              v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
          }
        } else if (t >= 0 && t < Tin) {
#pragma unroll
          for (int q = 0; q < 8; ++q) {
            const int c = c0 + ch + q;
            float v = c < Cin ? bf2f(xb[t * Cin + c]) : 0.f;
            if (pre_slope >= 0.f) v = lrelu_(v, pre_slope);
            v8[q] = f2bf(v);
          }
        } else {
#pragma unroll
          for (int q = 0; q < 8; ++q) v8[q] = f2bf(0.f);
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    }

    for (int tap0 = 0; tap0 < ntaps; tap0 += TC) {
      const int ntc = min(TC, ntaps - tap0);
      // ---- stage W taps: Ws[tc][n][kk] = w[tap][n0+n][c0+kk] ---------
      for (int tc = 0; tc < ntc; ++tc) {
        const long wbase = ((long)(tap0 + tc) * CoutP + n0) * CinP + c0;
        for (int base = sr_base0; base < BN; base += 128) {
          const int n = base + sr_l;
          const int ch = sr_c;
          *(ulonglong2*)&Ws[tc][n][ch] =
              *(const ulonglong2*)&w[wbase + (long)n * CinP + ch];
        }
      }
      __syncthreads();

      for (int tc = 0; tc < ntc; ++tc) {
        const int toff = (tap0 + tc) * dil;  // A-row shift of this tap
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] = *(const bf16x8*)&Ws[tc][wc * WN + nj * 16 + il][kl * 8];
#pragma unroll
        for (int mi = 0; mi < MT; ++mi) {
          const bf16x8 a_frag =
              *(const bf16x8*)&Xs[wr * WM + mi * 16 + il + toff][kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj)
            acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: bias + act (+ residual) + ragged mask + store -------
  bf16* ob = out + (long)b * Tout * Cout;
  const bf16* rb = resid ? resid + (long)b * Tout * Cout : nullptr;
  const long lim = out_lens ? min((long)out_lens[b], Tout) : Tout;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long t = t0 + wr * WM + mi * 16 + kl * 4 + rg;
      if (t >= Tout) continue;
      const bool live = t < lim;
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) {
        const int co = n0 + wc * WN + nj * 16 + il;
        if (co >= Cout) continue;
        float v = 0.f;
        if (live) {
          v = acc[mi][nj][rg];
          if (bias) v += bias[co];
          if (act_mode == ACT_LRELU) v = lrelu_(v, post_slope);
          else if (act_mode == ACT_TANH) v = tanhf(v);
          if (rb) v += bf2f(rb[t * Cout + co]);
        }
        ob[t * Cout + co] = f2bf(v);
      }
    }
  }
}

// --------------------------------------------------------------------------
// Hybrid conv: A (X window) staged in LDS, B (weight taps) loaded per
// wave STRAIGHT from global memory.  Weights are tiny and shared by
// thousands of blocks (L2-hot); skipping the Ws LDS stage removes the
// per-tap-chunk barrier pair entirely (one barrier pair per K-slice)
// and frees ~20 KB LDS -> more blocks/CU.  A-side stays staged: the
// fully-direct variant measured 1.6-2.6x slower (see note above).
// --------------------------------------------------------------------------
template <int BM, int BN, int WGM, int WGN>
__global__ __launch_bounds__(512) void conv1d_cl_wdirect_kernel(
    const bf16* __restrict__ x,     // [B][Tin][Cin]
    const bf16* __restrict__ w,     // [ntaps][CoutP][CinP]
    const float* __restrict__ bias,
    bf16* __restrict__ out,         // [B][Tout][Cout]
    const bf16* __restrict__ resid,
    const int* __restrict__ out_lens,
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int ntaps, int dil, int pad, float pre_slope, int act_mode,
    float post_slope) {
  constexpr int ROWS = BM + HALO_MAX;
  const long t0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int b = blockIdx.z;

  __shared__ bf16 Xs[ROWS][BKP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  constexpr int WM = BM / WGM;
  constexpr int WN = BN / WGN;
  constexpr int MT = WM / 16;
  constexpr int NT = WN / 16;
  const int kl = lane >> 4;
  const int il = lane & 15;
  // conflict-free staging lane map (see resblock_cl.hip: quads keep a
  // row's 4 chunks contiguous for 64B coalescing; phase-group quads
  // cover rows {0,4,8,12}+g so LDS bank slots (5r+c)%16 = 4q+c)
  const int sr_l = 4 * ((lane >> 2) & 3) + (lane >> 4);
  const int sr_c = (lane & 3) * 8;
  const int sr_base0 = wid * 16;

  f32x4 acc[MT][NT];
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Tin * Cin;
  const long row0 = t0 - pad;
  const int halo = (ntaps - 1) * dil;
  const int rows_used = BM + halo;
  const bool t_interior = (row0 >= 0) && (row0 + rows_used <= Tin);
  // this lane's weight row base for each nj fragment
  const int wcol = n0 + wc * WN + il;

  for (int c0 = 0; c0 < CinP; c0 += BK) {
    const bool c_interior = (c0 + BK) <= Cin;
    __syncthreads();  // protect Xs against the previous slice's readers
    if (t_interior && c_interior) {
      for (int base = sr_base0; base < rows_used; base += 128) {
        const int r = base + sr_l;
        if (r >= rows_used) continue;
        const int ch = sr_c;
        bf16 v8[8];
        *(ulonglong2*)v8 =
            *(const ulonglong2*)&xb[(row0 + r) * Cin + c0 + ch];
        if (pre_slope >= 0.f) {
#pragma unroll
          for (int q = 0; q < 8; ++q)
            v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    } else {
      for (int base = sr_base0; base < rows_used; base += 128) {
        const int r = base + sr_l;
        if (r >= rows_used) continue;
        const int ch = sr_c;
        const long t = row0 + r;
        bf16 v8[8];
        if (t >= 0 && t < Tin && c_interior) {
          *(ulonglong2*)v8 = *(const ulonglong2*)&xb[t * Cin + c0 + ch];
          if (pre_slope >= 0.f) {
#pragma unroll
            for (int q = 0; q < 8; ++q)
              v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
          }
        } else if (t >= 0 && t < Tin) {
#pragma unroll
          for (int q = 0; q < 8; ++q) {
            const int c = c0 + ch + q;
            float v = c < Cin ? bf2f(xb[t * Cin + c]) : 0.f;
            if (pre_slope >= 0.f) v = lrelu_(v, pre_slope);
            v8[q] = f2bf(v);
          }
        } else {
#pragma unroll
          for (int q = 0; q < 8; ++q) v8[q] = f2bf(0.f);
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    }
    __syncthreads();

    for (int tap = 0; tap < ntaps; ++tap) {
      bf16x8 b_frag[NT];
#pragma unroll
      for (int nj = 0; nj < NT; ++nj)
        b_frag[nj] = *(const bf16x8*)&w[
            ((long)tap * CoutP + wcol + nj * 16) * CinP + c0 + kl * 8];
      const int toff = tap * dil;
#pragma unroll
      for (int mi = 0; mi < MT; ++mi) {
        const bf16x8 a_frag =
            *(const bf16x8*)&Xs[wr * WM + mi * 16 + il + toff][kl * 8];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
      }
    }
  }

  bf16* ob = out + (long)b * Tout * Cout;
  const bf16* rb = resid ? resid + (long)b * Tout * Cout : nullptr;
  const long lim = out_lens ? min((long)out_lens[b], Tout) : Tout;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long t = t0 + wr * WM + mi * 16 + kl * 4 + rg;
      if (t >= Tout) continue;
      const bool live = t < lim;
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) {
        const int co = n0 + wc * WN + nj * 16 + il;
        if (co >= Cout) continue;
        float v = 0.f;
        if (live) {
          v = acc[mi][nj][rg];
          if (bias) v += bias[co];
          if (act_mode == ACT_LRELU) v = lrelu_(v, post_slope);
          else if (act_mode == ACT_TANH) v = tanhf(v);
          if (rb) v += bf2f(rb[t * Cout + co]);
        }
        ob[t * Cout + co] = f2bf(v);
      }
    }
  }
}

// --------------------------------------------------------------------------
// ConvTranspose1d, channel-last, phase-merged (k = 2*stride, KR = 2):
// each block computes ALL s phases of a BMV(v) x BN(co) tile.
// D[r][v][co] = sum_{m,ci} X[v-m][ci] * W[r*KR+m][co][ci];
// out row t = v*s + r - pad.
// --------------------------------------------------------------------------
template <int BMV, int BN, int WGM, int WGN, int S, int KR>
__global__ __launch_bounds__(512) void convt1d_cl_kernel(
    const bf16* __restrict__ x,   // [B][Tin][Cin]
    const bf16* __restrict__ w,   // [S*KR][CoutP][CinP]
    const float* __restrict__ bias,
    bf16* __restrict__ out,       // [B][Tout][Cout]
    const int* __restrict__ out_lens,
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int pad, long Vn, float pre_slope) {
  constexpr int ROWS = BMV + (KR - 1);
  constexpr int TAUC = 4;
  const long v0 = (long)blockIdx.x * BMV;
  const int n0 = blockIdx.y * BN;
  const int b = blockIdx.z;

  __shared__ bf16 Xs[ROWS][BKP];
  __shared__ bf16 Ws[TAUC][BN][BKP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  constexpr int WM = BMV / WGM;
  constexpr int WN = BN / WGN;
  constexpr int MT = WM / 16;
  constexpr int NT = WN / 16;
  const int kl = lane >> 4;
  const int il = lane & 15;
  // conflict-free staging lane map (see resblock_cl.hip: quads keep a
  // row's 4 chunks contiguous for 64B coalescing; phase-group quads
  // cover rows {0,4,8,12}+g so LDS bank slots (5r+c)%16 = 4q+c)
  const int sr_l = 4 * ((lane >> 2) & 3) + (lane >> 4);
  const int sr_c = (lane & 3) * 8;
  const int sr_base0 = wid * 16;

  f32x4 acc[S][MT][NT];
#pragma unroll
  for (int r = 0; r < S; ++r)
#pragma unroll
    for (int i = 0; i < MT; ++i)
#pragma unroll
      for (int j = 0; j < NT; ++j) acc[r][i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Tin * Cin;
  const long row0 = v0 - (KR - 1);  // tap m reads x[v - m]
  const bool t_interior = (row0 >= 0) && (row0 + ROWS <= Tin);

  for (int c0 = 0; c0 < CinP; c0 += BK) {
    const bool c_interior = (c0 + BK) <= Cin;
    if (t_interior && c_interior) {
      for (int base = sr_base0; base < ROWS; base += 128) {
        const int r = base + sr_l;
        if (r >= ROWS) continue;
        const int ch = sr_c;
        bf16 v8[8];
        *(ulonglong2*)v8 =
            *(const ulonglong2*)&xb[(row0 + r) * Cin + c0 + ch];
        if (pre_slope >= 0.f) {
#pragma unroll
          for (int q = 0; q < 8; ++q)
            v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    } else {
      for (int base = sr_base0; base < ROWS; base += 128) {
        const int r = base + sr_l;
        if (r >= ROWS) continue;
        const int ch = sr_c;
        const long t = row0 + r;
        bf16 v8[8];
        if (t >= 0 && t < Tin && c_interior) {
          *(ulonglong2*)v8 = *(const ulonglong2*)&xb[t * Cin + c0 + ch];
          if (pre_slope >= 0.f) {
#pragma unroll
            for (int q = 0; q < 8; ++q)
              v8[q] = f2bf(lrelu_(bf2f(v8[q]), pre_slope));
          }
        } else if (t >= 0 && t < Tin) {
#pragma unroll
          for (int q = 0; q < 8; ++q) {
            const int c = c0 + ch + q;
            float v = c < Cin ? bf2f(xb[t * Cin + c]) : 0.f;
            if (pre_slope >= 0.f) v = lrelu_(v, pre_slope);
            v8[q] = f2bf(v);
          }
        } else {
#pragma unroll
          for (int q = 0; q < 8; ++q) v8[q] = f2bf(0.f);
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    }

    constexpr int NTAU = S * KR;
    static_assert(NTAU % TAUC == 0, "tap chunking must divide evenly");
#pragma unroll
    for (int cc = 0; cc < NTAU / TAUC; ++cc) {
#pragma unroll
      for (int tc = 0; tc < TAUC; ++tc) {
        const long wbase = ((long)(cc * TAUC + tc) * CoutP + n0) * CinP + c0;
        for (int base = sr_base0; base < BN; base += 128) {
          const int n = base + sr_l;
          const int ch = sr_c;
          *(ulonglong2*)&Ws[tc][n][ch] =
              *(const ulonglong2*)&w[wbase + (long)n * CinP + ch];
        }
      }
      __syncthreads();

#pragma unroll
      for (int tc = 0; tc < TAUC; ++tc) {
        const int tau = cc * TAUC + tc;  // compile-time (loops unroll)
        const int r = tau / KR;
        const int m = tau % KR;
        const int toff = (KR - 1) - m;
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] = *(const bf16x8*)&Ws[tc][wc * WN + nj * 16 + il][kl * 8];
#pragma unroll
        for (int mi = 0; mi < MT; ++mi) {
          const bf16x8 a_frag =
              *(const bf16x8*)&Xs[wr * WM + mi * 16 + il + toff][kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj)
            acc[r][mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[r][mi][nj], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: per (v-row, phase) store out[v*S + r - pad][co] -----
  bf16* ob = out + (long)b * Tout * Cout;
  const long lim = out_lens ? min((long)out_lens[b], Tout) : Tout;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long v = v0 + wr * WM + mi * 16 + kl * 4 + rg;
      if (v >= Vn) continue;
#pragma unroll
      for (int r = 0; r < S; ++r) {
        const long t = v * S + r - pad;
        if (t < 0 || t >= Tout) continue;
        const bool live = t < lim;
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) {
          const int co = n0 + wc * WN + nj * 16 + il;
          if (co >= Cout) continue;
          float out_v = 0.f;
          if (live) {
            out_v = acc[r][mi][nj][rg];
            if (bias) out_v += bias[co];
          }
          ob[t * Cout + co] = f2bf(out_v);
        }
      }
    }
  }
}

// ========================================================================
// host wrappers
// ========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static inline hipStream_t cur_stream3() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

torch::Tensor conv1d_cl_fused(torch::Tensor x, torch::Tensor w_perm,
                              c10::optional<torch::Tensor> bias, long Cout,
                              long k, long padding, long dilation,
                              double pre_lrelu, long act_mode,
                              double post_slope,
                              c10::optional<torch::Tensor> residual,
                              c10::optional<torch::Tensor> out_lens) {
  // x: [B, Tin, Cin] channel-last bf16; w_perm: [k][CoutP][CinP]
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "conv_cl: bf16 only");
  TORCH_CHECK(w_perm.dim() == 3 && w_perm.size(0) == k);
  const long B = x.size(0), Tin = x.size(1), Cin = x.size(2);
  TORCH_CHECK(Cin % 8 == 0, "conv_cl: Cin must be a multiple of 8");
  TORCH_CHECK((k - 1) * dilation <= HALO_MAX, "conv_cl: halo too large");
  const long Tout = Tin + 2 * padding - dilation * (k - 1);
  const int CoutP = w_perm.size(1), CinP = w_perm.size(2);
  auto out = torch::empty({B, Tout, Cout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat ? *bias
                                               : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const bf16* res_p = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->sizes() == out.sizes() && residual->is_contiguous());
    res_p = (const bf16*)residual->data_ptr();
  }
  const int* lens_p = nullptr;
  if (out_lens.has_value()) {
    TORCH_CHECK(out_lens->scalar_type() == at::kInt && out_lens->is_cuda());
    lens_p = out_lens->data_ptr<int>();
  }
  hipStream_t st = cur_stream3();
#define LAUNCH_CL(BM, BN, WGM, WGN, TC)                                     \
  hipLaunchKernelGGL((conv1d_cl_kernel<BM, BN, WGM, WGN, TC>),              \
                     dim3(ceil_div(Tout, BM), ceil_div(Cout, BN), B),       \
                     dim3(512), 0, st, (const bf16*)x.data_ptr(),           \
                     (const bf16*)w_perm.data_ptr(), bias_p,                \
                     (bf16*)out.data_ptr(), res_p, lens_p, (int)Cin, CinP,  \
                     (int)Cout, CoutP, Tin, Tout, (int)k, (int)dilation,    \
                     (int)padding, (float)pre_lrelu, (int)act_mode,         \
                     (float)post_slope)
  // BM=256 t-tiles: skinny-K GEMMs (K = Cin*k <= 1408) are barrier-
  // amortization-bound, so double the MFMA work per barrier window.
  if (Cout >= 128) LAUNCH_CL(256, 128, 4, 2, 2);
  else if (Cout >= 64) LAUNCH_CL(256, 64, 4, 2, 4);
  else LAUNCH_CL(256, 32, 4, 2, 4);
#undef LAUNCH_CL
  return out;
}

torch::Tensor conv1d_cl_wdirect(torch::Tensor x, torch::Tensor w_perm,
                                c10::optional<torch::Tensor> bias, long Cout,
                                long k, long padding, long dilation,
                                double pre_lrelu, long act_mode,
                                double post_slope,
                                c10::optional<torch::Tensor> residual,
                                c10::optional<torch::Tensor> out_lens) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  const long B = x.size(0), Tin = x.size(1), Cin = x.size(2);
  TORCH_CHECK((k - 1) * dilation <= HALO_MAX);
  const long Tout = Tin + 2 * padding - dilation * (k - 1);
  const int CoutP = w_perm.size(1), CinP = w_perm.size(2);
  auto out = torch::empty({B, Tout, Cout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat ? *bias
                                               : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const bf16* res_p = nullptr;
  if (residual.has_value()) res_p = (const bf16*)residual->data_ptr();
  const int* lens_p = nullptr;
  if (out_lens.has_value()) lens_p = out_lens->data_ptr<int>();
  hipStream_t st = cur_stream3();
#define LAUNCH_WD(BM, BN, WGM, WGN)                                         \
  hipLaunchKernelGGL((conv1d_cl_wdirect_kernel<BM, BN, WGM, WGN>),          \
                     dim3(ceil_div(Tout, BM), ceil_div(Cout, BN), B),       \
                     dim3(512), 0, st, (const bf16*)x.data_ptr(),           \
                     (const bf16*)w_perm.data_ptr(), bias_p,                \
                     (bf16*)out.data_ptr(), res_p, lens_p, (int)Cin, CinP,  \
                     (int)Cout, CoutP, Tin, Tout, (int)k, (int)dilation,    \
                     (int)padding, (float)pre_lrelu, (int)act_mode,         \
                     (float)post_slope)
  if (Cout >= 128) LAUNCH_WD(256, 128, 4, 2);
  else if (Cout >= 64) LAUNCH_WD(256, 64, 4, 2);
  else LAUNCH_WD(256, 32, 4, 2);
#undef LAUNCH_WD
  return out;
}

torch::Tensor convtranspose1d_cl_fused(torch::Tensor x, torch::Tensor w_perm,
                                       c10::optional<torch::Tensor> bias,
                                       long Cout, long k, long stride,
                                       long padding, double pre_lrelu,
                                       c10::optional<torch::Tensor> out_lens) {
  // x: [B, Tin, Cin] channel-last bf16; w_perm: [s][kr][CoutP][CinP];
  // requires k == 2*stride (the HiFi-GAN upsampler family).
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "convt_cl: bf16 only");
  TORCH_CHECK(w_perm.dim() == 4 && w_perm.size(0) == stride &&
              w_perm.size(1) == 2 && k == 2 * stride,
              "convt_cl: expects k == 2*stride phase layout");
  const long B = x.size(0), Tin = x.size(1), Cin = x.size(2);
  TORCH_CHECK(Cin % 8 == 0, "convt_cl: Cin must be a multiple of 8");
  const long Tout = (Tin - 1) * stride - 2 * padding + k;
  const int CoutP = w_perm.size(2), CinP = w_perm.size(3);
  auto out = torch::empty({B, Tout, Cout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat ? *bias
                                               : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const int* lens_p = nullptr;
  if (out_lens.has_value()) {
    TORCH_CHECK(out_lens->scalar_type() == at::kInt && out_lens->is_cuda());
    lens_p = out_lens->data_ptr<int>();
  }
  const long Vn = (Tout - 1 + padding) / stride + 1;
  hipStream_t st = cur_stream3();
#define LAUNCH_TCL(BMV, BN, WGM, WGN, S)                                    \
  hipLaunchKernelGGL((convt1d_cl_kernel<BMV, BN, WGM, WGN, S, 2>),          \
                     dim3(ceil_div(Vn, BMV), ceil_div(Cout, BN), B),        \
                     dim3(512), 0, st, (const bf16*)x.data_ptr(),           \
                     (const bf16*)w_perm.data_ptr(), bias_p,                \
                     (bf16*)out.data_ptr(), lens_p, (int)Cin, CinP,         \
                     (int)Cout, CoutP, Tin, Tout, (int)padding, Vn,         \
                     (float)pre_lrelu)
  if (stride == 8) {
    if (Cout >= 64) LAUNCH_TCL(128, 64, 4, 2, 8);
    else LAUNCH_TCL(128, 32, 4, 2, 8);
  } else if (stride == 2) {
    if (Cout >= 64) LAUNCH_TCL(128, 64, 4, 2, 2);
    else LAUNCH_TCL(128, 32, 4, 2, 2);
  } else {
    TORCH_CHECK(false, "convt_cl: unsupported stride ", stride);
  }
#undef LAUNCH_TCL
  return out;
}

// --------------------------------------------------------------------------
// Direct (zero-LDS, zero-barrier) channel-last conv — MEASURED NEGATIVE
// RESULT, kept as documentation: both MFMA operands load straight from
// global (16-B per-lane reads) so the staged kernel\'s barrier cost
// vanishes, but per-fragment VMEM issue + in-register LeakyReLU lose
// 1.6-2.6x vs the staged kernel on every resblock shape
// (gpurun A/B 2026-09-13: staged 323/689/470 TF vs direct 197/269/225 TF
// on C=128 k3 / C=128 k11d5 / C=256 k3).  The LDS-staged structure wins
// on this hardware; not dispatched anywhere.
// --------------------------------------------------------------------------
__device__ __forceinline__ bf16x8 lrelu8_(bf16x8 v, float slope) {
  bf16x8 r;
#pragma unroll
  for (int q = 0; q < 8; ++q) {
    float f = __bfloat162float(((__hip_bfloat16*)&v)[q]);
    ((__hip_bfloat16*)&r)[q] = __float2bfloat16(f > 0.f ? f : f * slope);
  }
  return r;
}

template <int BM, int BN, int WGM, int WGN>
__global__ __launch_bounds__(512) void conv1d_direct_cl_kernel(
    const bf16* __restrict__ x,     // [B][Tin][Cin]
    const bf16* __restrict__ w,     // [ntaps][CoutP][CinP]
    const float* __restrict__ bias,
    bf16* __restrict__ out,         // [B][Tout][Cout]
    const bf16* __restrict__ resid,
    const int* __restrict__ out_lens,
    int Cin, int CinP, int Cout, int CoutP, long Tin, long Tout,
    int ntaps, int dil, int pad, float pre_slope, int act_mode,
    float post_slope) {
  const long t0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int b = blockIdx.z;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  constexpr int WM = BM / WGM;
  constexpr int WN = BN / WGN;
  constexpr int MT = WM / 16;
  constexpr int NT = WN / 16;
  const int kl = lane >> 4;
  const int il = lane & 15;
  // conflict-free staging lane map (see resblock_cl.hip: quads keep a
  // row's 4 chunks contiguous for 64B coalescing; phase-group quads
  // cover rows {0,4,8,12}+g so LDS bank slots (5r+c)%16 = 4q+c)
  const int sr_l = 4 * ((lane >> 2) & 3) + (lane >> 4);
  const int sr_c = (lane & 3) * 8;
  const int sr_base0 = wid * 16;

  f32x4 acc[MT][NT];
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* xb = x + (long)b * Tin * Cin;
  // per-lane row bases (constant across the K loop)
  long trow[MT];
  bool tok_base[MT];
#pragma unroll
  for (int mi = 0; mi < MT; ++mi)
    trow[mi] = t0 + wr * WM + mi * 16 + il - pad;

  const int wcol = n0 + wc * WN + il;  // this lane's co base (per nj +16)

  for (int c0 = 0; c0 < Cin; c0 += 32) {
    for (int tap = 0; tap < ntaps; ++tap) {
      bf16x8 b_frag[NT];
#pragma unroll
      for (int nj = 0; nj < NT; ++nj)
        b_frag[nj] = *(const bf16x8*)&w[
            ((long)tap * CoutP + wcol + nj * 16) * CinP + c0 + kl * 8];
      const int toff = tap * dil;
#pragma unroll
      for (int mi = 0; mi < MT; ++mi) {
        const long t = trow[mi] + toff;
        // clamped load + post-select (no branch around the load)
        const long tc = t < 0 ? 0 : (t >= Tin ? Tin - 1 : t);
        bf16x8 a = *(const bf16x8*)&xb[tc * Cin + c0 + kl * 8];
        if (pre_slope >= 0.f) a = lrelu8_(a, pre_slope);
        if (t < 0 || t >= Tin) a = bf16x8{};
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b_frag[nj], acc[mi][nj], 0, 0, 0);
      }
    }
  }

  bf16* ob = out + (long)b * Tout * Cout;
  const bf16* rb = resid ? resid + (long)b * Tout * Cout : nullptr;
  const long lim = out_lens ? min((long)out_lens[b], Tout) : Tout;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const long t = t0 + wr * WM + mi * 16 + kl * 4 + rg;
      if (t >= Tout) continue;
      const bool live = t < lim;
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) {
        const int co = n0 + wc * WN + nj * 16 + il;
        if (co >= Cout) continue;
        float v = 0.f;
        if (live) {
          v = acc[mi][nj][rg];
          if (bias) v += bias[co];
          if (act_mode == ACT_LRELU) v = lrelu_(v, post_slope);
          else if (act_mode == ACT_TANH) v = tanhf(v);
          if (rb) v += bf2f(rb[t * Cout + co]);
        }
        ob[t * Cout + co] = f2bf(v);
      }
    }
  }
}

torch::Tensor conv1d_direct_cl(torch::Tensor x, torch::Tensor w_perm,
                               c10::optional<torch::Tensor> bias, long Cout,
                               long k, long padding, long dilation,
                               double pre_lrelu, long act_mode,
                               double post_slope,
                               c10::optional<torch::Tensor> residual,
                               c10::optional<torch::Tensor> out_lens) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  const long B = x.size(0), Tin = x.size(1), Cin = x.size(2);
  TORCH_CHECK(Cin % 32 == 0, "conv_direct: Cin must be a multiple of 32");
  const long Tout = Tin + 2 * padding - dilation * (k - 1);
  const int CoutP = w_perm.size(1), CinP = w_perm.size(2);
  auto out = torch::empty({B, Tout, Cout}, x.options());
  if (out.numel() == 0) return out;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat ? *bias
                                               : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const bf16* res_p = nullptr;
  if (residual.has_value()) res_p = (const bf16*)residual->data_ptr();
  const int* lens_p = nullptr;
  if (out_lens.has_value()) lens_p = out_lens->data_ptr<int>();
  hipStream_t st = cur_stream3();
#define LAUNCH_DIR(BM, BN, WGM, WGN)                                        \
  hipLaunchKernelGGL((conv1d_direct_cl_kernel<BM, BN, WGM, WGN>),           \
                     dim3(ceil_div(Tout, BM), ceil_div(Cout, BN), B),       \
                     dim3(512), 0, st, (const bf16*)x.data_ptr(),           \
                     (const bf16*)w_perm.data_ptr(), bias_p,                \
                     (bf16*)out.data_ptr(), res_p, lens_p, (int)Cin, CinP,  \
                     (int)Cout, CoutP, Tin, Tout, (int)k, (int)dilation,    \
                     (int)padding, (float)pre_lrelu, (int)act_mode,         \
                     (float)post_slope)
  if (Cout >= 128) LAUNCH_DIR(256, 128, 4, 2);
  else if (Cout >= 64) LAUNCH_DIR(256, 64, 4, 2);
  else LAUNCH_DIR(256, 32, 4, 2);
#undef LAUNCH_DIR
  return out;
}
