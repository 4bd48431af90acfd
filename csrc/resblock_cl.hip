// Fused HiFi-GAN ResBlock conv pair, channel-last (gfx950).
//
//   out = conv2_{k,d=1}( lrelu( conv1_{k,d}( lrelu(x) ) ) ) + x
//
// One kernel per pair instead of two: the intermediate tensor xt never
// touches HBM — conv1's output tile is written to LDS (pre-activated)
// and conv2 consumes it in place.  The small-channel resblock stages
// (C = 32..128 at T up to 256*F) are HBM-bandwidth-bound (measured
// 2.5-2.9 TB/s, profiles/r01_conv_pmc_v2.txt), so removing xt's
// write+read halves their traffic.
//
// Geometry: xt tile is a fixed 128 rows; the block's OUTPUT tile is
// BM = 128-(k-1) rows (conv2 consumes (k-1)/2 halo rows per side of xt,
// recomputed per block).  GEMM orientation identical to conv1d_cl.hip:
// A = time rows (k-contiguous channel-last), B = [co][ci] weight taps,
// K = Cin in 32-deep LDS slices.
#include "common.h"

#define BK 32
#define BKP 40
#define XTROWS 128

template <int BN, int WGN, int TC, int XR, int XTR>
__global__ __launch_bounds__(512) void resblock_pair_cl_kernel(
    const bf16* __restrict__ x,    // [B][T][C]
    const bf16* __restrict__ w1,   // [k][CP][CP] (dilated conv)
    const float* __restrict__ b1,  // [C]
    const bf16* __restrict__ w2,   // [k][CP][CP] (d=1 conv)
    const float* __restrict__ b2,  // [C]
    bf16* __restrict__ out,        // [B][T][C]
    const bf16* __restrict__ accum,  // optional: add (xs running sum)
    const int* __restrict__ out_lens,
    int C, int CP, long T, int k, int dil, float out_scale) {
  constexpr int WGM = 4;
  constexpr int WM = XTR / WGM;
  constexpr int MT = WM / 16;
  constexpr int WN = BN / WGN;
  constexpr int NT = WN / 16;
  constexpr int XROWS_MAX = XR;     // 128 + (k-1)*dil, bucketed
  constexpr int XTP = BN + 8;       // xt row pitch: 16B-aligned rows,
                                    // 16 distinct banks across il lanes

  const int h2 = (k - 1) / 2;        // conv2 halo per side
  const int BM = XTR - (k - 1);      // output rows per block
  // XCD-aware swizzle: consecutive T-tiles land on the SAME XCD so halo
  // rows and the weight tensor stay hot in that XCD\'s L2 (the hardware
  // round-robins blockIdx across the 8 XCDs).  Bijective remap per the
  // CDNA4 guide (q/r split handles nwg % 8 != 0).
  const int nwg = gridDim.x;
  int tile = blockIdx.x;
  if (nwg > 8) {
    const int xcd = tile % 8, orig8 = tile / 8;
    const int q = nwg / 8, r = nwg % 8;
    tile = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig8;
  }
  const long t0 = (long)tile * BM;
  const int b = blockIdx.z;

  __shared__ bf16 Xs[XROWS_MAX][BKP];
  __shared__ bf16 Xt[XTR][XTP];
  __shared__ bf16 Ws[TC][BN][BKP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WGN;
  const int wc = wid % WGN;
  const int kl = lane >> 4;
  const int il = lane & 15;

  const bf16* xb = x + (long)b * T * C;
  const int pad1 = (k - 1) * dil / 2;
  // Xs[0] holds x row (t0 - h2 - pad1); GEMM1 xt row m taps rows m+j*dil
  const long row0 = t0 - h2 - pad1;
  const int xrows = XTR + (k - 1) * dil;
  const bool t_interior = (row0 >= 0) && (row0 + xrows <= T);

  // ================= GEMM1: xt = lrelu(conv1(lrelu(x))) ================
  f32x4 acc[MT][NT];
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // Staging lane map: each QUAD (4 lanes) keeps one row's 4 chunks
  // contiguous (64B global coalescing), but the 4 quads of a 16-lane
  // LDS phase group cover rows {0,4,8,12}+g instead of {0,1,2,3}: at
  // BKP=40 (row stride 20 banks) the bank slot is (5r+c) mod 16, and
  // rows 4 apart give slots 4q+c -- a clean permutation, so staging
  // writes are conflict-free (the row=u>>2 map collided rows 0 and 3
  // of each group; profiles/r02_pmc_rbpair.txt: LDS issue-stall 17%).
  const int sr_l = 4 * ((lane >> 2) & 3) + (lane >> 4);  // row-in-group
  const int sr_c = (lane & 3) * 8;     // staging chunk (bf16 offset)
  const int sr_base0 = wid * 16;       // wave's first row; stride 128

  for (int c0 = 0; c0 < CP; c0 += BK) {
    const bool c_interior = (c0 + BK) <= C;
    if (t_interior && c_interior) {
      for (int base = sr_base0; base < xrows; base += 128) {
        const int r = base + sr_l;
        if (r < xrows) {
          bf16 v8[8];
          *(ulonglong2*)v8 =
              *(const ulonglong2*)&xb[(row0 + r) * C + c0 + sr_c];
#pragma unroll
          for (int q = 0; q < 8; ++q)
            v8[q] = f2bf(lrelu_(bf2f(v8[q]), 0.1f));
          *(ulonglong2*)&Xs[r][sr_c] = *(ulonglong2*)v8;
        }
      }
    } else {
      for (int base = sr_base0; base < xrows; base += 128) {
        const int r = base + sr_l;
        if (r >= xrows) continue;
        const long t = row0 + r;
        bf16 v8[8];
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          const int c = c0 + sr_c + q;
          float v = (t >= 0 && t < T && c < C) ? bf2f(xb[t * C + c]) : 0.f;
          v8[q] = f2bf(lrelu_(v, 0.1f));
        }
        *(ulonglong2*)&Xs[r][sr_c] = *(ulonglong2*)v8;
      }
    }
    for (int tap0 = 0; tap0 < k; tap0 += TC) {
      const int ntc = min(TC, k - tap0);
      for (int tc = 0; tc < ntc; ++tc) {
        const long wbase = ((long)(tap0 + tc) * CP) * CP + c0;
        for (int base = sr_base0; base < BN; base += 128) {
          const int n = base + sr_l;
          *(ulonglong2*)&Ws[tc][n][sr_c] =
              *(const ulonglong2*)&w1[wbase + (long)n * CP + sr_c];
        }
      }
      __syncthreads();
      for (int tc = 0; tc < ntc; ++tc) {
        const int toff = (tap0 + tc) * dil;
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] =
              *(const bf16x8*)&Ws[tc][wc * WN + nj * 16 + il][kl * 8];
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

  // epilogue1 -> LDS xt (bias1 + lrelu; zero outside [0,T) and >= lim)
  const long lim = out_lens ? min((long)out_lens[b], T) : T;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const int m = wr * WM + mi * 16 + kl * 4 + rg;
      const long t = t0 - h2 + m;
      const bool live = (t >= 0) && (t < lim);
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) {
        const int co = wc * WN + nj * 16 + il;
        float v = 0.f;
        if (live && co < C)
          v = lrelu_(acc[mi][nj][rg] + b1[co], 0.1f);
        Xt[m][co] = f2bf(v);
      }
    }
  }
  __syncthreads();

  // ================= GEMM2: out = conv2(xt) + x =======================
#pragma unroll
  for (int i = 0; i < MT; ++i)
#pragma unroll
    for (int j = 0; j < NT; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int c0 = 0; c0 < CP; c0 += BK) {
    for (int tap0 = 0; tap0 < k; tap0 += TC) {
      const int ntc = min(TC, k - tap0);
      for (int tc = 0; tc < ntc; ++tc) {
        const long wbase = ((long)(tap0 + tc) * CP) * CP + c0;
        for (int base = sr_base0; base < BN; base += 128) {
          const int n = base + sr_l;
          *(ulonglong2*)&Ws[tc][n][sr_c] =
              *(const ulonglong2*)&w2[wbase + (long)n * CP + sr_c];
        }
      }
      __syncthreads();
      for (int tc = 0; tc < ntc; ++tc) {
        const int toff = tap0 + tc;  // d=1
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] =
              *(const bf16x8*)&Ws[tc][wc * WN + nj * 16 + il][kl * 8];
#pragma unroll
        for (int mi = 0; mi < MT; ++mi) {
          // conv2 out row m2 taps xt rows m2 + toff (xt row 0 = t0-h2)
          const bf16x8 a_frag =
              *(const bf16x8*)&Xt[wr * WM + mi * 16 + il + toff][c0 + kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj)
            acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
        }
      }
      __syncthreads();
    }
  }

  // epilogue2: bias2 + residual x (+ xs accumulation) + scale + mask
  bf16* ob = out + (long)b * T * C;
  const bf16* ab = accum ? accum + (long)b * T * C : nullptr;
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const int m2 = wr * WM + mi * 16 + kl * 4 + rg;
      const long t = t0 + m2;
      if (m2 >= BM || t >= T) continue;
      const bool live = t < lim;
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) {
        const int co = wc * WN + nj * 16 + il;
        if (co >= C) continue;
        float v = 0.f;
        if (live) {
          v = acc[mi][nj][rg] + b2[co] + bf2f(xb[t * C + co]);
          if (ab) v += bf2f(ab[t * C + co]);
          v *= out_scale;
        }
        ob[t * C + co] = f2bf(v);
      }
    }
  }
}

// ========================================================================
// Persistent small-C pair kernel: BOTH conv weights LDS-resident.
//
// The C=32/64 resblock stages run at huge T (F*128 / F*256 samples) with
// tiny per-window MFMA work (~8-16 MFMA between barriers in the generic
// kernel): they are BARRIER/overhead-bound, not compute-bound.  Here one
// block per CU stays resident, stages w1+w2 into LDS ONCE (fits for
// CP=32 any k, CP=64 k=3: 2*k*CP*(CP+8)*2B <= 56 KB), and loops over
// (b, t-tile)s with the whole K dim resident per row: each GEMM then
// runs ALL taps x K-slices with ZERO interior barriers - 3 barriers per
// tile total (vs ~24 in the generic kernel at k=11).
// ========================================================================
#define PERSIST_XTR 256
#define PERSIST_XTROWS (PERSIST_XTR + 10)  // + (k-1) for k<=11

template <int CP_T, int KMAX, int XR>
__global__ __launch_bounds__(512) void resblock_pair_persist_kernel(
    const bf16* __restrict__ x,    // [B][T][C]
    const bf16* __restrict__ w1,   // [k][CP][CP]
    const float* __restrict__ b1,
    const bf16* __restrict__ w2,   // [k][CP][CP]
    const float* __restrict__ b2,
    bf16* __restrict__ out,
    const bf16* __restrict__ accum,
    const int* __restrict__ out_lens,
    int C, long T, int k, int dil, float out_scale, int tiles_per_b,
    int total_tiles) {
  constexpr int CPP = CP_T + 8;        // row pitch (40 / 72: conflict-free)
  constexpr int NT = CP_T / 16;
  constexpr int KS = CP_T / 32;        // K slices per row
  constexpr int MTMAX = 3;             // ceil(266/16/8) wave row-tiles

  __shared__ bf16 Ws1[KMAX][CP_T][CPP];
  __shared__ bf16 Ws2[KMAX][CP_T][CPP];
  __shared__ bf16 Xs[XR][CPP];
  __shared__ bf16 Xt[PERSIST_XTROWS][CPP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int kl = lane >> 4;
  const int il = lane & 15;

  // ---- stage both weight tensors once ------------------------------- //
  const int wchunks = k * CP_T * (CP_T / 8);
  for (int u = tid; u < wchunks; u += 512) {
    const int tap = u / (CP_T * CP_T / 8);
    const int rem = u % (CP_T * CP_T / 8);
    const int n = rem / (CP_T / 8), ch = (rem % (CP_T / 8)) * 8;
    const long src = ((long)tap * CP_T + n) * CP_T + ch;
    *(ulonglong2*)&Ws1[tap][n][ch] = *(const ulonglong2*)&w1[src];
    *(ulonglong2*)&Ws2[tap][n][ch] = *(const ulonglong2*)&w2[src];
  }

  const int h2 = (k - 1) / 2;
  const int BM = PERSIST_XTR - (k - 1);
  const int xtrows = PERSIST_XTR + (k - 1);
  const int pad1 = (k - 1) * dil / 2;
  const int xrows = xtrows + (k - 1) * dil;
  const int mtiles1 = (xtrows + 15) / 16;   // GEMM1 row tiles (<= 17)
  const int mtiles2 = (BM + 15) / 16;       // GEMM2 row tiles (<= 16)

  for (int idx = blockIdx.x; idx < total_tiles; idx += gridDim.x) {
    const int b = idx / tiles_per_b;
    const long t0 = (long)(idx % tiles_per_b) * BM;
    const bf16* xb = x + (long)b * T * C;
    const long row0 = t0 - h2 - pad1;
    const long lim = out_lens ? min((long)out_lens[b], T) : T;

    __syncthreads();  // prior tile's GEMM2/Xt reads complete
    // ---- stage x rows (full K per row, pre-lrelu) ------------------- //
    const bool interior = (row0 >= 0) && (row0 + xrows <= T);
    if (interior) {
      for (int u = tid; u < xrows * (CP_T / 8); u += 512) {
        const int r = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
        bf16 v8[8];
        *(ulonglong2*)v8 = *(const ulonglong2*)&xb[(row0 + r) * C + ch];
#pragma unroll
        for (int q = 0; q < 8; ++q) v8[q] = f2bf(lrelu_(bf2f(v8[q]), 0.1f));
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    } else {
      for (int u = tid; u < xrows * (CP_T / 8); u += 512) {
        const int r = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
        const long t = row0 + r;
        bf16 v8[8];
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          float v = (t >= 0 && t < T) ? bf2f(xb[t * C + ch + q]) : 0.f;
          v8[q] = f2bf(lrelu_(v, 0.1f));
        }
        *(ulonglong2*)&Xs[r][ch] = *(ulonglong2*)v8;
      }
    }
    __syncthreads();

    // ---- GEMM1: xt = lrelu(b1 + conv1(Xs)) — zero interior barriers - //
    f32x4 acc[MTMAX][NT];
    for (int mt = wid, mi = 0; mt < mtiles1; mt += 8, ++mi) {
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) acc[mi][nj] = {0.f, 0.f, 0.f, 0.f};
    }
    for (int tap = 0; tap < k; ++tap) {
      const int toff = tap * dil;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] = *(const bf16x8*)&Ws1[tap][nj * 16 + il][ks * 32 + kl * 8];
        for (int mt = wid, mi = 0; mt < mtiles1; mt += 8, ++mi) {
          const int m = mt * 16 + il;
          const bf16x8 a_frag =
              *(const bf16x8*)&Xs[min(m + toff, XR - 1)][ks * 32 + kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj)
            acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
        }
      }
    }
    // epilogue1 -> Xt
    for (int mt = wid, mi = 0; mt < mtiles1; mt += 8, ++mi) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const int m = mt * 16 + kl * 4 + rg;
        if (m >= xtrows) continue;
        const long t = t0 - h2 + m;
        const bool live = (t >= 0) && (t < lim);
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) {
          const int co = nj * 16 + il;
          float v = 0.f;
          if (live && co < C) v = lrelu_(acc[mi][nj][rg] + b1[co], 0.1f);
          Xt[m][co] = f2bf(v);
        }
      }
    }
    __syncthreads();

    // ---- GEMM2: out = b2 + conv2(Xt) + x (+accum) ------------------- //
    for (int mt = wid, mi = 0; mt < mtiles2; mt += 8, ++mi) {
#pragma unroll
      for (int nj = 0; nj < NT; ++nj) acc[mi][nj] = {0.f, 0.f, 0.f, 0.f};
    }
    for (int tap = 0; tap < k; ++tap) {
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        bf16x8 b_frag[NT];
#pragma unroll
        for (int nj = 0; nj < NT; ++nj)
          b_frag[nj] = *(const bf16x8*)&Ws2[tap][nj * 16 + il][ks * 32 + kl * 8];
        for (int mt = wid, mi = 0; mt < mtiles2; mt += 8, ++mi) {
          const int m = mt * 16 + il;
          const bf16x8 a_frag =
              *(const bf16x8*)&Xt[min(m + tap, PERSIST_XTROWS - 1)]
                                 [ks * 32 + kl * 8];
#pragma unroll
          for (int nj = 0; nj < NT; ++nj)
            acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
        }
      }
    }
    bf16* ob = out + (long)b * T * C;
    const bf16* ab = accum ? accum + (long)b * T * C : nullptr;
    for (int mt = wid, mi = 0; mt < mtiles2; mt += 8, ++mi) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const int m2 = mt * 16 + kl * 4 + rg;
        const long t = t0 + m2;
        if (m2 >= BM || t >= T) continue;
        const bool live = t < lim;
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) {
          const int co = nj * 16 + il;
          if (co >= C) continue;
          float v = 0.f;
          if (live) {
            v = acc[mi][nj][rg] + b2[co] + bf2f(xb[t * C + co]);
            if (ab) v += bf2f(ab[t * C + co]);
            v *= out_scale;
          }
          ob[t * C + co] = f2bf(v);
        }
      }
    }
  }
}

// ========================================================================
// Whole-resblock CHAIN kernel: the three conv pairs of one HiFi-GAN
// ResBlock (dilations d1,d2,d3, same k) fused into ONE kernel.
//
// Why: the C=32 / C=64-k3 pair shapes are HBM-bound (1.6-2.7 TB/s
// measured); running the pairs separately reads + writes the full
// [B,T,C] tensor 3x (in, out, residual re-read).  Chaining keeps every
// intermediate in LDS: global traffic drops ~3x per resblock at the
// cost of halo recompute (stage rows grow by (k-1)*(d_i+1) per pair).
//
// Same-origin indexing: all LDS buffers map row r -> time row0+r, so
// pair p's residual is simply its input buffer at the same row; valid
// rows shrink toward the final [LEAD, LEAD+BM) window and edge garbage
// never reaches it (halo arithmetic below).
//
// Buffers hold PRE-ACTIVATED (lrelu) values — the next GEMM's A operand
// needs them — and the residual reconstructs the raw value through the
// exact-ish inverse (z>=0 ? z : 10z; bf16 storage makes the negative
// branch differ by <=2^-8 relative, inside kernel parity tolerance).
// 3 rotating buffers + per-window W staging: 42-76 KB -> 2-3 blocks/CU
// (the cross-block overlap the persistent experiment showed is vital).
// ========================================================================
#define CHAIN_XTR 128

__device__ __forceinline__ float inv_lrelu_(float z) {
  return z >= 0.0f ? z : z * 10.0f;
}

template <int CP_T, int SR>
__global__ __launch_bounds__(512) void resblock_chain_cl_kernel(
    const bf16* __restrict__ x,       // [B][T][C]
    const bf16* __restrict__ w_all,   // [6][k][CP][CP] (w1,w2 per pair)
    const float* __restrict__ b_all,  // [6][C]
    bf16* __restrict__ out,           // [B][T][C]
    const bf16* __restrict__ accum,
    const int* __restrict__ out_lens,
    int C, long T, int k, int d1, int d2, int d3, float out_scale) {
  constexpr int CPP = CP_T + 8;
  constexpr int NT = CP_T / 16;
  constexpr int KS = CP_T / 32;
  constexpr int MTMAX = 2;  // ceil(SR/16)/8 waves, SR <= 256
  constexpr int TC = 2;

  const int km1 = k - 1;
  const int BM = CHAIN_XTR;
  const int LEAD = km1 * (d1 + d2 + d3 + 3) / 2;   // left halo total
  const int S0 = BM + 2 * LEAD;                    // x rows staged

  // XCD swizzle (bijective; see pair kernel above)
  const int nwg = gridDim.x;
  int tile = blockIdx.x;
  if (nwg > 8) {
    const int xcd = tile % 8, orig8 = tile / 8;
    const int q = nwg / 8, r = nwg % 8;
    tile = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig8;
  }
  const long t0 = (long)tile * BM;
  const int b = blockIdx.z;
  const long row0 = t0 - LEAD;

  __shared__ bf16 Buf[3][SR][CPP];
  __shared__ bf16 Ws[TC][CP_T][CPP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int kl = lane >> 4;
  const int il = lane & 15;

  const bf16* xb = x + (long)b * T * C;
  const long lim = out_lens ? min((long)out_lens[b], T) : T;

  // ---- stage x rows (pre-activated, full K per row) ------------------ //
  const bool interior = (row0 >= 0) && (row0 + S0 <= T);
  if (interior) {
    for (int u = tid; u < S0 * (CP_T / 8); u += 512) {
      const int r = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
      bf16 v8[8];
      *(ulonglong2*)v8 = *(const ulonglong2*)&xb[(row0 + r) * C + ch];
#pragma unroll
      for (int q = 0; q < 8; ++q) v8[q] = f2bf(lrelu_(bf2f(v8[q]), 0.1f));
      *(ulonglong2*)&Buf[0][r][ch] = *(ulonglong2*)v8;
    }
  } else {
    for (int u = tid; u < S0 * (CP_T / 8); u += 512) {
      const int r = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
      const long t = row0 + r;
      bf16 v8[8];
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float v = (t >= 0 && t < T) ? bf2f(xb[t * C + ch + q]) : 0.f;
        v8[q] = f2bf(lrelu_(v, 0.1f));
      }
      *(ulonglong2*)&Buf[0][r][ch] = *(ulonglong2*)v8;
    }
  }

  const int mtiles = (S0 + 15) / 16;
  int in_buf = 0;  // holds lrelu(pair input)
  const int dils[3] = {d1, d2, d3};

  for (int pair = 0; pair < 3; ++pair) {
    const int dil = dils[pair];
    const int xt_buf = (in_buf + 1) % 3;
    const int out_buf = (in_buf + 2) % 3;
    const bf16* w1 = w_all + ((long)(2 * pair) * k) * CP_T * CP_T;
    const bf16* w2 = w_all + ((long)(2 * pair + 1) * k) * CP_T * CP_T;
    const float* b1 = b_all + (2 * pair) * C;
    const float* b2 = b_all + (2 * pair + 1) * C;

    // ---- GEMM_a: xt = lrelu(b1 + conv1_{k,dil}(Buf[in])) ------------- //
    {
      f32x4 acc[MTMAX][NT];
      for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi)
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) acc[mi][nj] = {0.f, 0.f, 0.f, 0.f};
      const int pad = km1 * dil / 2;
      for (int tap0 = 0; tap0 < k; tap0 += TC) {
        const int ntc = min(TC, k - tap0);
        for (int tc = 0; tc < ntc; ++tc) {
          const long wbase = ((long)(tap0 + tc) * CP_T) * CP_T;
          for (int u = tid; u < CP_T * (CP_T / 8); u += 512) {
            const int n = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
            *(ulonglong2*)&Ws[tc][n][ch] =
                *(const ulonglong2*)&w1[wbase + (long)n * CP_T + ch];
          }
        }
        __syncthreads();
        for (int tc = 0; tc < ntc; ++tc) {
          const int toff = (tap0 + tc) * dil - pad;
#pragma unroll
          for (int ks = 0; ks < KS; ++ks) {
            bf16x8 b_frag[NT];
#pragma unroll
            for (int nj = 0; nj < NT; ++nj)
              b_frag[nj] =
                  *(const bf16x8*)&Ws[tc][nj * 16 + il][ks * 32 + kl * 8];
            for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi) {
              const int m = mt * 16 + il;
              const int src = min(max(m + toff, 0), S0 - 1);
              const bf16x8 a_frag =
                  *(const bf16x8*)&Buf[in_buf][src][ks * 32 + kl * 8];
#pragma unroll
              for (int nj = 0; nj < NT; ++nj)
                acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
            }
          }
        }
        __syncthreads();
      }
      // epilogue_a -> Buf[xt] (pre-activated)
      for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi) {
#pragma unroll
        for (int rg = 0; rg < 4; ++rg) {
          const int m = mt * 16 + kl * 4 + rg;
          if (m >= S0) continue;
          const long t = row0 + m;
          const bool live = (t >= 0) && (t < lim);
#pragma unroll
          for (int nj = 0; nj < NT; ++nj) {
            const int co = nj * 16 + il;
            float v = 0.f;
            if (live && co < C) v = lrelu_(acc[mi][nj][rg] + b1[co], 0.1f);
            Buf[xt_buf][m][co] = f2bf(v);
          }
        }
      }
      __syncthreads();
    }

    // ---- GEMM_b: out = b2 + conv2_{k,1}(xt) + raw(Buf[in]) ----------- //
    {
      f32x4 acc[MTMAX][NT];
      for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi)
#pragma unroll
        for (int nj = 0; nj < NT; ++nj) acc[mi][nj] = {0.f, 0.f, 0.f, 0.f};
      const int pad2 = km1 / 2;
      for (int tap0 = 0; tap0 < k; tap0 += TC) {
        const int ntc = min(TC, k - tap0);
        for (int tc = 0; tc < ntc; ++tc) {
          const long wbase = ((long)(tap0 + tc) * CP_T) * CP_T;
          for (int u = tid; u < CP_T * (CP_T / 8); u += 512) {
            const int n = u / (CP_T / 8), ch = (u % (CP_T / 8)) * 8;
            *(ulonglong2*)&Ws[tc][n][ch] =
                *(const ulonglong2*)&w2[wbase + (long)n * CP_T + ch];
          }
        }
        __syncthreads();
        for (int tc = 0; tc < ntc; ++tc) {
          const int toff = (tap0 + tc) - pad2;
#pragma unroll
          for (int ks = 0; ks < KS; ++ks) {
            bf16x8 b_frag[NT];
#pragma unroll
            for (int nj = 0; nj < NT; ++nj)
              b_frag[nj] =
                  *(const bf16x8*)&Ws[tc][nj * 16 + il][ks * 32 + kl * 8];
            for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi) {
              const int m = mt * 16 + il;
              const int src = min(max(m + toff, 0), S0 - 1);
              const bf16x8 a_frag =
                  *(const bf16x8*)&Buf[xt_buf][src][ks * 32 + kl * 8];
#pragma unroll
              for (int nj = 0; nj < NT; ++nj)
                acc[mi][nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, b_frag[nj], acc[mi][nj], 0, 0, 0);
            }
          }
        }
        __syncthreads();
      }
      const bool last_pair = pair == 2;
      bf16* ob = out + (long)b * T * C;
      const bf16* ab = accum ? accum + (long)b * T * C : nullptr;
      for (int mt = wid, mi = 0; mt < mtiles; mt += 8, ++mi) {
#pragma unroll
        for (int rg = 0; rg < 4; ++rg) {
          const int m = mt * 16 + kl * 4 + rg;
          if (m >= S0) continue;
          const long t = row0 + m;
          const bool live = (t >= 0) && (t < lim);
#pragma unroll
          for (int nj = 0; nj < NT; ++nj) {
            const int co = nj * 16 + il;
            if (co >= C) continue;
            float v = 0.f;
            if (live) {
              const float resid = inv_lrelu_(bf2f(Buf[in_buf][m][co]));
              v = acc[mi][nj][rg] + b2[co] + resid;
            }
            if (!last_pair) {
              Buf[out_buf][m][co] = f2bf(lrelu_(v, 0.1f));
            } else if (m >= LEAD && m < LEAD + BM && t >= t0 && t < T) {
              float o = v;
              if (live && ab) o += bf2f(ab[t * C + co]);
              if (live) o *= out_scale;
              ob[t * C + co] = f2bf(live ? o : 0.f);
            }
          }
        }
      }
      __syncthreads();
    }
    in_buf = out_buf;
  }
}

// ========================================================================
// host wrapper
// ========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstring>

static inline hipStream_t cur_stream4() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

torch::Tensor resblock_chain_cl_fused(
    torch::Tensor x, torch::Tensor w_all, torch::Tensor b_all, long k,
    long d1, long d2, long d3, c10::optional<torch::Tensor> out_lens,
    c10::optional<torch::Tensor> accum, double out_scale) {
  // x: [B, T, C] bf16; w_all: [6][k][CP][CP] (w1,w2 per pair, permuted);
  // b_all: [6][C] f32
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "chain: bf16 only");
  TORCH_CHECK(w_all.dim() == 4 && w_all.size(0) == 6 && w_all.size(1) == k);
  TORCH_CHECK(w_all.is_contiguous() && b_all.is_contiguous());
  TORCH_CHECK(b_all.scalar_type() == at::kFloat);
  const long B = x.size(0), T = x.size(1), C = x.size(2);
  const long CP = w_all.size(3);
  TORCH_CHECK(w_all.size(2) == CP && CP == C, "chain: square channels");
  const long S0 = CHAIN_XTR + (k - 1) * (d1 + d2 + d3 + 3);
  TORCH_CHECK((CP == 32 && S0 <= 248) || (CP == 64 && S0 <= 152),
              "chain: unsupported geometry");
  auto out = torch::empty_like(x);
  if (out.numel() == 0) return out;
  const int* lens_p = nullptr;
  if (out_lens.has_value()) {
    TORCH_CHECK(out_lens->scalar_type() == at::kInt && out_lens->is_cuda());
    lens_p = out_lens->data_ptr<int>();
  }
  const bf16* accum_p = nullptr;
  if (accum.has_value()) {
    TORCH_CHECK(accum->sizes() == x.sizes() && accum->is_contiguous());
    accum_p = (const bf16*)accum->data_ptr();
  }
  hipStream_t st = cur_stream4();
  dim3 grid(ceil_div(T, CHAIN_XTR), 1, B);
#define LAUNCH_CHAIN(CPT, SRB)                                              \
  hipLaunchKernelGGL((resblock_chain_cl_kernel<CPT, SRB>), grid, dim3(512), \
                     0, st, (const bf16*)x.data_ptr(),                      \
                     (const bf16*)w_all.data_ptr(),                         \
                     b_all.data_ptr<float>(), (bf16*)out.data_ptr(),        \
                     accum_p, lens_p, (int)C, T, (int)k, (int)d1, (int)d2,  \
                     (int)d3, (float)out_scale)
  if (CP == 64) LAUNCH_CHAIN(64, 152);
  else if (S0 <= 152) LAUNCH_CHAIN(32, 152);
  else if (S0 <= 200) LAUNCH_CHAIN(32, 200);
  else LAUNCH_CHAIN(32, 248);
#undef LAUNCH_CHAIN
  return out;
}

torch::Tensor resblock_pair_cl_fused(torch::Tensor x, torch::Tensor w1_perm,
                                     torch::Tensor b1, torch::Tensor w2_perm,
                                     torch::Tensor b2, long k, long dil,
                                     c10::optional<torch::Tensor> out_lens,
                                     c10::optional<torch::Tensor> accum,
                                     double out_scale) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "resblock_cl: bf16 only");
  const long B = x.size(0), T = x.size(1), C = x.size(2);
  TORCH_CHECK(w1_perm.size(0) == k && w2_perm.size(0) == k);
  const int CP = w1_perm.size(2);
  TORCH_CHECK(w1_perm.size(1) == CP && w2_perm.size(1) == CP &&
              w2_perm.size(2) == CP, "resblock_cl: square channel conv");
  TORCH_CHECK((k - 1) * dil <= 52, "resblock_cl: halo too large");
  auto out = torch::empty_like(x);
  if (out.numel() == 0) return out;
  auto b1f = b1.scalar_type() == at::kFloat ? b1 : b1.to(at::kFloat).contiguous();
  auto b2f = b2.scalar_type() == at::kFloat ? b2 : b2.to(at::kFloat).contiguous();
  const int* lens_p = nullptr;
  if (out_lens.has_value()) {
    TORCH_CHECK(out_lens->scalar_type() == at::kInt && out_lens->is_cuda());
    lens_p = out_lens->data_ptr<int>();
  }
  const bf16* accum_p = nullptr;
  if (accum.has_value()) {
    TORCH_CHECK(accum->sizes() == x.sizes() && accum->is_contiguous());
    accum_p = (const bf16*)accum->data_ptr();
  }
  hipStream_t st = cur_stream4();
  // Persistent W-resident variant (opt-in, SONATA_PERSIST_RB=1):
  // DOCUMENTED NEGATIVE RESULT.  Both weight tensors LDS-resident, each
  // GEMM runs tap x K with zero interior barriers (3 barriers/tile vs
  // ~24) - but measured 2.4-16x SLOWER than the generic kernel
  // (profiles/r02_rbpair_persist_ab.log): one serial block per CU
  // (LDS-locked at 99-133 KB) exposes the x staging latency and every
  // barrier drain that the generic kernel's 2-block-per-CU overlap
  // hides.  Matches round-1's pipelining findings: cross-block overlap
  // beats intra-block overhead elimination at these window sizes.
  // Kept for documentation + as the substrate for a future prefetched
  // variant; parity-tested (ragged/accum/scale) under the env flag.
  const char* persist_env = getenv("SONATA_PERSIST_RB");
  const bool persist_on = persist_env && persist_env[0] == '1';
  static int n_cu = 0;
  if (persist_on && n_cu == 0) {
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, 0));
    n_cu = prop.multiProcessorCount;
  }
  const bool persist_ok =
      persist_on && ((CP == 32) || (CP == 64 && k == 3));
  if (persist_ok) {
    const int BMp = 256 - (int)(k - 1);
    const int tiles_per_b = (int)((T + BMp - 1) / BMp);
    const long total = (long)B * tiles_per_b;
    if (total >= n_cu) {
      const int grid = (int)std::min<long>(total, n_cu);
      const int xrows = 256 + (int)(k - 1) * (1 + (int)dil);
#define LAUNCH_PERSIST(CPT, KM, XR)                                         \
  hipLaunchKernelGGL((resblock_pair_persist_kernel<CPT, KM, XR>),           \
                     dim3(grid), dim3(512), 0, st,                          \
                     (const bf16*)x.data_ptr(),                             \
                     (const bf16*)w1_perm.data_ptr(),                       \
                     b1f.data_ptr<float>(),                                 \
                     (const bf16*)w2_perm.data_ptr(),                       \
                     b2f.data_ptr<float>(), (bf16*)out.data_ptr(), accum_p, \
                     lens_p, (int)C, T, (int)k, (int)dil,                   \
                     (float)out_scale, tiles_per_b, (int)total)
      if (CP == 64) LAUNCH_PERSIST(64, 3, 272);
      else if (xrows <= 268) LAUNCH_PERSIST(32, 11, 268);
      else if (xrows <= 292) LAUNCH_PERSIST(32, 11, 292);
      else LAUNCH_PERSIST(32, 11, 316);
#undef LAUNCH_PERSIST
      return out;
    }
  }
  // small-C stages run at huge T with tiny per-block work: use taller
  // 256-row xt tiles there (2x MFMA per block, occupancy still 2-3).
  // SONATA_RB_GEOM=xtr64 (experiment): 64-row xt tiles for C=128/256 —
  // Xt shrinks 2x (C=256's Xt[128][264]=66 KB locks it to ONE block/CU
  // today), trading W-restage traffic (L2-hot) for occupancy.
  // Geometry per channel class (A/B in profiles/r02_rb256_ab.log):
  //   C=256: 64-row xt tiles BY DEFAULT — Xt[128][264]=66 KB locked the
  //     128-row variant to ONE block/CU (no cross-block overlap);
  //     xt64 runs 2 blocks/CU: +24% k3, +6% k7, +1% k11 (medians).
  //   C=128: 128-row stays — xt64 measured 10-38% SLOWER there.
  // SONATA_RB_GEOM={xtr64,xtr128} overrides for experiments.
  const char* geom_env = getenv("SONATA_RB_GEOM");
  bool geom64 = CP >= 256;
  if (geom_env && strcmp(geom_env, "xtr64") == 0) geom64 = CP >= 128;
  if (geom_env && strcmp(geom_env, "xtr128") == 0) geom64 = false;
  const long XTRh = (CP <= 32) ? 256 : (geom64 ? 64 : 128);
  const long BM = XTRh - (k - 1);
#define LAUNCH_RB(BN, WGN, TC, XR, XTR)                                     \
  hipLaunchKernelGGL((resblock_pair_cl_kernel<BN, WGN, TC, XR, XTR>),       \
                     dim3(ceil_div(T, BM), 1, B), dim3(512), 0, st,         \
                     (const bf16*)x.data_ptr(),                             \
                     (const bf16*)w1_perm.data_ptr(),                       \
                     b1f.data_ptr<float>(),                                 \
                     (const bf16*)w2_perm.data_ptr(),                       \
                     b2f.data_ptr<float>(), (bf16*)out.data_ptr(), accum_p, \
                     lens_p, (int)C, CP, T, (int)k, (int)dil,               \
                     (float)out_scale)
  // XR bucket = 128 + (k-1)*dil rounded up; TC=3 covers k=3 in one
  // chunk and k=7/11 in 3/4 chunks (fewer barrier pairs); C=256 keeps
  // TC=2 for LDS.
  const int xrows = (int)XTRh + (int)((k - 1) * dil);
#define RB_XR128(BN, WGN, TC)                                               \
  do {                                                                      \
    if (xrows <= 144) LAUNCH_RB(BN, WGN, TC, 144, 128);                     \
    else if (xrows <= 160) LAUNCH_RB(BN, WGN, TC, 160, 128);                \
    else LAUNCH_RB(BN, WGN, TC, 180, 128);                                  \
  } while (0)
#define RB_XR256(BN, WGN, TC)                                               \
  do {                                                                      \
    if (xrows <= 272) LAUNCH_RB(BN, WGN, TC, 272, 256);                     \
    else if (xrows <= 288) LAUNCH_RB(BN, WGN, TC, 288, 256);                \
    else LAUNCH_RB(BN, WGN, TC, 308, 256);                                  \
  } while (0)
#define RB_XR64(BN, WGN, TC)                                                \
  do {                                                                      \
    if (xrows <= 80) LAUNCH_RB(BN, WGN, TC, 80, 64);                        \
    else if (xrows <= 96) LAUNCH_RB(BN, WGN, TC, 96, 64);                   \
    else LAUNCH_RB(BN, WGN, TC, 124, 64);                                   \
  } while (0)
  if (CP == 256) {
    if (geom64) RB_XR64(256, 2, 1);   // 63 KB -> 2 blocks/CU (default)
    else RB_XR128(256, 2, 2);         // 120 KB -> 1 block/CU
  } else if (CP == 128) {
    if (geom64) RB_XR64(128, 2, 2);   // 47 KB -> 3 blocks (measured slower)
    else RB_XR128(128, 2, 3);         // default
  } else if (CP == 64) RB_XR128(64, 2, 2);
  else if (CP == 32) RB_XR256(32, 2, 2);
  else TORCH_CHECK(false, "resblock_cl: unsupported CP ", CP);
#undef RB_XR128
#undef RB_XR256
#undef RB_XR64
#undef LAUNCH_RB
  return out;
}
