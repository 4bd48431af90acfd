// Fused relative-position multi-head attention, channel-last (gfx950).
//
// Replaces the TextEncoder attention's hipBLASLt matmul chain
// (QK^T -> +rel_to_abs(Q rel_k^T) -> masked softmax -> PV ->
//  +abs_to_rel(P) rel_v) with ONE kernel per layer: the VITS relative
// terms touch only a +/-window diagonal band (the rel tables are
// zero-padded outside +/-window, vits.py:111-116), so instead of the
// oracle's [B,h,T,2T-1] pad/reshape traffic this kernel
//   - precomputes band logits R[m][delta] = Q[m] . rel_k[delta] once,
//   - adds them along the band while assembling the score tile,
//   - records the band scores during the K sweep and applies the
//     closed-form band output  sum_d exp(s_band - m_row)/l_row * rel_v
//     in the epilogue (no second softmax pass, no rescale chain).
// Online (flash-style) softmax over column tiles; everything stays in
// LDS/registers; one store of [B,T,H,D] at the end.
//
// Layouts (all bf16, channel-last rows):
//   qkv  [B][T][3][H*D]   packed rows from ONE fused linear
//   rel_k/rel_v [2w+1][D] (heads share the table, vits.py:84-90)
//   out  [B][T][H*D]
// Oracle: sonata_amd/models/vits.py RelativeAttention.forward_cl
// (reference semantics: upstream VITS attentions.py via the ONNX graph,
//  SURVEY.md section 2.2 TextEncoder row).
//
// Tiling: one workgroup = (64 query rows) x (one head) x (one batch);
// 4 waves, each owning 16 rows; column tiles of 64 keys swept with
// mfma_f32_16x16x32_bf16 for both QK^T and PV (P round-trips through
// LDS to convert C-layout -> A-fragment layout).
#include "common.h"

#define ATT_BM 64
#define ATT_BN 64
#define ATT_BNP 72       // +8 bf16: conflict-free b128 rows
#define ATT_SLOTS 16     // band slots (window<=7), padded
#define NEG_BAND -3.0e38f

template <int DPAD>
__global__ __launch_bounds__(256) void attn_relpos_cl_kernel(
    const bf16* __restrict__ qkv,   // [B][T][3][HD]
    const bf16* __restrict__ relk,  // [2w+1][D]
    const bf16* __restrict__ relv,  // [2w+1][D]
    const int* __restrict__ lens,   // [B] or null (=> all T)
    bf16* __restrict__ out,         // [B][T][HD]
    int B, long T, int H, int D, int w, float scale) {
  constexpr int DP = DPAD + 8;
  constexpr int DT = DPAD / 16;
  const long t0 = (long)blockIdx.x * ATT_BM;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int HD = H * D;
  const long rowp = 3L * HD;  // qkv row pitch

  __shared__ bf16 Qs[ATT_BM][DP];
  __shared__ bf16 Ks[ATT_BN][DP];
  __shared__ bf16 Vt[DPAD][ATT_BNP];
  __shared__ bf16 Ps[ATT_BM][ATT_BNP];
  __shared__ bf16 Rk[ATT_SLOTS][DP];
  __shared__ bf16 Rv[ATT_SLOTS][DP];
  __shared__ float Bqk[ATT_BM][ATT_SLOTS];    // q . rel_k band logits
  __shared__ float Sband[ATT_BM][ATT_SLOTS];  // recorded band scores

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;      // wave id = row strip
  const int kl = lane >> 4;      // 0..3
  const int il = lane & 15;      // 0..15

  const long len_b = lens ? min((long)lens[b], T) : T;
  const int nslot = 2 * w + 1;
  const bf16 zero = f2bf(0.f);

  // ---- stage Q (scaled) + rel tables --------------------------------- //
  const bf16* qb = qkv + ((long)b * T) * rowp + (long)h * D;  // q row base
  for (int u = tid; u < ATT_BM * (DP / 8); u += 256) {
    const int r = u / (DP / 8), ch = (u % (DP / 8)) * 8;
    bf16 v8[8];
    const long t = t0 + r;
    if (t < T && ch < D) {
      if (ch + 8 <= D) {
        *(ulonglong2*)v8 = *(const ulonglong2*)&qb[t * rowp + ch];
      } else {
        for (int q = 0; q < 8; ++q)
          v8[q] = (ch + q < D) ? qb[t * rowp + ch + q] : zero;
      }
      for (int q = 0; q < 8; ++q) v8[q] = f2bf(bf2f(v8[q]) * scale);
    } else {
      for (int q = 0; q < 8; ++q) v8[q] = zero;
    }
    *(ulonglong2*)&Qs[r][ch] = *(ulonglong2*)v8;
  }
  for (int u = tid; u < ATT_SLOTS * (DP / 8); u += 256) {
    const int s = u / (DP / 8), ch = (u % (DP / 8)) * 8;
    bf16 k8[8], v8[8];
    for (int q = 0; q < 8; ++q) {
      const bool live = (s < nslot) && (ch + q < D);
      k8[q] = live ? relk[(long)s * D + ch + q] : zero;
      v8[q] = live ? relv[(long)s * D + ch + q] : zero;
    }
    *(ulonglong2*)&Rk[s][ch] = *(ulonglong2*)k8;
    *(ulonglong2*)&Rv[s][ch] = *(ulonglong2*)v8;
  }
  // init band score record
  for (int u = tid; u < ATT_BM * ATT_SLOTS; u += 256) {
    Sband[u / ATT_SLOTS][u % ATT_SLOTS] = NEG_BAND;
  }
  __syncthreads();

  // ---- band logits: Bqk[m][s] = Qs[m] . Rk[s] (Q already scaled) ----- //
  for (int u = tid; u < ATT_BM * ATT_SLOTS; u += 256) {
    const int r = u / ATT_SLOTS, s = u % ATT_SLOTS;
    float acc = 0.f;
    if (s < nslot) {
#pragma unroll 8
      for (int d = 0; d < DPAD; ++d) acc += bf2f(Qs[r][d]) * bf2f(Rk[s][d]);
    }
    Bqk[r][s] = acc;
  }

  // ---- online softmax state (rows owned by this lane: kl*4+rg) ------- //
  float m_run[4], l_run[4];
#pragma unroll
  for (int rg = 0; rg < 4; ++rg) {
    m_run[rg] = -3.0e38f;
    l_run[rg] = 0.f;
  }
  f32x4 accO[DT];
#pragma unroll
  for (int dj = 0; dj < DT; ++dj) accO[dj] = {0.f, 0.f, 0.f, 0.f};

  const bf16* kb = qb + HD;       // k rows
  const bf16* vb = qb + 2L * HD;  // v rows
  const int r_loc0 = wid * 16;    // this wave's row strip

  for (long j0 = 0; j0 < T; j0 += ATT_BN) {
    __syncthreads();  // previous tile's PV reads done before restage
    // ---- stage K tile rows + V tile transposed ---------------------- //
    for (int u = tid; u < ATT_BN * (DP / 8); u += 256) {
      const int r = u / (DP / 8), ch = (u % (DP / 8)) * 8;
      const long j = j0 + r;
      bf16 v8[8];
      if (j < T && ch < D) {
        if (ch + 8 <= D) {
          *(ulonglong2*)v8 = *(const ulonglong2*)&kb[j * rowp + ch];
        } else {
          for (int q = 0; q < 8; ++q)
            v8[q] = (ch + q < D) ? kb[j * rowp + ch + q] : zero;
        }
      } else {
        for (int q = 0; q < 8; ++q) v8[q] = zero;
      }
      *(ulonglong2*)&Ks[r][ch] = *(ulonglong2*)v8;
      // transpose into Vt (scalar writes; V rows are contiguous reads)
      bf16 t8[8];
      if (j < T && ch < D) {
        if (ch + 8 <= D) {
          *(ulonglong2*)t8 = *(const ulonglong2*)&vb[j * rowp + ch];
        } else {
          for (int q = 0; q < 8; ++q)
            t8[q] = (ch + q < D) ? vb[j * rowp + ch + q] : zero;
        }
      } else {
        for (int q = 0; q < 8; ++q) t8[q] = zero;
      }
#pragma unroll
      for (int q = 0; q < 8; ++q)
        if (ch + q < DPAD) Vt[ch + q][r] = t8[q];
    }
    __syncthreads();

    // ---- S strip = Qs(rows of this wave) x Ks^T --------------------- //
    f32x4 accS[ATT_BN / 16];
#pragma unroll
    for (int nj = 0; nj < ATT_BN / 16; ++nj) accS[nj] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int k0 = 0; k0 < DPAD; k0 += 32) {
      const bf16x8 a_frag = *(const bf16x8*)&Qs[r_loc0 + il][k0 + kl * 8];
#pragma unroll
      for (int nj = 0; nj < ATT_BN / 16; ++nj) {
        const bf16x8 b_frag = *(const bf16x8*)&Ks[nj * 16 + il][k0 + kl * 8];
        accS[nj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, accS[nj], 0, 0, 0);
      }
    }

    // ---- band add + mask; tile row max ------------------------------ //
    float s_val[ATT_BN / 16][4];
    float tmax[4] = {-3.0e38f, -3.0e38f, -3.0e38f, -3.0e38f};
#pragma unroll
    for (int nj = 0; nj < ATT_BN / 16; ++nj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const int r = r_loc0 + kl * 4 + rg;      // local row
        const long gr = t0 + r;                  // global row
        const long gj = j0 + nj * 16 + il;       // global col
        float s = accS[nj][rg];
        const long delta = gj - gr;
        if (delta >= -w && delta <= w) s += Bqk[r][delta + w];
        if (gj >= len_b) s = (gj < T) ? -1e4f : NEG_BAND;
        s_val[nj][rg] = s;
        tmax[rg] = fmaxf(tmax[rg], s);
      }
    }
    // butterfly max across the 16 il lanes (same kl group)
#pragma unroll
    for (int sh = 1; sh < 16; sh <<= 1) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg)
        tmax[rg] = fmaxf(tmax[rg],
                         __shfl_xor(tmax[rg], sh, 64));
    }

    // ---- online softmax update + write P tile ----------------------- //
    float tsum[4] = {0.f, 0.f, 0.f, 0.f};
    float fscale[4];
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      const float m_new = fmaxf(m_run[rg], tmax[rg]);
      fscale[rg] = __expf(m_run[rg] - m_new);
      m_run[rg] = m_new;
    }
#pragma unroll
    for (int nj = 0; nj < ATT_BN / 16; ++nj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        const float p = __expf(s_val[nj][rg] - m_run[rg]);
        tsum[rg] += p;
        Ps[r_loc0 + kl * 4 + rg][nj * 16 + il] = f2bf(p);
        // record band scores for the epilogue rel_v term
        const int r = r_loc0 + kl * 4 + rg;
        const long delta = (j0 + nj * 16 + il) - (t0 + r);
        if (delta >= -w && delta <= w) Sband[r][delta + w] = s_val[nj][rg];
      }
    }
#pragma unroll
    for (int sh = 1; sh < 16; sh <<= 1) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg)
        tsum[rg] += __shfl_xor(tsum[rg], sh, 64);
    }
#pragma unroll
    for (int rg = 0; rg < 4; ++rg) {
      l_run[rg] = l_run[rg] * fscale[rg] + tsum[rg];
    }
    // rescale O accumulators (rows kl*4+rg match accO's C-layout rows)
#pragma unroll
    for (int dj = 0; dj < DT; ++dj)
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) accO[dj][rg] *= fscale[rg];

    __syncthreads();  // Ps visible to... (same wave only; conservative)

    // ---- O strip += P x V ------------------------------------------- //
#pragma unroll
    for (int k0 = 0; k0 < ATT_BN; k0 += 32) {
      const bf16x8 a_frag = *(const bf16x8*)&Ps[r_loc0 + il][k0 + kl * 8];
#pragma unroll
      for (int dj = 0; dj < DT; ++dj) {
        const bf16x8 b_frag = *(const bf16x8*)&Vt[dj * 16 + il][k0 + kl * 8];
        accO[dj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, accO[dj], 0, 0, 0);
      }
    }
  }
  __syncthreads();  // Sband complete before epilogue reads

  // ---- epilogue: band rel_v term + 1/l + store ----------------------- //
  bf16* ob = out + ((long)b * T) * HD + (long)h * D;
#pragma unroll
  for (int rg = 0; rg < 4; ++rg) {
    const int r = r_loc0 + kl * 4 + rg;
    const long t = t0 + r;
    if (t >= T) continue;
    const float inv_l = (l_run[rg] > 0.f) ? 1.0f / l_run[rg] : 0.f;
    // band probabilities for this row (<=15 slots, shared across d)
    float pband[ATT_SLOTS];
#pragma unroll
    for (int s = 0; s < ATT_SLOTS; ++s)
      pband[s] = (s < nslot && Sband[r][s] > -1.0e38f)
                     ? __expf(Sband[r][s] - m_run[rg])
                     : 0.f;
#pragma unroll
    for (int dj = 0; dj < DT; ++dj) {
      const int d = dj * 16 + il;
      if (d >= D) continue;
      float v = accO[dj][rg];
      for (int s = 0; s < ATT_SLOTS; ++s)
        if (pband[s] != 0.f) v += pband[s] * bf2f(Rv[s][d]);
      ob[t * HD + d] = f2bf(v * inv_l);
    }
  }
}

// --------------------------------------------------------------------- //
// host launcher
// --------------------------------------------------------------------- //
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static inline hipStream_t attn_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

torch::Tensor attn_relpos_cl(torch::Tensor qkv, torch::Tensor rel_k,
                             torch::Tensor rel_v,
                             c10::optional<torch::Tensor> lens, long H,
                             long window, double scale) {
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_cuda() && qkv.is_contiguous(),
              "attn: qkv must be [B,T,3*H*D] contiguous cuda");
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16, "attn: bf16 only");
  const long B = qkv.size(0), T = qkv.size(1);
  TORCH_CHECK(qkv.size(2) % (3 * H) == 0);
  const long D = qkv.size(2) / (3 * H);
  TORCH_CHECK(D % 8 == 0 && D <= 128, "attn: head_dim must be <=128, 8|D");
  TORCH_CHECK(rel_k.is_contiguous() && rel_v.is_contiguous());
  TORCH_CHECK(rel_k.size(-1) == D && rel_v.size(-1) == D);
  TORCH_CHECK(2 * window + 1 <= ATT_SLOTS, "attn: window too large");
  TORCH_CHECK(rel_k.scalar_type() == at::kBFloat16 &&
              rel_v.scalar_type() == at::kBFloat16);
  auto out = torch::empty({B, T, H * D}, qkv.options());
  if (out.numel() == 0) return out;
  const int* lens_p = nullptr;
  if (lens.has_value()) {
    TORCH_CHECK(lens->scalar_type() == at::kInt && lens->is_cuda());
    lens_p = lens->data_ptr<int>();
  }
  hipStream_t st = attn_stream();
  dim3 grid(ceil_div(T, ATT_BM), H, B);
#define LAUNCH_ATT(DPAD)                                                     \
  hipLaunchKernelGGL((attn_relpos_cl_kernel<DPAD>), grid, dim3(256), 0, st,  \
                     (const bf16*)qkv.data_ptr(),                            \
                     (const bf16*)rel_k.data_ptr(),                          \
                     (const bf16*)rel_v.data_ptr(), lens_p,                  \
                     (bf16*)out.data_ptr(), (int)B, T, (int)H, (int)D,       \
                     (int)window, (float)scale)
  if (D <= 32) LAUNCH_ATT(32);
  else if (D <= 64) LAUNCH_ATT(64);
  else if (D <= 96) LAUNCH_ATT(96);
  else LAUNCH_ATT(128);
#undef LAUNCH_ATT
  return out;
}
