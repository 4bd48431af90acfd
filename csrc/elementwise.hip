// Fused elementwise / normalization kernels for the VITS graph.
//
// Ops (SURVEY.md §2.2 kernel inventory):
//   - layer_norm_ct: LayerNorm across channels of [B, C, T]
//   - fused_gate:    WaveNet tanh(a+ga)·sigmoid(b+gb) gate
//   - prior_sample:  z = (m + eps·exp(logs)·noise_scale)·mask
//   - expand_states: duration length-regulator gather
//
// All are HBM-bandwidth-bound: one coalesced read per input element, one
// write per output, activations fused so no intermediate tensors hit HBM.
#include "common.h"

// --------------------------------------------------------------------------
// layer_norm_ct: x [B, C, T] -> per-(b,t) normalize across C.
// Thread t-major: lane i handles time position t0+i so every c-iteration
// reads a contiguous [blockDim.x] segment (fully coalesced along T).
// --------------------------------------------------------------------------
template <typename T, bool HAS_RES>
__global__ void layer_norm_ct_kernel(const T* __restrict__ x,
                                     const T* __restrict__ res,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     T* __restrict__ out, int C, long T_len,
                                     float eps, long n_bt) {
  const long bt = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (bt >= n_bt) return;
  const long b = bt / T_len;
  const long t = bt % T_len;
  const long base = (b * C) * T_len + t;
  const T* xp = x + base;
  const T* rp = HAS_RES ? res + base : nullptr;
  // single pass: sum + sum of squares (C is small, f32 accumulate)
  float s = 0.f, ss = 0.f;
  for (int c = 0; c < C; ++c) {
    float v = ld_f(xp + (long)c * T_len);
    if (HAS_RES) v += ld_f(rp + (long)c * T_len);
    s += v;
    ss += v * v;
  }
  const float mean = s / C;
  const float var = ss / C - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  T* op = out + base;
  for (int c = 0; c < C; ++c) {
    float v = ld_f(xp + (long)c * T_len);
    if (HAS_RES) v += ld_f(rp + (long)c * T_len);
    st_f(op + (long)c * T_len, (v - mean) * rstd * gamma[c] + beta[c]);
  }
}

// Lane-split variant for small B*T (encoder: ~8k tokens underfills the
// 256-CU chip with one thread per token).  SPLIT lanes cooperate on one
// token: lane = part*(64/SPLIT) + token, so each 64-lane wave holds
// 64/SPLIT tokens; stats reduced with SPLIT-1 shfl_xor rounds.  Each
// part owns a contiguous channel chunk (coalesced along T within a
// part-group).
template <typename T, bool HAS_RES, int SPLIT>
__global__ void layer_norm_ct_split_kernel(const T* __restrict__ x,
                                           const T* __restrict__ res,
                                           const float* __restrict__ gamma,
                                           const float* __restrict__ beta,
                                           T* __restrict__ out, int C,
                                           long T_len, float eps,
                                           long n_bt) {
  constexpr int TOK = 64 / SPLIT;  // tokens per wave
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int part = lane / TOK;
  const int tok = lane % TOK;
  const long bt = ((long)blockIdx.x * (blockDim.x >> 6) + wid) * TOK + tok;
  const bool live = bt < n_bt;
  const long b = live ? bt / T_len : 0;
  const long t = live ? bt % T_len : 0;
  const long base = (b * C) * T_len + t;
  const T* xp = x + base;
  const T* rp = HAS_RES ? res + base : nullptr;
  const int cq = (C + SPLIT - 1) / SPLIT;
  const int c_lo = part * cq;
  const int c_hi = min(c_lo + cq, C);
  float s = 0.f, ss = 0.f;
  if (live) {
    for (int c = c_lo; c < c_hi; ++c) {
      float v = ld_f(xp + (long)c * T_len);
      if (HAS_RES) v += ld_f(rp + (long)c * T_len);
      s += v;
      ss += v * v;
    }
  }
  // combine the SPLIT part-lanes of each token (they sit TOK apart)
#pragma unroll
  for (int d = TOK; d < 64; d <<= 1) {
    s += __shfl_xor(s, d, 64);
    ss += __shfl_xor(ss, d, 64);
  }
  if (!live) return;
  const float mean = s / C;
  const float var = ss / C - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  T* op = out + base;
  for (int c = c_lo; c < c_hi; ++c) {
    float v = ld_f(xp + (long)c * T_len);
    if (HAS_RES) v += ld_f(rp + (long)c * T_len);
    st_f(op + (long)c * T_len, (v - mean) * rstd * gamma[c] + beta[c]);
  }
}

// --------------------------------------------------------------------------
// fused_gate: x [B, 2C, T] (+ optional g) -> tanh·sigmoid gate [B, C, T]
// --------------------------------------------------------------------------
template <typename T, bool HAS_G>
__global__ void fused_gate_kernel(const T* __restrict__ x,
                                  const T* __restrict__ g,
                                  T* __restrict__ out, long C_T, long CT2,
                                  long T_len, long g_T,  // g time size: T or 1
                                  long n) {
  // n = B*C*T ; C_T = C*T (half-channel offset); CT2 = 2*C*T (batch stride)
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  long b = i / C_T;
  long r = i % C_T;
  long ia = b * CT2 + r;
  long ib = ia + C_T;
  float va = ld_f(x + ia), vb = ld_f(x + ib);
  if (HAS_G) {
    // g may be time-broadcast [B, 2C, 1] (speaker conditioning)
    long c = r / T_len;
    long t = g_T == 1 ? 0 : r % T_len;
    long C = C_T / T_len;
    long ga = (b * 2 * C + c) * g_T + t;
    va += ld_f(g + ga);
    vb += ld_f(g + (ga + C * g_T));
  }
  st_f(out + i, tanhf(va) * sigmoidf_(vb));
}

// --------------------------------------------------------------------------
// prior_sample
// --------------------------------------------------------------------------
template <typename T>
__global__ void prior_sample_kernel(const T* __restrict__ m,
                                    const T* __restrict__ logs,
                                    const T* __restrict__ mask,  // [B,1,T]
                                    const T* __restrict__ noise,
                                    T* __restrict__ out, long C_T, long T_len,
                                    float ns, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  long b = i / C_T;
  long t = i % T_len;
  float msk = ld_f(mask + b * T_len + t);
  float v = ld_f(m + i) + ld_f(noise + i) * __expf(ld_f(logs + i)) * ns;
  st_f(out + i, v * msk);
}

// --------------------------------------------------------------------------
// expand_states: stats [B, C, T] + durations [B, T] -> out [B, C, F]
// Per block: batch row b.  Thread 0 builds the cumulative-duration ->
// phoneme-index table in LDS (T is a few hundred), then all threads gather
// coalesced along F.
// --------------------------------------------------------------------------
template <typename T>
__global__ void expand_states_kernel(const T* __restrict__ stats,
                                     const int* __restrict__ durs,
                                     T* __restrict__ out, int C, int T_ph,
                                     int F_max) {
  extern __shared__ int idx_lds[];  // [F_max] phoneme index per frame
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  // serial cumsum + fill (T_ph and F_max are small; ~µs)
  if (tid == 0) {
    int f = 0;
    for (int p = 0; p < T_ph && f < F_max; ++p) {
      int d = durs[b * T_ph + p];
      for (int k = 0; k < d && f < F_max; ++k) idx_lds[f++] = p;
    }
    // pad remaining frames with last phoneme (masked out later anyway)
    int last = T_ph - 1;
    while (f < F_max) idx_lds[f++] = last;
  }
  __syncthreads();
  const T* sp = stats + (long)b * C * T_ph;
  T* op = out + (long)b * C * F_max;
  for (long cf = tid; cf < (long)C * F_max; cf += blockDim.x) {
    int c = cf / F_max;
    int f = cf % F_max;
    op[cf] = sp[(long)c * T_ph + idx_lds[f]];
  }
}

// ========================================================================
// host wrappers
// ========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

#define DISPATCH_FT(TENSOR, NAME, ...)                                     \
  do {                                                                     \
    if ((TENSOR).scalar_type() == at::kFloat) {                            \
      using scalar_t = float;                                              \
      __VA_ARGS__;                                                         \
    } else if ((TENSOR).scalar_type() == at::kBFloat16) {                  \
      using scalar_t = bf16;                                               \
      __VA_ARGS__;                                                         \
    } else {                                                               \
      TORCH_CHECK(false, NAME ": unsupported dtype");                      \
    }                                                                      \
  } while (0)

torch::Tensor layer_norm_ct(torch::Tensor x, c10::optional<torch::Tensor> res,
                            torch::Tensor gamma, torch::Tensor beta,
                            double eps) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), C = x.size(1), T = x.size(2);
  auto out = torch::empty_like(x);
  auto gamma_f = gamma.to(at::kFloat).contiguous();
  auto beta_f = beta.to(at::kFloat).contiguous();
  // Fill the chip: one thread per token underfills at encoder sizes
  // (B*T ~ 8k vs 256 CUs), so split each token's channel reduction over
  // SPLIT lanes when the token count is small.
  const long n_bt = B * T;
  const int threads = 256;
#define LN_LAUNCH(KER, GRID)                                                \
  DISPATCH_FT(x, "layer_norm_ct", {                                        \
    if (res.has_value()) {                                                 \
      TORCH_CHECK(res->sizes() == x.sizes() && res->is_contiguous());      \
      hipLaunchKernelGGL((KER<scalar_t, true>), GRID, dim3(threads), 0,    \
                         cur_stream(), (const scalar_t*)x.data_ptr(),      \
                         (const scalar_t*)res->data_ptr(),                 \
                         gamma_f.data_ptr<float>(),                        \
                         beta_f.data_ptr<float>(),                         \
                         (scalar_t*)out.data_ptr(), (int)C, T, (float)eps, \
                         n_bt);                                            \
    } else {                                                               \
      hipLaunchKernelGGL((KER<scalar_t, false>), GRID, dim3(threads), 0,   \
                         cur_stream(), (const scalar_t*)x.data_ptr(),      \
                         (const scalar_t*)nullptr,                         \
                         gamma_f.data_ptr<float>(),                        \
                         beta_f.data_ptr<float>(),                         \
                         (scalar_t*)out.data_ptr(), (int)C, T, (float)eps, \
                         n_bt);                                            \
    }                                                                      \
  })
  if (n_bt <= 16384) {
    dim3 grid(ceil_div(n_bt, (long)threads / 8));
    DISPATCH_FT(x, "layer_norm_ct", {
      if (res.has_value()) {
        TORCH_CHECK(res->sizes() == x.sizes() && res->is_contiguous());
        hipLaunchKernelGGL((layer_norm_ct_split_kernel<scalar_t, true, 8>),
                           grid, dim3(threads), 0, cur_stream(),
                           (const scalar_t*)x.data_ptr(),
                           (const scalar_t*)res->data_ptr(),
                           gamma_f.data_ptr<float>(),
                           beta_f.data_ptr<float>(),
                           (scalar_t*)out.data_ptr(), (int)C, T, (float)eps,
                           n_bt);
      } else {
        hipLaunchKernelGGL((layer_norm_ct_split_kernel<scalar_t, false, 8>),
                           grid, dim3(threads), 0, cur_stream(),
                           (const scalar_t*)x.data_ptr(),
                           (const scalar_t*)nullptr,
                           gamma_f.data_ptr<float>(),
                           beta_f.data_ptr<float>(),
                           (scalar_t*)out.data_ptr(), (int)C, T, (float)eps,
                           n_bt);
      }
    });
  } else {
    dim3 grid(ceil_div(n_bt, (long)threads));
    LN_LAUNCH(layer_norm_ct_kernel, grid);
  }
#undef LN_LAUNCH
  return out;
}

torch::Tensor fused_gate(torch::Tensor x, c10::optional<torch::Tensor> g,
                         long n_channels) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), C2 = x.size(1), T = x.size(2);
  TORCH_CHECK(C2 == 2 * n_channels, "fused_gate: channel mismatch");
  auto out = torch::empty({B, n_channels, T}, x.options());
  const long n = B * n_channels * T;
  const int threads = 256;
  const long blocks = (n + threads - 1) / threads;
  DISPATCH_FT(x, "fused_gate", {
    if (g.has_value()) {
      TORCH_CHECK(g->dim() == 3 && g->size(0) == B && g->size(1) == C2 &&
                      (g->size(2) == T || g->size(2) == 1),
                  "fused_gate: g must be [B,2C,T] or [B,2C,1]");
      hipLaunchKernelGGL((fused_gate_kernel<scalar_t, true>), dim3(blocks),
                         dim3(threads), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)g->data_ptr(),
                         (scalar_t*)out.data_ptr(), n_channels * T,
                         2 * n_channels * T, T, g->size(2), n);
    } else {
      hipLaunchKernelGGL((fused_gate_kernel<scalar_t, false>), dim3(blocks),
                         dim3(threads), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(), (const scalar_t*)nullptr,
                         (scalar_t*)out.data_ptr(), n_channels * T,
                         2 * n_channels * T, T, 1, n);
    }
  });
  return out;
}

torch::Tensor prior_sample(torch::Tensor m, torch::Tensor logs,
                           torch::Tensor mask, torch::Tensor noise,
                           double noise_scale) {
  TORCH_CHECK(m.is_cuda() && m.is_contiguous() && m.dim() == 3);
  const long B = m.size(0), C = m.size(1), T = m.size(2);
  auto out = torch::empty_like(m);
  const long n = B * C * T;
  const int threads = 256;
  DISPATCH_FT(m, "prior_sample", {
    hipLaunchKernelGGL(prior_sample_kernel<scalar_t>,
                       dim3((n + threads - 1) / threads), dim3(threads), 0,
                       cur_stream(), (const scalar_t*)m.data_ptr(),
                       (const scalar_t*)logs.data_ptr(),
                       (const scalar_t*)mask.data_ptr(),
                       (const scalar_t*)noise.data_ptr(),
                       (scalar_t*)out.data_ptr(), C * T, T,
                       (float)noise_scale, n);
  });
  return out;
}

torch::Tensor expand_states(torch::Tensor stats, torch::Tensor durs,
                            long F_max) {
  TORCH_CHECK(stats.dim() == 3 && stats.is_cuda() && stats.is_contiguous());
  TORCH_CHECK(durs.scalar_type() == at::kInt && durs.is_contiguous());
  const long B = stats.size(0), C = stats.size(1), T = stats.size(2);
  auto out = torch::empty({B, C, F_max}, stats.options());
  size_t lds = F_max * sizeof(int);
  TORCH_CHECK(lds <= 160 * 1024, "expand_states: F too large for LDS");
  DISPATCH_FT(stats, "expand_states", {
    hipLaunchKernelGGL(expand_states_kernel<scalar_t>, dim3(B), dim3(256),
                       lds, cur_stream(), (const scalar_t*)stats.data_ptr(),
                       durs.data_ptr<int>(), (scalar_t*)out.data_ptr(),
                       (int)C, (int)T, (int)F_max);
  });
  return out;
}

// --------------------------------------------------------------------------
// mask_tail_: in-place zero of x[b, :, lens[b]:] for padded batches.
// Makes ragged-batch decoding bit-equal to single-utterance decoding when
// applied after each conv stage (padding region never feeds valid taps).
// One block per (b, c) row; threads sweep the tail only.
// --------------------------------------------------------------------------
template <typename T>
__global__ void mask_tail_kernel(T* __restrict__ x,
                                 const int* __restrict__ lens, int C,
                                 long T_len) {
  const int b = blockIdx.y;
  const int c = blockIdx.x;
  const long lo = lens[b];
  for (long t = lo + threadIdx.x; t < T_len; t += blockDim.x)
    x[((long)b * C + c) * T_len + t] = T(0);
}

torch::Tensor mask_tail_(torch::Tensor x, torch::Tensor lens) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(lens.scalar_type() == at::kInt && lens.is_cuda() &&
              lens.is_contiguous() && lens.size(0) == x.size(0));
  const long B = x.size(0), C = x.size(1), T_len = x.size(2);
  DISPATCH_FT(x, "mask_tail_", {
    hipLaunchKernelGGL(mask_tail_kernel<scalar_t>, dim3(C, B), dim3(256), 0,
                       cur_stream(), (scalar_t*)x.data_ptr(),
                       lens.data_ptr<int>(), (int)C, T_len);
  });
  return x;
}

// --------------------------------------------------------------------------
// fused_gate_cl: channel-last WaveNet gate.
// x [B, F, 2C] (+ optional g [B, 2C] speaker bias or [B, F, 2C]) ->
// out [B, F, C] = tanh(xa + ga) * sigmoid(xb + gb).
// Channel-last rows are contiguous: coalesced along C.
// --------------------------------------------------------------------------
template <typename T, int GMODE>  // 0: none, 1: [B,2C] broadcast, 2: full
__global__ void fused_gate_cl_kernel(const T* __restrict__ x,
                                     const T* __restrict__ g,
                                     T* __restrict__ out, long C, long F_len,
                                     long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;  // n = B*F*C
  const long c = i % C;
  const long bf = i / C;
  const long b = bf / F_len;
  const long ia = bf * 2 * C + c;
  float va = ld_f(x + ia), vb = ld_f(x + ia + C);
  if (GMODE == 1) {
    va += ld_f(g + b * 2 * C + c);
    vb += ld_f(g + b * 2 * C + C + c);
  } else if (GMODE == 2) {
    va += ld_f(g + ia);
    vb += ld_f(g + ia + C);
  }
  st_f(out + i, tanhf(va) * sigmoidf_(vb));
}

torch::Tensor fused_gate_cl(torch::Tensor x, c10::optional<torch::Tensor> g,
                            long n_channels) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), F_len = x.size(1), C2 = x.size(2);
  TORCH_CHECK(C2 == 2 * n_channels, "fused_gate_cl: channel mismatch");
  auto out = torch::empty({B, F_len, n_channels}, x.options());
  const long n = B * F_len * n_channels;
  const int threads = 256;
  const long blocks = (n + threads - 1) / threads;
  DISPATCH_FT(x, "fused_gate_cl", {
    if (!g.has_value()) {
      hipLaunchKernelGGL((fused_gate_cl_kernel<scalar_t, 0>), dim3(blocks),
                         dim3(threads), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)nullptr,
                         (scalar_t*)out.data_ptr(), n_channels, F_len, n);
    } else if (g->dim() == 2 ||
               (g->dim() == 3 && g->size(1) == 1)) {
      TORCH_CHECK(g->numel() == B * C2, "fused_gate_cl: bad g");
      hipLaunchKernelGGL((fused_gate_cl_kernel<scalar_t, 1>), dim3(blocks),
                         dim3(threads), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)g->contiguous().data_ptr(),
                         (scalar_t*)out.data_ptr(), n_channels, F_len, n);
    } else {
      TORCH_CHECK(g->sizes() == x.sizes(), "fused_gate_cl: bad g");
      hipLaunchKernelGGL((fused_gate_cl_kernel<scalar_t, 2>), dim3(blocks),
                         dim3(threads), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)g->contiguous().data_ptr(),
                         (scalar_t*)out.data_ptr(), n_channels, F_len, n);
    }
  });
  return out;
}

// --------------------------------------------------------------------------
// depthwise_cl: channel-last depthwise Conv1d (DDSConv separable stage).
// out[b,t,c] = bias[c] + sum_j x[b, t + j*dil - pad, c] * w[c, j]
// Channel-last rows are contiguous: thread handles 8 channels of one row
// (b128 loads/stores); taps walk rows.  Memory-bound by design.
// --------------------------------------------------------------------------
template <typename T>
__global__ void depthwise_cl_kernel(const T* __restrict__ x,
                                    const T* __restrict__ w,  // [C][k]
                                    const float* __restrict__ bias,
                                    T* __restrict__ out, long Tlen, int C,
                                    int k, int dil, int pad, long n8) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n8) return;  // n8 = B*T*C/8
  const long c8 = (i % (C / 8)) * 8;
  const long bt = i / (C / 8);
  const long b = bt / Tlen;
  const long t = bt % Tlen;
  const T* xb = x + (b * Tlen) * C;
  float acc[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) acc[q] = bias ? bias[c8 + q] : 0.f;
  for (int j = 0; j < k; ++j) {
    const long tt = t + (long)j * dil - pad;
    if (tt < 0 || tt >= Tlen) continue;
    const T* xr = xb + tt * C + c8;
#pragma unroll
    for (int q = 0; q < 8; ++q)
      acc[q] += ld_f(xr + q) * ld_f(w + (c8 + q) * k + j);
  }
  T* or_ = out + (b * Tlen + t) * C + c8;
#pragma unroll
  for (int q = 0; q < 8; ++q) st_f(or_ + q, acc[q]);
}

torch::Tensor depthwise_cl(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, long dil,
                           long pad) {
  // x: [B, T, C] channel-last; w: [C, 1, k] (nn.Conv1d groups=C weight)
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), Tlen = x.size(1), C = x.size(2);
  TORCH_CHECK(C % 8 == 0, "depthwise_cl: C % 8 != 0");
  auto w2 = w.reshape({C, w.size(-1)}).contiguous();
  const long k = w2.size(1);
  auto out = torch::empty_like(x);
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->scalar_type() == at::kFloat
                 ? *bias : bias->to(at::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  const long n8 = B * Tlen * (C / 8);
  const int threads = 256;
  DISPATCH_FT(x, "depthwise_cl", {
    hipLaunchKernelGGL(depthwise_cl_kernel<scalar_t>,
                       dim3((n8 + threads - 1) / threads), dim3(threads), 0,
                       cur_stream(), (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w2.data_ptr(), bias_p,
                       (scalar_t*)out.data_ptr(), Tlen, (int)C, (int)k,
                       (int)dil, (int)pad, n8);
  });
  return out;
}

// ------------------------------------------------------------------------- //
// seeded_noise: per-utterance deterministic standard-normal noise in ONE
// launch.  Replaces the per-row torch::randn loops (B x 2 generator
// launches per batch + per-row .item() syncs) that showed up as 1.4k
// distribution kernels per bench step.  Value at (b, c, t) depends ONLY
// on (seeds[b], c, t): noise is independent of batch composition, rank
// AND padding by construction (SURVEY.md section 7 hard part 7).
// RNG: splitmix64 counter hash -> Box-Muller.
// ------------------------------------------------------------------------- //
__device__ __forceinline__ unsigned long long sm64_(unsigned long long z) {
  z += 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

template <typename T>
__global__ void seeded_noise_kernel(T* __restrict__ out,
                                    const long* __restrict__ seeds,
                                    const int* __restrict__ lens,
                                    long B, long C, long Tm) {
  const long total = B * C * Tm;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const long b = i / (C * Tm);
    const long rem = i - b * C * Tm;
    const long c = rem / Tm, t = rem - (rem / Tm) * Tm;
    float v = 0.f;
    if (t < (long)lens[b]) {
      const unsigned long long ctr =
          ((unsigned long long)(c + 1) << 40) ^ (unsigned long long)t;
      const unsigned long long r1 =
          sm64_((unsigned long long)seeds[b] ^ sm64_(ctr));
      const unsigned long long r2 = sm64_(r1);
      const float u1 = (float)((r1 >> 11) + 1) * 1.1102230246251565e-16f;
      const float u2 = (float)(r2 >> 11) * 1.1102230246251565e-16f;
      v = sqrtf(-2.0f * __logf(u1)) * __cosf(6.28318530717958f * u2);
    }
    st_f(out + i, v);
  }
}

torch::Tensor seeded_noise(long B, long C, long T_max, torch::Tensor lens,
                           torch::Tensor seeds, torch::ScalarType dtype) {
  TORCH_CHECK(lens.is_cuda() && lens.scalar_type() == at::kInt);
  TORCH_CHECK(seeds.is_cuda() && seeds.scalar_type() == at::kLong);
  TORCH_CHECK(lens.numel() == B && seeds.numel() == B);
  auto out = torch::empty(
      {B, C, T_max},
      torch::TensorOptions().device(lens.device()).dtype(dtype));
  if (out.numel() == 0) return out;
  const long total = B * C * T_max;
  const int blocks = (int)std::min<long>((total + 255) / 256, 4096);
  if (dtype == at::kBFloat16) {
    hipLaunchKernelGGL(seeded_noise_kernel<bf16>, dim3(blocks), dim3(256), 0,
                       cur_stream(), (bf16*)out.data_ptr(),
                       seeds.data_ptr<long>(), lens.data_ptr<int>(), B, C,
                       T_max);
  } else {
    TORCH_CHECK(dtype == at::kFloat, "seeded_noise: f32/bf16 only");
    hipLaunchKernelGGL(seeded_noise_kernel<float>, dim3(blocks), dim3(256),
                       0, cur_stream(), (float*)out.data_ptr(),
                       seeds.data_ptr<long>(), lens.data_ptr<int>(), B, C,
                       T_max);
  }
  return out;
}

// ------------------------------------------------------------------------- //
// row_ln_cl: LayerNorm over the last (channel) dim of channel-last rows
// with FUSED residual add: y = LN(x + r) * gamma + beta.  The encoder
// calls torch::layer_norm(x + y) ~400x per step (two launches + an
// intermediate tensor each); this is one launch, one pass.
// One wave per row (C <= 8*64 via per-lane accumulation + butterfly).
// ------------------------------------------------------------------------- //
template <typename T>
__global__ void row_ln_cl_kernel(const T* __restrict__ x,
                                 const T* __restrict__ r,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 T* __restrict__ out, long rows, int C,
                                 float eps) {
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const T* xr = x + row * C;
  const T* rr = r ? r + row * C : nullptr;
  float v[8];
  float sum = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int c = lane + i * 64;
    float t = 0.f;
    if (c < C) {
      t = ld_f(xr + c);
      if (rr) t += ld_f(rr + c);
    }
    v[i] = t;
    sum += t;
  }
#pragma unroll
  for (int sh = 32; sh > 0; sh >>= 1) sum += __shfl_xor(sum, sh, 64);
  const float mean = sum / C;
  float var = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int c = lane + i * 64;
    if (c < C) {
      v[i] -= mean;
      var += v[i] * v[i];
    }
  }
#pragma unroll
  for (int sh = 32; sh > 0; sh >>= 1) var += __shfl_xor(var, sh, 64);
  const float inv = rsqrtf(var / C + eps);
  T* orow = out + row * C;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int c = lane + i * 64;
    if (c < C) st_f(orow + c, v[i] * inv * gamma[c] + beta[c]);
  }
}

torch::Tensor row_ln_cl(torch::Tensor x, c10::optional<torch::Tensor> resid,
                        torch::Tensor gamma, torch::Tensor beta,
                        double eps) {
  TORCH_CHECK(x.dim() == 3 && x.is_cuda() && x.is_contiguous());
  const long B = x.size(0), T = x.size(1), C = x.size(2);
  TORCH_CHECK(C <= 512, "row_ln_cl: C too large");
  auto g32 = gamma.scalar_type() == at::kFloat ? gamma.contiguous()
                                               : gamma.to(at::kFloat).contiguous();
  auto b32 = beta.scalar_type() == at::kFloat ? beta.contiguous()
                                              : beta.to(at::kFloat).contiguous();
  auto out = torch::empty_like(x);
  const long rows = B * T;
  if (!rows) return out;
  const void* rp = nullptr;
  if (resid.has_value()) {
    TORCH_CHECK(resid->is_contiguous() && resid->sizes() == x.sizes());
    rp = resid->data_ptr();
  }
  const int waves_per_block = 4;  // 256 threads
  const int blocks = (int)((rows + waves_per_block - 1) / waves_per_block);
  DISPATCH_FT_CONV(x, hipLaunchKernelGGL(
      row_ln_cl_kernel<scalar_t>, dim3(blocks), dim3(64 * waves_per_block),
      0, cur_stream(), (const scalar_t*)x.data_ptr(),
      (const scalar_t*)rp, g32.data_ptr<float>(), b32.data_ptr<float>(),
      (scalar_t*)out.data_ptr(), rows, (int)C, (float)eps));
  return out;
}
