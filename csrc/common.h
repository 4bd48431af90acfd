// Common helpers for sonata_amd CDNA4 (gfx950) kernels.
//
// Target: MI355X only — wave64, MFMA bf16 16x16x32, 160 KiB LDS/CU,
// 8 XCDs.  No CUDA-compat shims, no multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;

// MFMA fragment vector types (gfx950 mfma_f32_16x16x32_bf16):
// A/B: 8 bf16 per lane (4 VGPRs), C/D: 4 f32 per lane.
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

template <typename T> __device__ __forceinline__ float ld_f(const T* p);
template <> __device__ __forceinline__ float ld_f<float>(const float* p) { return *p; }
template <> __device__ __forceinline__ float ld_f<bf16>(const bf16* p) { return bf2f(*p); }

template <typename T> __device__ __forceinline__ void st_f(T* p, float v);
template <> __device__ __forceinline__ void st_f<float>(float* p, float v) { *p = v; }
template <> __device__ __forceinline__ void st_f<bf16>(bf16* p, float v) { *p = f2bf(v); }

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

__device__ __forceinline__ float lrelu_(float x, float slope) {
  return x > 0.0f ? x : x * slope;
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",      \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// dtype dispatch over float / bf16 torch tensors
#define DISPATCH_FT_CONV(TENSOR, ...)                                      \
  do {                                                                     \
    if ((TENSOR).scalar_type() == at::kFloat) {                            \
      using scalar_t = float;                                              \
      __VA_ARGS__;                                                         \
    } else if ((TENSOR).scalar_type() == at::kBFloat16) {                  \
      using scalar_t = bf16;                                               \
      __VA_ARGS__;                                                         \
    } else {                                                               \
      TORCH_CHECK(false, "unsupported dtype");                             \
    }                                                                      \
  } while (0)
