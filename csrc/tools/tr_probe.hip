// Probe the exact semantics of __builtin_amdgcn_ds_read_tr16_b64_v4bf16
// on gfx950: fill LDS with bf16(index), issue the read with several
// address patterns, dump [lane][elem] -> source index.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4;
#define __LDS_ADDR __attribute__((address_space(3)))

__global__ void probe(float* out, int pattern) {
  __shared__ __bf16 lds[512];
  int tid = threadIdx.x;
  for (int i = tid; i < 512; i += 64) lds[i] = (__bf16)(float)i;
  __syncthreads();
  int lane = tid & 63;
  int off;
  switch (pattern) {
    case 0: off = 0; break;                          // uniform base
    case 1: off = (lane & 15) * 4; break;            // lane row of 4
    case 2: off = (lane >> 4) * 64; break;           // group base
    case 3: off = (lane & 15) * 4 + (lane >> 4) * 64; break;
    case 4: off = lane * 4; break;                   // fully linear
    case 5: off = 1 + (lane & 15) * 4; break;        // UNALIGNED (2B) test
    case 6: off = 2 + (lane & 15) * 4; break;        // 4B-aligned test
    default: {                                       // intended conv use:
      // B[k][n] tile rows pitch 40; lane m: row m/4 (in 16-group), col 4*(m%4)
      int il = lane & 15, kl = lane >> 4;
      off = (kl * 8 + il / 4) * 40 + 4 * (il % 4);
      break;
    }
  }
  auto p = (__LDS_ADDR bf16x4*)((__LDS_ADDR __bf16*)lds + off);
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  for (int j = 0; j < 4; ++j) out[(lane * 4 + j)] = (float)v[j];
}

int main() {
  float* d;
  hipMalloc(&d, 64 * 4 * sizeof(float));
  float h[256];
  for (int pat = 0; pat <= 7; ++pat) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, pat);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("pattern %d:\n", pat);
    for (int l = 0; l < 64; ++l) {
      printf("L%02d:", l);
      for (int j = 0; j < 4; ++j) printf(" %4.0f", h[l * 4 + j]);
      printf(l % 4 == 3 ? "\n" : "   ");
    }
    printf("\n");
  }
  hipFree(d);
  return 0;
}
