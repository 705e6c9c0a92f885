// Common device helpers for the sdwd_amd gfx950 kernels.
// CDNA4 only: wave64, 256 CUs / 8 XCDs, LDS 160 KiB/CU (see
// /opt/skills guide notes mirrored in ops/README.md).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__device__ __forceinline__ float bf2f(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ __hip_bfloat16 f2bf(float v) {
  return __float2bfloat16(v);
}

// exact-GELU (erf form, matches torch.nn.functional.gelu default)
__device__ __forceinline__ float gelu_f(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

__device__ __forceinline__ float silu_f(float x) {
  return x / (1.0f + __expf(-x));
}

// block-wide sum reduction of up to 2 values, 256 threads, returns on lane 0
// of wave 0 and broadcasts via LDS.
template <int BLOCK>
__device__ __forceinline__ void block_reduce2(float &a, float &b,
                                              float *lds /*>= 2*BLOCK/WAVE*/) {
  // wave reduce
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    a += __shfl_down(a, off, WAVE);
    b += __shfl_down(b, off, WAVE);
  }
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  constexpr int NW = BLOCK / WAVE;
  if (lane == 0) {
    lds[wid * 2] = a;
    lds[wid * 2 + 1] = b;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float sa = 0.f, sb = 0.f;
#pragma unroll
    for (int i = 0; i < NW; ++i) {
      sa += lds[i * 2];
      sb += lds[i * 2 + 1];
    }
    lds[0] = sa;
    lds[1] = sb;
  }
  __syncthreads();
  a = lds[0];
  b = lds[1];
}
