// Row softmax, in-place, fp32 [R, S]: interim fused-softmax for the
// matmul-composed attention path (replaced by the MFMA flash-forward kernel
// for the hot shapes; this stays for debug & odd shapes).
// One wave per row; 3 register passes (max, exp-sum, scale) over L2-resident
// rows; scale folded into the first pass load.
#include "common.h"

__global__ void row_softmax_f32_kernel(float *__restrict__ x, long R, int S,
                                       float scale) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= R) return;
  float *base = x + row * S;

  float m = -1e30f;
  for (int i = lane; i < S; i += WAVE) m = fmaxf(m, base[i] * scale);
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_down(m, off, WAVE));
  m = __shfl(m, 0, WAVE);

  float s = 0.f;
  for (int i = lane; i < S; i += WAVE) {
    float e = __expf(base[i] * scale - m);
    base[i] = e;
    s += e;
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) s += __shfl_down(s, off, WAVE);
  s = __shfl(s, 0, WAVE);
  const float inv = 1.0f / s;
  for (int i = lane; i < S; i += WAVE) base[i] *= inv;
}

__global__ void row_softmax_bf16_kernel(__hip_bfloat16 *__restrict__ x, long R,
                                        int S, float scale) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (row >= R) return;
  __hip_bfloat16 *base = x + row * S;

  float m = -1e30f;
  for (int i = lane; i < S; i += WAVE) m = fmaxf(m, bf2f(base[i]) * scale);
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_down(m, off, WAVE));
  m = __shfl(m, 0, WAVE);

  float s = 0.f;
  for (int i = lane; i < S; i += WAVE) {
    float e = __expf(bf2f(base[i]) * scale - m);
    base[i] = f2bf(e);
    s += e;
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) s += __shfl_down(s, off, WAVE);
  s = __shfl(s, 0, WAVE);
  const float inv = 1.0f / s;
  for (int i = lane; i < S; i += WAVE) base[i] = f2bf(bf2f(base[i]) * inv);
}
