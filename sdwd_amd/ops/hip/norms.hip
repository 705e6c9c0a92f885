// Fused normalisation kernels.
//
// group_norm_silu: NCHW GroupNorm with the SiLU folded into the second pass
//   (one kernel instead of GN + act: saves one full HBM round trip of the
//   activation tensor — the UNet/VAE call this before every conv).
//   One 256-thread block per (sample, group); a group's elements are
//   contiguous in NCHW so both passes stream 16 B/lane.
//
// layer_norm: row-wise LN over the last dim (transformer blocks).
//   One block per row batch; fp32 statistics.
#include "common.h"

// ---------------------------------------------------------------------------
// GroupNorm(+SiLU), bf16
// grid.x = N * G   block = 256
// ---------------------------------------------------------------------------
template <bool SILU, bool VEC>
__global__ void group_norm_silu_bf16_kernel(
    const __hip_bfloat16 *__restrict__ x, const float *__restrict__ w,
    const float *__restrict__ b, __hip_bfloat16 *__restrict__ out, int N,
    int C, long HW, int G, float eps) {
  __shared__ float lds[16];
  const int ng = blockIdx.x;
  const int n = ng / G, g = ng % G;
  const int cg = C / G;            // channels per group
  const long len = (long)cg * HW;  // elements per (n, g), contiguous
  const __hip_bfloat16 *base = x + (long)n * C * HW + (long)g * len;
  __hip_bfloat16 *obase = out + (base - x);

  float sum = 0.f, sumsq = 0.f;
  if (VEC) {
    const long nv = len / 8;
    for (long i = threadIdx.x; i < nv; i += blockDim.x) {
      bf16x8 v = ((const bf16x8 *)base)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        sum += f;
        sumsq += f * f;
      }
    }
  } else {
    for (long i = threadIdx.x; i < len; i += blockDim.x) {
      float f = bf2f(base[i]);
      sum += f;
      sumsq += f * f;
    }
  }
  block_reduce2<256>(sum, sumsq, lds);
  const float mean = sum / (float)len;
  const float var = sumsq / (float)len - mean * mean;
  const float rstd = rsqrtf(var + eps);

  if (VEC) {
    const long nv = len / 8;
    const long hv = HW / 8;  // vectors per channel (HW % 8 == 0 in VEC mode)
    for (long i = threadIdx.x; i < nv; i += blockDim.x) {
      const int c = g * cg + (int)(i / hv);
      const float scale = rstd * w[c];
      const float shift = b[c] - mean * scale;
      bf16x8 v = ((const bf16x8 *)base)[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j] * scale + shift;
        o[j] = (__bf16)(SILU ? silu_f(f) : f);
      }
      ((bf16x8 *)obase)[i] = o;
    }
  } else {
    for (long i = threadIdx.x; i < len; i += blockDim.x) {
      const int c = g * cg + (int)(i / HW);
      float f = (bf2f(base[i]) - mean) * rstd * w[c] + b[c];
      obase[i] = f2bf(SILU ? silu_f(f) : f);
    }
  }
}

// fp32 variant (debug / CPU-parity checks on GPU)
template <bool SILU>
__global__ void group_norm_silu_f32_kernel(const float *__restrict__ x,
                                           const float *__restrict__ w,
                                           const float *__restrict__ b,
                                           float *__restrict__ out, int N,
                                           int C, long HW, int G, float eps) {
  __shared__ float lds[16];
  const int ng = blockIdx.x;
  const int n = ng / G, g = ng % G;
  const int cg = C / G;
  const long len = (long)cg * HW;
  const float *base = x + (long)n * C * HW + (long)g * cg * HW;
  float *obase = out + (base - x);
  float sum = 0.f, sumsq = 0.f;
  for (long i = threadIdx.x; i < len; i += blockDim.x) {
    float f = base[i];
    sum += f;
    sumsq += f * f;
  }
  block_reduce2<256>(sum, sumsq, lds);
  const float mean = sum / (float)len;
  const float rstd = rsqrtf(sumsq / (float)len - mean * mean + eps);
  for (long i = threadIdx.x; i < len; i += blockDim.x) {
    const int c = g * cg + (int)(i / HW);
    float f = (base[i] - mean) * rstd * w[c] + b[c];
    obase[i] = SILU ? silu_f(f) : f;
  }
}

// ---------------------------------------------------------------------------
// LayerNorm over rows [R, D], bf16. One wave per row (D <= a few K).
// block = 256 = 4 waves -> 4 rows per block.
// ---------------------------------------------------------------------------
__global__ void layer_norm_bf16_kernel(const __hip_bfloat16 *__restrict__ x,
                                       const float *__restrict__ w,
                                       const float *__restrict__ b,
                                       __hip_bfloat16 *__restrict__ out,
                                       long R, int D, float eps) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long row = (long)blockIdx.x * 4 + wid;
  if (row >= R) return;
  const __hip_bfloat16 *base = x + row * D;
  __hip_bfloat16 *obase = out + row * D;

  float sum = 0.f, sumsq = 0.f;
  const bool vec = (D % 8) == 0;
  if (vec) {
    const int nv = D / 8;
    for (int i = lane; i < nv; i += WAVE) {
      bf16x8 v = ((const bf16x8 *)base)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        sum += f;
        sumsq += f * f;
      }
    }
  } else {
    for (int i = lane; i < D; i += WAVE) {
      float f = bf2f(base[i]);
      sum += f;
      sumsq += f * f;
    }
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off, WAVE);
    sumsq += __shfl_down(sumsq, off, WAVE);
  }
  sum = __shfl(sum, 0, WAVE);
  sumsq = __shfl(sumsq, 0, WAVE);
  const float mean = sum / D;
  const float rstd = rsqrtf(sumsq / D - mean * mean + eps);

  if (vec) {
    const int nv = D / 8;
    for (int i = lane; i < nv; i += WAVE) {
      bf16x8 v = ((const bf16x8 *)base)[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d = i * 8 + j;
        o[j] = (__bf16)(((float)v[j] - mean) * rstd * w[d] + b[d]);
      }
      ((bf16x8 *)obase)[i] = o;
    }
  } else {
    for (int i = lane; i < D; i += WAVE)
      obase[i] = f2bf((bf2f(base[i]) - mean) * rstd * w[i] + b[i]);
  }
}

__global__ void layer_norm_f32_kernel(const float *__restrict__ x,
                                      const float *__restrict__ w,
                                      const float *__restrict__ b,
                                      float *__restrict__ out, long R, int D,
                                      float eps) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long row = (long)blockIdx.x * 4 + wid;
  if (row >= R) return;
  const float *base = x + row * D;
  float sum = 0.f, sumsq = 0.f;
  for (int i = lane; i < D; i += WAVE) {
    float f = base[i];
    sum += f;
    sumsq += f * f;
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off, WAVE);
    sumsq += __shfl_down(sumsq, off, WAVE);
  }
  sum = __shfl(sum, 0, WAVE);
  sumsq = __shfl(sumsq, 0, WAVE);
  const float mean = sum / D;
  const float rstd = rsqrtf(sumsq / D - mean * mean + eps);
  for (int i = lane; i < D; i += WAVE)
    out[row * D + i] = (base[i] - mean) * rstd * w[i] + b[i];
}

// ---------------------------------------------------------------------------
// GroupNorm(+SiLU) for channels_last (NHWC) — the conv-native layout on
// gfx950 (MIOpen NHWC solvers are ~30% faster and need no transposes).
// 3 phases: per-channel partial sums (coalesced C-contiguous loads),
// per-(n,g) stat finalize, fused normalize+SiLU sweep.
// ---------------------------------------------------------------------------
// Threads own vec8 channel columns: thread (ro, vc) reads rows hw0+ro,
// hw0+ro+RP, ... at 16 B per lane (consecutive vc -> consecutive 16 B:
// fully coalesced). RP rows are reduced per block pass; the stats kernel
// sums over S*RP partial slots.
__global__ void gn_nhwc_partial_bf16(const __hip_bfloat16 *__restrict__ x,
                                     float *__restrict__ partial,
                                     // [N*S*RP][2C]
                                     int C, long HW, int S, int RP) {
  const int ns = blockIdx.x;      // n * S + s
  const int n = ns / S, sc = ns % S;
  const long chunk = (HW + S - 1) / S;
  const long hw0 = sc * chunk;
  const long hw1 = min(HW, hw0 + chunk);
  const __hip_bfloat16 *base = x + (long)n * HW * C;

  const int VC = C / 8;                       // vec8 columns
  const int cpt8 = (VC + 255) / 256;          // vec cols per thread (<=2)
  const int vc0 = (int)threadIdx.x % (cpt8 == 1 ? VC : 256);
  const int ro = cpt8 == 1 ? (int)threadIdx.x / VC : 0;
  float sum[2][8], sumsq[2][8];
#pragma unroll
  for (int j2 = 0; j2 < 2; ++j2)
#pragma unroll
    for (int j = 0; j < 8; ++j) sum[j2][j] = sumsq[j2][j] = 0.f;

  if (ro < RP) {
    for (long hw = hw0 + ro; hw < hw1; hw += RP) {
      const __hip_bfloat16 *row = base + hw * C;
      for (int j2 = 0; j2 < cpt8; ++j2) {
        const int vc = j2 * 256 + vc0;
        if (vc < VC) {
          bf16x8 v = ((const bf16x8 *)row)[vc];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float f = (float)v[j];
            sum[j2][j] += f;
            sumsq[j2][j] += f * f;
          }
        }
      }
    }
  }
  float *out = partial + ((long)ns * RP + ro) * 2 * C;
  if (ro < RP) {
    for (int j2 = 0; j2 < cpt8; ++j2) {
      const int vc = j2 * 256 + vc0;
      if (vc < VC) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          out[vc * 8 + j] = sum[j2][j];
          out[C + vc * 8 + j] = sumsq[j2][j];
        }
      }
    }
  }
}

__global__ void gn_nhwc_stats(const float *__restrict__ partial,
                              float *__restrict__ stats,  // [N*G][2]
                              int C, long HW, int G, int S, float eps) {
  const int ng = blockIdx.x;
  const int n = ng / G, g = ng % G;
  const int cg = C / G;
  float sum = 0.f, sumsq = 0.f;
  for (int idx = threadIdx.x; idx < S * cg; idx += WAVE) {
    const int s = idx / cg, c = g * cg + idx % cg;
    const float *p = partial + ((long)n * S + s) * 2 * C;
    sum += p[c];
    sumsq += p[C + c];
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off, WAVE);
    sumsq += __shfl_down(sumsq, off, WAVE);
  }
  if (threadIdx.x == 0) {
    const float cnt = (float)(HW * cg);
    const float mean = sum / cnt;
    const float rstd = rsqrtf(sumsq / cnt - mean * mean + eps);
    stats[ng * 2] = mean;
    stats[ng * 2 + 1] = rstd;
  }
}

template <bool SILU>
__global__ void gn_nhwc_norm_bf16(const __hip_bfloat16 *__restrict__ x,
                                  const float *__restrict__ stats,
                                  const float *__restrict__ w,
                                  const float *__restrict__ b,
                                  __hip_bfloat16 *__restrict__ out, int C,
                                  long HW, int G, long N) {
  const int cg = C / G;
  const long total = N * HW * (C / 8);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cv = (int)(i % (C / 8));       // 8-channel vector index
    const long nhw = i / (C / 8);
    const long n = nhw / HW;
    bf16x8 v = ((const bf16x8 *)x)[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = cv * 8 + j;
      const int g = c / cg;
      const float mean = stats[(n * G + g) * 2];
      const float rstd = stats[(n * G + g) * 2 + 1];
      float f = ((float)v[j] - mean) * rstd * w[c] + b[c];
      o[j] = (__bf16)(SILU ? silu_f(f) : f);
    }
    ((bf16x8 *)out)[i] = o;
  }
}

// ---------------------------------------------------------------------------
// Fused residual-add + LayerNorm: sum = x + res; ln = LN(sum)*w + b.
// The transformer pre-norm chain needs both tensors; fusing saves one full
// HBM round trip per residual. One wave per row, row held in registers
// (D <= 1536).
// ---------------------------------------------------------------------------
__global__ void add_layer_norm_bf16_kernel(
    const __hip_bfloat16 *__restrict__ x, const __hip_bfloat16 *__restrict__ r,
    const float *__restrict__ w, const float *__restrict__ b,
    __hip_bfloat16 *__restrict__ out_sum, __hip_bfloat16 *__restrict__ out_ln,
    long R, int D, float eps) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long row = (long)blockIdx.x * 4 + wid;
  if (row >= R) return;
  const int nv = D / 8;  // D % 8 == 0 guaranteed by dispatch
  const __hip_bfloat16 *xb = x + row * D;
  const __hip_bfloat16 *rb = r + row * D;

  float vals[24 * 8];  // up to 3 vec8 per lane at D=1536
  float sum = 0.f, sumsq = 0.f;
  int cnt = 0;
  for (int i = lane; i < nv; i += WAVE, ++cnt) {
    bf16x8 vx = ((const bf16x8 *)xb)[i];
    bf16x8 vr = ((const bf16x8 *)rb)[i];
    bf16x8 vs;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)vx[j] + (float)vr[j];
      vals[cnt * 8 + j] = f;
      sum += f;
      sumsq += f * f;
      vs[j] = (__bf16)f;
    }
    ((bf16x8 *)(out_sum + row * D))[i] = vs;
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off, WAVE);
    sumsq += __shfl_down(sumsq, off, WAVE);
  }
  sum = __shfl(sum, 0, WAVE);
  sumsq = __shfl(sumsq, 0, WAVE);
  const float mean = sum / D;
  const float rstd = rsqrtf(sumsq / D - mean * mean + eps);

  cnt = 0;
  for (int i = lane; i < nv; i += WAVE, ++cnt) {
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = i * 8 + j;
      o[j] = (__bf16)((vals[cnt * 8 + j] - mean) * rstd * w[d] + b[d]);
    }
    ((bf16x8 *)(out_ln + row * D))[i] = o;
  }
}
