// MFMA fragment-layout probes: one wave computes a single MFMA tile from
// plain row-major inputs using the assumed lane->element mappings; the GPU
// test compares against torch.matmul on asymmetric random data (a transposed
// or mis-split mapping fails loudly). Assumed gfx950 mappings:
//   32x32x16 bf16: A[row=l%32][k=8*(l/32)+i], B[k=8*(l/32)+i][col=l%32],
//                  C col=l%32, row=(reg&3)+8*(reg>>2)+4*(l>>5), reg in [0,16)
//   16x16x32 bf16: A[row=l%16][k=8*(l/16)+i], B[k=8*(l/16)+i][col=l%16],
//                  C col=l%16, row=4*(l>>4)+reg, reg in [0,4)
#include "common.h"

__global__ void probe_mfma_32x32x16(const float *__restrict__ A,
                                    const float *__restrict__ B,
                                    float *__restrict__ C) {
  const int l = threadIdx.x;  // one wave
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)A[(l % 32) * 16 + 8 * (l / 32) + i];
    b[i] = (__bf16)B[(8 * (l / 32) + i) * 32 + (l % 32)];
  }
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    C[row * 32 + (l % 32)] = c[r];
  }
}

// ds_read_b64_tr_b16 semantics probe: LDS holds bf16 element e = e (its own
// index); each lane issues one tr read at addr = base + lane*stride and the
// 4 received bf16 are dumped per lane. The host decodes which LDS element
// each (lane, j) slot received, settling the exact transpose mapping for
// the attention V path (guide T10 has no worked example on disk).
__global__ void probe_tr_b16(float *__restrict__ out, int stride_bytes,
                             int base_bytes) {
  __shared__ __align__(16) __bf16 buf[2048];
  const int l = threadIdx.x;
  for (int i = l; i < 2048; i += 64) buf[i] = (__bf16)(float)i;
  __syncthreads();
  unsigned addr =
      (unsigned)(unsigned long long)(const char *)buf + base_bytes +
      (unsigned)l * (unsigned)stride_bytes;
  unsigned long long d = 0;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
      : "=v"(d)
      : "v"(addr)
      : "memory");
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const unsigned short bits = (unsigned short)(d >> (16 * j));
    union {
      unsigned u;
      float f;
    } cv;
    cv.u = ((unsigned)bits) << 16;  // bf16 -> f32
    out[l * 4 + j] = cv.f;
  }
}

__global__ void probe_mfma_16x16x32(const float *__restrict__ A,
                                    const float *__restrict__ B,
                                    float *__restrict__ C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)A[(l % 16) * 32 + 8 * (l / 16) + i];
    b[i] = (__bf16)B[(8 * (l / 16) + i) * 16 + (l % 16)];
  }
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = 4 * (l / 16) + r;
    C[row * 16 + (l % 16)] = c[r];
  }
}
