// Plain bf16 GEMM on the conv_v4 skeleton: the guide's 256x256 8-phase
// deep-pipeline structure with linear row addressing instead of im2col.
// Serves the UNet projection / FF shapes (Y[M,N] = X[M,K] @ W[N,K]^T + b)
// where hipBLASLt's picks measured 15-45% MfmaBusy; dispatched per shape
// from the measured winner table in ops/__init__.py.
//
// Same schedule as conv_v4: 512 threads (8 waves, 2Mx4N), BK=64, TWO
// whole-tile LDS buffers (128 KiB, 1 block/CU), 4 quadrant phases per
// K-tile, 2 glds pieces per thread per phase staged B-first/A-last so the
// counted vmcnt(2)/vmcnt(4) waits never drain the queue.
#include "common.h"

#define G2_BM 256
#define G2_BN 256
#define G2_BK 64
#define G2_ATILE (G2_BM * G2_BK)

template <bool HAS_BIAS>
__launch_bounds__(512, 2) __global__ void gemm_v2_kernel(
    const __hip_bfloat16 *__restrict__ X,  // [M,K]
    const __hip_bfloat16 *__restrict__ Wt, // [N,K]
    const float *__restrict__ bias,        // [N] or null
    const __hip_bfloat16 *__restrict__ Zero,
    __hip_bfloat16 *__restrict__ Y,        // [M,N]
    long M, int N, int K) {
  __shared__ __align__(16) __bf16 smem[4 * G2_ATILE];

  const long m0 = ((long)blockIdx.y + (long)blockIdx.z * 32768) * G2_BM;
  const int n0 = blockIdx.x * G2_BN;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wm = (wid >> 2) * 128;
  const int wn = (wid & 3) * 64;

  const int prow = tid / 8;
  const int schunk = tid % 8;
  const int sc8 = schunk * 8;

  long arow[4];
  int bco[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const long m = m0 + prow + 64 * i;
    arow[i] = (m < M) ? m : (M - 1);
    bco[i] = n0 + prow + 64 * i;
  }

  const int NT = K / G2_BK;

  auto stage2 = [&](int t, int b, int ph) {
    const int k0 = t * G2_BK;
    __bf16 *abuf = smem + b * (2 * G2_ATILE);
    __bf16 *bbuf = abuf + G2_ATILE;
    constexpr int PIECES[4][2] = {{4, 5}, {6, 7}, {0, 2}, {1, 3}};
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int i = PIECES[ph][kk];
      if (i < 4) {
        const int row = prow + 64 * i;
        const int swz = cswz(row, schunk);
        const __hip_bfloat16 *asrc = X + arow[i] * K + k0 + swz * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)asrc,
            (__attribute__((address_space(3))) unsigned int
                 *)(abuf + (long)i * 4096 + (long)tid * 8),
            16, 0, 0);
      } else {
        const int bi = i - 4;
        const int row = prow + 64 * bi;
        const int swz = cswz(row, schunk);
        const bool bv = bco[bi] < N;
        const __hip_bfloat16 *bsrc =
            bv ? (Wt + (long)bco[bi] * K + k0 + swz * 8) : Zero;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)bsrc,
            (__attribute__((address_space(3))) unsigned int
                 *)(bbuf + (long)bi * 4096 + (long)tid * 8),
            16, 0, 0);
      }
    }
  };
  (void)sc8;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){};

#pragma unroll
  for (int ph = 0; ph < 4; ++ph) stage2(0, 0, ph);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int l16 = lane % 16;
  const int kq4 = lane / 16;

  for (int t = 0; t < NT; ++t) {
    const __bf16 *abuf = smem + (t & 1) * (2 * G2_ATILE);
    const __bf16 *bbuf = abuf + G2_ATILE;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int mh = ph >> 1, nh = ph & 1;
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int s = 0; s < 2; ++s) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int ar = wm + mh * 64 + i * 16 + l16;
          const int ck = cswz(ar, s * 4 + kq4);
          af[i][s] = *(const bf16x8 *)((const char *)(abuf +
                                                      (long)ar * G2_BK) +
                                       ck * 16);
        }
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const int br = wn + nh * 32 + j * 16 + l16;
          const int ck = cswz(br, s * 4 + kq4);
          bf[j][s] = *(const bf16x8 *)((const char *)(bbuf +
                                                      (long)br * G2_BK) +
                                       ck * 16);
        }
      }
      if (t + 1 < NT) stage2(t + 1, (t + 1) & 1, ph);
      if (ph == 1) {
        if (t + 1 < NT)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else if (ph == 3) {
        if (t + 1 < NT)
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 2; ++s)
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[mh * 4 + i][nh * 2 + j] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af[i][s], bf[j][s], acc[mh * 4 + i][nh * 2 + j], 0, 0,
                    0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  const int r4 = (lane / 16) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int co = n0 + wn + (j >> 1) * 32 + (j & 1) * 16 + l16;
      if (co >= N) continue;
      const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wm + (i >> 2) * 64 + (i & 3) * 16 + r4 + r;
        if (m >= M) continue;
        Y[m * N + co] = f2bf(acc[i][j][r] + bv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
#if defined(__HIP_PLATFORM_AMD__) && !defined(SDWD_NO_TORCH)
bool gemm_v2_supported(long M, long N, long K) {
  return K % 64 == 0 && K >= 128 && N >= 64 && M >= 4096;
}

torch::Tensor gemm_v2(torch::Tensor x2d, torch::Tensor w,
                      c10::optional<torch::Tensor> bias) {
  // x2d: [M,K] bf16 contiguous; w: [N,K] bf16 contiguous (nn.Linear layout)
  TORCH_CHECK(x2d.dim() == 2 && x2d.is_contiguous() &&
              x2d.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous() &&
              w.scalar_type() == torch::kBFloat16);
  const long M = x2d.size(0);
  const int K = (int)x2d.size(1), N = (int)w.size(0);
  TORCH_CHECK(gemm_v2_supported(M, N, K));
  auto y = torch::empty({M, (long)N}, x2d.options());
  static torch::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x2d.device())
    zero_page = torch::zeros({64}, x2d.options());
  torch::Tensor bf32;
  const float *bptr = nullptr;
  if (bias.has_value()) {
    bf32 = bias->to(torch::kFloat).contiguous();
    bptr = bf32.data_ptr<float>();
  }
  const long mtiles = (M + G2_BM - 1) / G2_BM;
  dim3 grid((unsigned)((N + G2_BN - 1) / G2_BN),
            (unsigned)std::min<long>(mtiles, 32768),
            (unsigned)((mtiles + 32767) / 32768));
  dim3 block(512);
  auto kern = bptr ? gemm_v2_kernel<true> : gemm_v2_kernel<false>;
  hipLaunchKernelGGL(kern, grid, block, 0, cur_stream(),
                     (const __hip_bfloat16 *)x2d.data_ptr(),
                     (const __hip_bfloat16 *)w.data_ptr(), bptr,
                     (const __hip_bfloat16 *)zero_page.data_ptr(),
                     (__hip_bfloat16 *)y.data_ptr(), M, N, K);
  return y;
}
#endif
