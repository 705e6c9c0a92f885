// Linear-layer GEMM for gfx950: out[M,N] = x[M,K] @ W^T (+ bias), bf16.
// W is torch nn.Linear layout [N,K] row-major, so the B-tile is read
// exactly like conv.hip's weight rows. Same pipelined structure as the
// conv kernel (128x128 tile, BK=64, global_load_lds double buffering, raw
// barrier + counted vmcnt, source-side XOR chunk swizzle).
// STATUS: experimental / unused by the models — measured on MI355X,
// hipBLASLt beats this 2-phase structure on the SD projection shapes
// (666-1360 TF vs 480-630; tools/gemm_perf.py). The models keep F.linear;
// this stays as the baseline for an 8-phase rewrite (guide §5 template).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4g;

#define GEMM_BM 128
#define GEMM_BN 128
#define GEMM_BK 64
#define GEMM_TILE (GEMM_BM * GEMM_BK)

__device__ __forceinline__ int gswz(int row, int chunk) {
  return chunk ^ ((row >> 1) & 7);
}

template <bool HAS_BIAS>
__launch_bounds__(256, 2) __global__ void linear_bf16_kernel(
    const __hip_bfloat16 *__restrict__ X,   // [M,K]
    const __hip_bfloat16 *__restrict__ Wm,  // [N,K]
    const float *__restrict__ bias,         // [N] or null
    const __hip_bfloat16 *__restrict__ Zero,
    __hip_bfloat16 *__restrict__ Y,         // [M,N]
    long M, int N, int K) {
  __shared__ __align__(16) __bf16 smem[4 * GEMM_TILE];

  const long m0 = (long)blockIdx.x * GEMM_BM;
  const int n0 = blockIdx.y * GEMM_BN;
  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  long arow[4];
  int brow[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = i * 32 + wid * 8 + lane / 8;
    arow[i] = (m0 + row < M) ? (m0 + row) : (M - 1);
    brow[i] = n0 + row;
  }
  const int schunk = lane % 8;
  const int NT = K / GEMM_BK;

  auto stage = [&](int t, int b) {
    const long koff = (long)t * GEMM_BK;
    __bf16 *abuf = smem + b * 2 * GEMM_TILE;
    __bf16 *bbuf = abuf + GEMM_TILE;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = i * 32 + wid * 8 + lane / 8;
      const int sc = gswz(row, schunk);
      const __hip_bfloat16 *asrc = X + arow[i] * K + koff + sc * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)asrc,
          (__attribute__((address_space(3))) unsigned int
               *)(abuf + i * 2048 + wid * 512),
          16, 0, 0);
      const __hip_bfloat16 *bsrc =
          (brow[i] < N) ? (Wm + (long)brow[i] * K + koff + sc * 8) : Zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)bsrc,
          (__attribute__((address_space(3))) unsigned int
               *)(bbuf + i * 2048 + wid * 512),
          16, 0, 0);
    }
  };

  f32x4g acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4g){};

  stage(0, 0);
  if (NT > 1) stage(1, 1);

  const int l16 = lane % 16;
  const int kq = (lane / 16) * 8;

  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const __bf16 *abuf = smem + (t & 1) * 2 * GEMM_TILE;
    const __bf16 *bbuf = abuf + GEMM_TILE;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = wm + i * 16 + l16;
        const int ck = gswz(am, (s * 32 + kq) / 8);
        af[i] = *(const bf16x8 *)((const char *)(abuf + am * GEMM_BK) +
                                  ck * 16);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int bn = wn + j * 16 + l16;
        const int ck = gswz(bn, (s * 32 + kq) / 8);
        bf[j] = *(const bf16x8 *)((const char *)(bbuf + bn * GEMM_BK) +
                                  ck * 16);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
    if (t + 2 < NT) stage(t + 2, t & 1);
  }

  const int r4 = (lane / 16) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wn + j * 16 + l16;
      if (n >= N) continue;
      const float bv = HAS_BIAS ? bias[n] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wm + i * 16 + r4 + r;
        if (m >= M) continue;
        Y[m * N + n] = f2bf(acc[i][j][r] + bv);
      }
    }
  }
}

#ifdef __HIP_PLATFORM_AMD__
bool linear_supported(long k) { return k % 64 == 0 && k >= 64; }

torch::Tensor linear_bf16(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias) {
  // x: [..., K] contiguous bf16; w: [N, K] contiguous bf16
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int K = xc.size(-1);
  const int N = wc.size(0);
  const long M = xc.numel() / K;
  TORCH_CHECK(linear_supported(K), "linear_bf16: K % 64 != 0");
  TORCH_CHECK(xc.scalar_type() == torch::kBFloat16);
  auto sizes = xc.sizes().vec();
  sizes.back() = N;
  auto y = torch::empty(sizes, xc.options());
  static torch::Tensor zpage;
  if (!zpage.defined() || zpage.device() != x.device())
    zpage = torch::zeros({64}, xc.options());
  torch::Tensor bf32;
  const float *bptr = nullptr;
  if (bias.has_value()) {
    bf32 = bias->to(torch::kFloat).contiguous();
    bptr = bf32.data_ptr<float>();
  }
  dim3 grid((unsigned)((M + GEMM_BM - 1) / GEMM_BM),
            (unsigned)((N + GEMM_BN - 1) / GEMM_BN));
  dim3 block(256);
  auto kern = bptr ? linear_bf16_kernel<true> : linear_bf16_kernel<false>;
  hipLaunchKernelGGL(kern, grid, block, 0, cur_stream(),
                     (const __hip_bfloat16 *)xc.data_ptr(),
                     (const __hip_bfloat16 *)wc.data_ptr(), bptr,
                     (const __hip_bfloat16 *)zpage.data_ptr(),
                     (__hip_bfloat16 *)y.data_ptr(), M, N, K);
  return y;
}
#endif
