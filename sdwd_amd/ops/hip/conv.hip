// Implicit-GEMM 3x3 convolution for gfx950, NHWC (channels_last), bf16.
//
// GEMM view: C[M=N*Ho*Wo, N'=Cout] = A[M, K=9*Cin] x B[K, N'], where A is
// the implicit im2col of the activations (never materialised: per-plane
// (dy,dx) address offsets are constant across pixels, so each staged row
// caches its base address + (ho,wo) once and adds a plane offset) and
// B is the weight in [Cout][3][3][Cin] (torch channels_last) layout read
// as rows of K.
//
// Structure = the canonical 128x128-tile MFMA GEMM (guide §5): 4 waves in
// 2x2, each computing a 64x64 sub-tile as 4x4 mfma_f32_16x16x32_bf16
// fragments; A/B tiles staged in LDS as [128][BK=32] rows of 64 B with a
// ((row&3)<<4) XOR byte swizzle so the b128 fragment reads are
// conflict-free; bias and an optional residual add are fused into the
// epilogue (the ResBlock skip-add never touches HBM separately).
//
// Requirements: Cin % 32 == 0 (all SD/VAE hot shapes; small-Cin stem convs
// stay on MIOpen), any Cout, stride 1 or 2, pad 1, kernel 3x3.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4c;

#define CONV_BM 128
#define CONV_BN 128
#define CONV_BK 32

// XOR-permute the 4 16-byte chunks of a 64-byte LDS row by bits 2-3 of the
// row index: a b128 fragment read's 16-lane group touches 16 consecutive
// rows at one chunk, and rows r/r+4/r+8/r+12 would share a bank with a
// plain layout (row stride 16 dwords); (row>>2)&3 separates them.
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return byte_in_row ^ (((row >> 2) & 3) << 4);
}

template <bool HAS_BIAS, bool HAS_RES, bool HAS_CB>
__launch_bounds__(256, 2) __global__ void conv3x3_nhwc_bf16_kernel(
    const __hip_bfloat16 *__restrict__ X,   // [N,H,W,Cin]
    const __hip_bfloat16 *__restrict__ Wt,  // [Cout,3,3,Cin]
    const float *__restrict__ bias,         // [Cout] or null
    const __hip_bfloat16 *__restrict__ Res, // [N,Ho,Wo,Cout] or null
    const __hip_bfloat16 *__restrict__ CB,  // [N,Cout] per-sample chan bias
    __hip_bfloat16 *__restrict__ Y,         // [N,Ho,Wo,Cout]
    int Nn, int H, int W, int Cin, int Cout, int Ho, int Wo, int stride) {
  __shared__ __align__(16) __bf16 at[CONV_BM][CONV_BK];
  __shared__ __align__(16) __bf16 bt[CONV_BN][CONV_BK];

  const long M = (long)Nn * Ho * Wo;
  const long m0 = (long)blockIdx.x * CONV_BM;
  const int n0 = blockIdx.y * CONV_BN;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wm = (wid >> 1) * 64;  // wave row offset in tile
  const int wn = (wid & 1) * 64;   // wave col offset

  // ---- per-thread staged-row bookkeeping --------------------------------
  // staging: 128 rows x 32 cols = 4096 bf16 = 512 vec8; 256 threads -> 2
  // vec8 each; thread t stages rows r = t/2 (A) with col half t%2.
  const int arow = tid >> 1;        // 0..127: two threads stage one row
  const int ehalf = (tid & 1) * 16; // element offset of this half-row
  long abase;
  int ho_s, wo_s;
  bool mvalid;
  {
    const long m = m0 + arow;
    mvalid = m < M;
    const long mm = mvalid ? m : (M - 1);
    const int n_img = (int)(mm / ((long)Ho * Wo));
    const int rem = (int)(mm % ((long)Ho * Wo));
    ho_s = rem / Wo;
    wo_s = rem % Wo;
    abase = (((long)n_img * H + ho_s * stride) * W + wo_s * stride) * Cin;
  }

  f32x4c acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4c){};

  const int kc_per_plane = Cin / CONV_BK;
  for (int plane = 0; plane < 9; ++plane) {
    const int dy = plane / 3 - 1, dx = plane % 3 - 1;
    const int hi = ho_s * stride + dy;
    const int wi = wo_s * stride + dx;
    const bool pvalid =
        mvalid && hi >= 0 && hi < H && wi >= 0 && wi < W;
    const long poff = ((long)dy * W + dx) * Cin;

    for (int kc = 0; kc < kc_per_plane; ++kc) {
      __syncthreads();
      // stage A: rows = output pixels, 32 ci of this plane/chunk
      {
        bf16x8 v0 = (bf16x8){}, v1 = (bf16x8){};
        if (pvalid) {
          const __hip_bfloat16 *src = X + abase + poff + kc * CONV_BK;
          v0 = *(const bf16x8 *)(src + ehalf);
          v1 = *(const bf16x8 *)(src + ehalf + 8);
        }
        char *arow_p = (char *)&at[arow][0];
        *(bf16x8 *)(arow_p + swz(arow, ehalf * 2)) = v0;
        *(bf16x8 *)(arow_p + swz(arow, ehalf * 2 + 16)) = v1;
      }
      // stage B: rows = cout, same 32-k chunk from [Cout][9*Cin]
      {
        const int co = n0 + arow;
        bf16x8 v0 = (bf16x8){}, v1 = (bf16x8){};
        if (co < Cout) {
          const __hip_bfloat16 *src =
              Wt + (long)co * 9 * Cin + plane * Cin + kc * CONV_BK;
          v0 = *(const bf16x8 *)(src + ehalf);
          v1 = *(const bf16x8 *)(src + ehalf + 8);
        }
        char *brow_p = (char *)&bt[arow][0];
        *(bf16x8 *)(brow_p + swz(arow, ehalf * 2)) = v0;
        *(bf16x8 *)(brow_p + swz(arow, ehalf * 2 + 16)) = v1;
      }
      __syncthreads();

      // compute: each wave 4x4 fragments of 16x16x32 over this k-chunk
      const int l16 = lane % 16;
      const int k8 = (lane / 16) * 8;  // 8 consecutive k per lane
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = wm + i * 16 + l16;
        bf16x8 af;
        {
          const char *p = (const char *)&at[am][0];
          af = *(const bf16x8 *)(p + swz(am, k8 * 2));
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int bn = wn + j * 16 + l16;
          bf16x8 bf;
          const char *p = (const char *)&bt[bn][0];
          bf = *(const bf16x8 *)(p + swz(bn, k8 * 2));
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bf, acc[i][j], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: bias, residual, store ----------------------------------
  // C frag 16x16: col = lane%16, row = 4*(lane/16) + reg
  const int l16 = lane % 16;
  const int r4 = (lane / 16) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int co = n0 + wn + j * 16 + l16;
      if (co >= Cout) continue;
      const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wm + i * 16 + r4 + r;
        if (m >= M) continue;
        float v = acc[i][j][r] + bv;
        if (HAS_CB) {
          const int ni = (int)(m / ((long)Ho * Wo));
          v += (float)CB[(long)ni * Cout + co];
        }
        if (HAS_RES) v += (float)Res[m * Cout + co];
        Y[m * Cout + co] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
#ifdef __HIP_PLATFORM_AMD__
bool conv3x3_supported(long cin) { return cin % 32 == 0 && cin >= 32; }

torch::Tensor conv3x3_nhwc(torch::Tensor x, torch::Tensor w_prep,
                           c10::optional<torch::Tensor> bias,
                           c10::optional<torch::Tensor> residual,
                           c10::optional<torch::Tensor> chan_bias,
                           long stride) {
  // x: [N,C,H,W] channels_last; w_prep: [Cout,3,3,Cin] contiguous
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3: x must be channels_last");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_prep.size(0);
  TORCH_CHECK(conv3x3_supported(Cin), "conv3x3: Cin % 32 != 0");
  const int Ho = (H + 2 - 3) / stride + 1;
  const int Wo = (W + 2 - 3) / stride + 1;
  auto y = torch::empty({N, Cout, Ho, Wo},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  const long M = (long)N * Ho * Wo;
  dim3 grid((unsigned)((M + CONV_BM - 1) / CONV_BM),
            (unsigned)((Cout + CONV_BN - 1) / CONV_BN));
  dim3 block(256);
  const bool has_b = bias.has_value();
  const bool has_r = residual.has_value();
  torch::Tensor bf32;
  const float *bptr = nullptr;
  if (has_b) {
    bf32 = bias->to(torch::kFloat).contiguous();
    bptr = bf32.data_ptr<float>();
  }
  const __hip_bfloat16 *rptr = nullptr;
  if (has_r) {
    TORCH_CHECK(
        residual->is_contiguous(torch::MemoryFormat::ChannelsLast));
    rptr = (const __hip_bfloat16 *)residual->data_ptr();
  }
  const bool has_cb = chan_bias.has_value();
  torch::Tensor cbt;
  const __hip_bfloat16 *cbptr = nullptr;
  if (has_cb) {
    cbt = chan_bias->to(torch::kBFloat16).contiguous();
    cbptr = (const __hip_bfloat16 *)cbt.data_ptr();
  }
  auto stream = cur_stream();
#define PICK(B_, R_, C_) conv3x3_nhwc_bf16_kernel<B_, R_, C_>
  auto kern =
      has_b ? (has_r ? (has_cb ? PICK(true, true, true)
                               : PICK(true, true, false))
                     : (has_cb ? PICK(true, false, true)
                               : PICK(true, false, false)))
            : (has_r ? (has_cb ? PICK(false, true, true)
                               : PICK(false, true, false))
                     : (has_cb ? PICK(false, false, true)
                               : PICK(false, false, false)));
#undef PICK
  hipLaunchKernelGGL(kern, grid, block, 0, stream,
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (const __hip_bfloat16 *)w_prep.data_ptr(), bptr, rptr,
                     cbptr, (__hip_bfloat16 *)y.data_ptr(), N, H, W, Cin,
                     Cout, Ho, Wo, (int)stride);
  return y;
}
#endif
