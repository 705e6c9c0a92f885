// Direct 3x3 convolution for tiny input-channel counts (the stem/IO convs:
// UNet conv_in Cin=4/9, VAE encoder conv_in Cin=3, VAE decoder conv_in
// Cin=4). K = 9*Cin is far too small for the implicit-GEMM MFMA path, and
// MIOpen's fallback solvers fire workspace warnings inside the headline
// bench — this kernel keeps the whole pipeline native.
//
// Shape of the work: M = N*Ho*Wo pixels x Cout outputs, 9*Cin MACs each.
// Weights live in LDS transposed to [tap][Cout] so each (pixel, 8-cout)
// thread slot reads its 8 weights per tap as ONE ds_read_b128 (broadcast
// across the pixels of a chunk); activations are 9*Cin bf16 per pixel,
// held in registers.
#include "common.h"

template <bool HAS_BIAS, int CIN>
__launch_bounds__(256, 4) __global__ void conv3x3_smallcin_kernel(
    const __hip_bfloat16 *__restrict__ X,   // [N,H,W,Cin] channels_last
    const __hip_bfloat16 *__restrict__ Wt,  // [Cout,3,3,Cin]
    const float *__restrict__ bias,         // [Cout] or null
    __hip_bfloat16 *__restrict__ Y,         // [N,Ho,Wo,Cout]
    int Nn, int H, int W, int Cout, int Ho, int Wo, int stride) {
  constexpr int TAPS = 9 * CIN;
  constexpr int PIX = 64;                  // pixels per block
  extern __shared__ __align__(16) __bf16 wl[];  // [TAPS][Cout]

  const long M = (long)Nn * Ho * Wo;
  const long m0 = (long)blockIdx.x * PIX;

  // cooperative W transpose into LDS: [cout][tap] -> [tap][cout]
  for (int idx = threadIdx.x; idx < TAPS * Cout; idx += 256) {
    const int co = idx / TAPS;
    const int t = idx % TAPS;
    wl[t * Cout + co] = ((const __bf16 *)Wt)[co * TAPS + t];
  }
  __syncthreads();

  // thread slot: pixel = m0 + tid/4, cout chunks tid%4 + 4*i
  const int pl = threadIdx.x / 4;
  const int c0 = threadIdx.x % 4;
  const long m = m0 + pl;
  const long mm = (m < M) ? m : (M - 1);
  const int n_img = (int)(mm / ((long)Ho * Wo));
  const int rem = (int)(mm % ((long)Ho * Wo));
  const int ho = rem / Wo, wo = rem % Wo;
  const int hi = ho * stride, wi = wo * stride;

  // activations for the 9 taps (zero outside the image)
  float xa[TAPS];
#pragma unroll
  for (int t = 0; t < 9; ++t) {
    const int dy = t / 3 - 1, dx = t % 3 - 1;
    const int h = hi + dy, w = wi + dx;
    const bool ok = h >= 0 && h < H && w >= 0 && w < W;
    const __hip_bfloat16 *src =
        X + (((long)n_img * H + h) * W + w) * CIN;
#pragma unroll
    for (int c = 0; c < CIN; ++c)
      xa[t * CIN + c] = ok ? (float)src[c] : 0.0f;
  }

  const int nchunk = (Cout + 7) / 8;
  for (int ch = c0; ch < nchunk; ch += 4) {
    const int co = ch * 8;
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] = (HAS_BIAS && co + j < Cout) ? bias[co + j] : 0.0f;
#pragma unroll
    for (int t = 0; t < TAPS; ++t) {
      const bf16x8 w8 = *(const bf16x8 *)&wl[t * Cout + co];
      const float xv = xa[t];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += xv * (float)w8[j];
    }
    if (m < M) {
      __hip_bfloat16 *dst = Y + m * Cout + co;
      if (co + 8 <= Cout) {
        bf16x8 o8;
#pragma unroll
        for (int j = 0; j < 8; ++j) o8[j] = (__bf16)acc[j];
        *(bf16x8 *)dst = o8;
      } else {
        for (int j = 0; j < Cout - co; ++j) dst[j] = f2bf(acc[j]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
#if defined(__HIP_PLATFORM_AMD__) && !defined(SDWD_NO_TORCH)
bool conv3x3_small_supported(long cin, long cout) {
  // the [tap][Cout] LDS weight cache (9*Cin*Cout bf16) must fit a CU
  return (cin == 3 || cin == 4 || cin == 9) && cout % 8 == 0 &&
         cout <= 1536 && 9 * cin * cout * 2 <= 160 * 1024;
}

torch::Tensor conv3x3_small(torch::Tensor x, torch::Tensor w_prep,
                            c10::optional<torch::Tensor> bias,
                            long stride) {
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3_small: x must be channels_last");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_prep.size(0);
  TORCH_CHECK(conv3x3_small_supported(Cin, Cout));
  const int Ho = (H + 2 - 3) / (int)stride + 1;
  const int Wo = (W + 2 - 3) / (int)stride + 1;
  auto y = torch::empty({N, Cout, Ho, Wo},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  const long M = (long)N * Ho * Wo;
  const bool has_b = bias.has_value();
  torch::Tensor bf32;
  const float *bptr = nullptr;
  if (has_b) {
    bf32 = bias->to(torch::kFloat).contiguous();
    bptr = bf32.data_ptr<float>();
  }
  dim3 grid((unsigned)((M + 63) / 64)), block(256);
  const size_t lds = (size_t)9 * Cin * Cout * 2;
  auto stream = cur_stream();
#define LAUNCH_SC(CIN_, HB_)                                                \
  hipLaunchKernelGGL((conv3x3_smallcin_kernel<HB_, CIN_>), grid, block,     \
                     lds, stream, (const __hip_bfloat16 *)x.data_ptr(),     \
                     (const __hip_bfloat16 *)w_prep.data_ptr(), bptr,       \
                     (__hip_bfloat16 *)y.data_ptr(), N, H, W, Cout, Ho,     \
                     Wo, (int)stride)
  if (Cin == 3) {
    if (has_b) LAUNCH_SC(3, true); else LAUNCH_SC(3, false);
  } else if (Cin == 4) {
    if (has_b) LAUNCH_SC(4, true); else LAUNCH_SC(4, false);
  } else {
    if (has_b) LAUNCH_SC(9, true); else LAUNCH_SC(9, false);
  }
#undef LAUNCH_SC
  return y;
}
#endif
