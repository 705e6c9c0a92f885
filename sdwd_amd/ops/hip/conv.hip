// Implicit-GEMM 3x3 convolution for gfx950, NHWC (channels_last), bf16.
//
// GEMM view: C[M=N*Ho*Wo, Cout] = A[M, K=9*Cin] x B[K, Cout]; A is the
// implicit im2col of the activations (per-(dy,dx)-plane address offsets are
// constant across pixels, so each staged row caches its base address once),
// B is the weight in [Cout][3][3][Cin] (torch channels_last) layout.
//
// v2 structure (guide §5 step-3 + 'glds, 2 LDS buffers' row):
//  * 128x128 tile, BK=64; 4 waves in 2x2, each wave a 64x64 sub-tile as
//    4x4(x2 k-sub) mfma_f32_16x16x32_bf16;
//  * A/B tiles staged by global_load_lds (16 B/lane direct-to-LDS DMA),
//    double-buffered in ONE __shared__ array (a second __shared__ object
//    makes hipcc drain vmcnt before every ds_read - guide §5 trap (a));
//  * tile t+1's 8 DMA instructions stay in flight across the barrier:
//    raw s_barrier + counted `s_waitcnt vmcnt(8)` (a plain __syncthreads
//    would drain the pipeline);
//  * LDS rows are 128 B; the 8 16-B chunks are XOR-permuted by (row>>1)&7
//    applied to the DMA *source* address (rule 21: the glds destination is
//    lane-linear) so the b128 fragment reads are bank-conflict-free;
//  * padding rows ride a 64-B zero page (glds is wave-uniform, lanes with
//    out-of-image taps just read zeros);
//  * bias, per-(sample,channel) bias (time embedding) and the ResBlock
//    residual add are fused into the epilogue.
//
// Requirements: Cin % 64 == 0 (every SD1.5/SDXL/VAE hot shape; small-Cin
// stem convs stay on MIOpen), any Cout, stride 1 or 2, pad 1, kernel 3x3.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4c;

#define CONV_BM 128
#define CONV_BN 128
#define CONV_BK 64
#define TILE_ELEMS (CONV_BM * CONV_BK)  // per operand per buffer

// XOR-permute the 8 16-byte chunks of a 128-byte LDS row: a b128 fragment
// read's 16-lane group touches 16 consecutive rows at one chunk; rows of
// equal parity would collide every 2 rows at a 32-dword stride, so spread
// by (row>>1)&7.
// cswz() lives in conv_v4.hip (included first)

// BN = 128 (default) or 64 (exact tiling for Cout % 128 == 64, e.g. the
// SD1.5 320-channel level: 5 exact 64-col tiles instead of 3 x 128 with a
// 17% masked-FLOP tail). NJ = column fragments per wave.
// UPS: the input is a VIRTUAL nearest-2x upsample of X (H/W are the REAL
// input dims, Ho/Wo the upsampled-output dims): tap (h,w) of the virtual
// image reads X[h>>1][w>>1], fusing Upsample+conv into one kernel with no
// 4x intermediate tensor.
template <bool HAS_BIAS, bool HAS_RES, bool HAS_CB, int BN = 128,
          bool UPS = false>
__launch_bounds__(256, 2) __global__ void conv3x3_nhwc_bf16_kernel(
    const __hip_bfloat16 *__restrict__ X,   // [N,H,W,Cin]
    const __hip_bfloat16 *__restrict__ Wt,  // [Cout,3,3,Cin]
    const float *__restrict__ bias,         // [Cout] or null
    const __hip_bfloat16 *__restrict__ Res, // [N,Ho,Wo,Cout] or null
    const __hip_bfloat16 *__restrict__ CB,  // [N,Cout] or null
    const __hip_bfloat16 *__restrict__ Zero,// >=16B of zeros
    __hip_bfloat16 *__restrict__ Y,         // [N,Ho,Wo,ldY] (+col offset)
    int Nn, int H, int W, int Cin, int Cout, int Ho, int Wo, int stride,
    int ldY, float *__restrict__ GNP = nullptr) {
  // GNP != null: emit per-(M-tile, channel) partial (sum, sumsq) of the
  // STORED values into GNP[mtile][2][ldY] so GroupNorm's stats pass can
  // skip its full-tensor read (host enables only when Ho*Wo % BM == 0,
  // i.e. every tile lies inside one image). Deterministic: wave shfl
  // reduce + fixed-order cross-wave combine (no atomics).
  // one __shared__ object: [buf0: A(8K elems) B(8K)][buf1: A B]
  __shared__ __align__(16) __bf16 smem[4 * TILE_ELEMS];

  const long M = (long)Nn * Ho * Wo;
  // grid: x = Cout tiles, y (+z overflow) = M tiles, so consecutively-
  // dispatched blocks are the column tiles of ONE row tile and share its
  // A reads in L2/L3
  constexpr int NJ = BN / 32;          // 4 or 2 column fragments per wave
  constexpr int BTILE = BN * CONV_BK;  // B-tile elements
  const long m0 =
      ((long)blockIdx.y + (long)blockIdx.z * 32768) * CONV_BM;
  const int n0 = blockIdx.x * BN;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * (BN / 2);

  // ---- per-lane DMA row bookkeeping -------------------------------------
  // stage instr i (0..3) per wave writes LDS bytes
  // [i*4096 + wid*1024 + lane*16]; linear row = byte/128:
  //   row_i = i*32 + wid*8 + lane/8, chunk = lane%8 (then source-swizzled)
  long abase[4];
  int hs[4], ws[4];   // ho*stride, wo*stride per staged A row
  int bco[4];         // cout row per staged B row (NB instrs used)
  constexpr int NB = BN / 32;  // B stage instrs (rows BN over 4 waves x 8)
  int nimg[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = i * 32 + wid * 8 + lane / 8;
    const long m = m0 + row;
    const long mm = (m < M) ? m : (M - 1);
    const int n_img = (int)(mm / ((long)Ho * Wo));
    const int rem = (int)(mm % ((long)Ho * Wo));
    hs[i] = (rem / Wo) * stride;
    ws[i] = (rem % Wo) * stride;
    nimg[i] = n_img;
    abase[i] = (((long)n_img * H + hs[i]) * W + ws[i]) * Cin;
    bco[i] = n0 + row;
  }
  const int schunk = lane % 8;

  const int kc_per_plane = Cin / CONV_BK;
  const int NT = 9 * kc_per_plane;

  // stage tile t into buffer b (8 glds: 4 A + 4 B)
  auto stage = [&](int t, int b) {
    const int plane = t / kc_per_plane;
    const int kc = t % kc_per_plane;
    const int dy = plane / 3 - 1, dx = plane % 3 - 1;
    const long poff = ((long)dy * W + dx) * Cin + (long)kc * CONV_BK;
    __bf16 *abuf = smem + b * (TILE_ELEMS + BTILE);
    __bf16 *bbuf = abuf + TILE_ELEMS;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = i * 32 + wid * 8 + lane / 8;
      const int sc = cswz(row, schunk);
      bool av;
      const __hip_bfloat16 *asrc;
      if constexpr (UPS) {
        const int vh = hs[i] + dy, vw = ws[i] + dx;  // virtual 2H x 2W
        av = vh >= 0 && vh < 2 * H && vw >= 0 && vw < 2 * W;
        asrc = av ? (X +
                     (((long)nimg[i] * H + (vh >> 1)) * W + (vw >> 1)) *
                         Cin +
                     (long)kc * CONV_BK + sc * 8)
                  : Zero;
      } else {
        av = (hs[i] + dy >= 0) && (hs[i] + dy < H) &&
             (ws[i] + dx >= 0) && (ws[i] + dx < W);
        asrc = av ? (X + abase[i] + poff + sc * 8) : Zero;
      }
      (void)poff;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)asrc,
          (__attribute__((address_space(3))) unsigned int
               *)(abuf + i * 2048 + wid * 512),
          16, 0, 0);
      if (i < NB) {
        const bool bv = bco[i] < Cout;
        const __hip_bfloat16 *bsrc =
            bv ? (Wt + (long)bco[i] * 9 * Cin + plane * Cin +
                  kc * CONV_BK + sc * 8)
               : Zero;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)bsrc,
            (__attribute__((address_space(3))) unsigned int
                 *)(bbuf + i * 2048 + wid * 512),
            16, 0, 0);
      }
    }
  };

  f32x4c acc[4][NJ];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = (f32x4c){};

  stage(0, 0);
  if (NT > 1) stage(1, 1);

  const int l16 = lane % 16;
  const int kq = (lane / 16) * 8;  // k offset of this lane's 8 elements

  for (int t = 0; t < NT; ++t) {
    // current tile's DMAs landed; the next tile's stay in flight
    if (t + 1 < NT)
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(4 + NB) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const __bf16 *abuf = smem + (t & 1) * (TILE_ELEMS + BTILE);
    const __bf16 *bbuf = abuf + TILE_ELEMS;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      bf16x8 af[4], bf[NJ];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = wm + i * 16 + l16;
        const int ck = cswz(am, (s * 32 + kq) / 8);
        af[i] = *(const bf16x8 *)((const char *)(abuf + am * CONV_BK) +
                                  ck * 16);
      }
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        const int bn = wn + j * 16 + l16;
        const int ck = cswz(bn, (s * 32 + kq) / 8);
        bf[j] = *(const bf16x8 *)((const char *)(bbuf + bn * CONV_BK) +
                                  ck * 16);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();  // everyone done reading this buffer
    if (t + 2 < NT) stage(t + 2, t & 1);
  }

  // ---- epilogue: bias, channel-bias, residual, store --------------------
  const int r4 = (lane / 16) * 4;
  float gs[NJ], gq[NJ];
#pragma unroll
  for (int j = 0; j < NJ; ++j) gs[j] = gq[j] = 0.0f;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
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
          v += (float)CB[(long)ni * ldY + co];
        }
        if (HAS_RES) v += (float)Res[m * ldY + co];
        const __hip_bfloat16 vb = f2bf(v);
        Y[m * ldY + co] = vb;
        if (GNP) {
          const float vr = bf2f(vb);  // stats over the STORED values
          gs[j] += vr;
          gq[j] += vr * vr;
        }
      }
    }
  }
  if (GNP) {
    // wave reduce: fold the 4 lane-quarters holding one column
    float *lds = (float *)smem;  // staging LDS is free after the k-loop
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
#pragma unroll
      for (int off = 16; off < 64; off <<= 1) {
        gs[j] += __shfl_down(gs[j], off, WAVE);
        gq[j] += __shfl_down(gq[j], off, WAVE);
      }
      // lanes 0..15 hold the wave's column sums -> LDS [wid][j][l16][2]
      if (lane < 16) {
        float *slot = lds + ((wid * NJ + j) * 16 + lane) * 2;
        slot[0] = gs[j];
        slot[1] = gq[j];
      }
    }
    __syncthreads();
    // fixed-order cross-wave combine: col c's two waves are
    // {c/(BN/2), +2} (wave grid 2Mx2N); one thread per column writes
    const long mtile = (long)blockIdx.y + (long)blockIdx.z * 32768;
    for (int c = tid; c < BN; c += 256) {
      const int co = n0 + c;
      if (co >= Cout) continue;
      const int h = c / (BN / 2);
      const int jj = (c % (BN / 2)) / 16;
      const int lc = c % 16;
      const float *s0 = lds + ((h * NJ + jj) * 16 + lc) * 2;
      const float *s1 = lds + (((h + 2) * NJ + jj) * 16 + lc) * 2;
      float *dst = GNP + (mtile * 2) * (long)ldY;
      dst[co] = s0[0] + s1[0];
      dst[ldY + co] = s0[1] + s1[1];
    }
  }
}

// ---------------------------------------------------------------------------
#ifdef __HIP_PLATFORM_AMD__
bool conv3x3_supported(long cin) { return cin % 64 == 0 && cin >= 64; }

torch::Tensor conv3x3_nhwc_impl(torch::Tensor x, torch::Tensor w_prep,
                                c10::optional<torch::Tensor> bias,
                                c10::optional<torch::Tensor> residual,
                                c10::optional<torch::Tensor> chan_bias,
                                long stride, torch::Tensor *gnp_out) {
  // x: [N,C,H,W] channels_last; w_prep: [Cout,3,3,Cin] contiguous
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3: x must be channels_last");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_prep.size(0);
  TORCH_CHECK(conv3x3_supported(Cin), "conv3x3: Cin % 64 != 0");
  const int Ho = (H + 2 - 3) / stride + 1;
  const int Wo = (W + 2 - 3) / stride + 1;
  auto y = torch::empty({N, Cout, Ho, Wo},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  static torch::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = torch::zeros({64}, x.options());
  const long M = (long)N * Ho * Wo;
  const long mtiles = (M + CONV_BM - 1) / CONV_BM;
  dim3 grid((unsigned)((Cout + CONV_BN - 1) / CONV_BN),
            (unsigned)std::min<long>(mtiles, 32768),
            (unsigned)((mtiles + 32767) / 32768));
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
  // GroupNorm partial side-channel (one (sum,sumsq) pair per 128-row tile
  // per channel): only when every 128-row tile lies inside one image
  float *gnp_ptr = nullptr;
  if (gnp_out != nullptr && ((long)Ho * Wo) % 128 == 0) {
    *gnp_out = torch::empty(
        {M / 128, 2, (long)Cout},
        x.options().dtype(torch::kFloat));
    gnp_ptr = gnp_out->data_ptr<float>();
  }
  // v2 (2 blocks/CU, 2-buffer glds) measured faster than the deeper v3
  // pipeline (18.8 vs 21.1 ms on the shape set): at 2 blocks/CU the
  // block-level overlap already hides the DMA. v3 stays opt-in.
  static const bool use_v3 = getenv("SDWD_CONV_V3") != nullptr;
  if (use_v3) {
    // v3: deep-pipelined 256x128 tile (conv_v3.hip)
    const long mt3 = (M + 255) / 256;
    dim3 g3((unsigned)((Cout + 127) / 128),
            (unsigned)std::min<long>(mt3, 32768),
            (unsigned)((mt3 + 32767) / 32768));
    dim3 b3(512);
#define PICK3(B_, R_, C_) conv3x3_v3_kernel<B_, R_, C_>
    auto k3 = has_b ? (has_r ? (has_cb ? PICK3(true, true, true)
                                       : PICK3(true, true, false))
                             : (has_cb ? PICK3(true, false, true)
                                       : PICK3(true, false, false)))
                    : (has_r ? (has_cb ? PICK3(false, true, true)
                                       : PICK3(false, true, false))
                             : (has_cb ? PICK3(false, false, true)
                                       : PICK3(false, false, false)));
#undef PICK3
    hipLaunchKernelGGL(k3, g3, b3, 0, stream,
                       (const __hip_bfloat16 *)x.data_ptr(),
                       (const __hip_bfloat16 *)w_prep.data_ptr(), bptr, rptr,
                       cbptr,
                       (const __hip_bfloat16 *)zero_page.data_ptr(),
                       (__hip_bfloat16 *)y.data_ptr(), N, H, W, Cin, Cout,
                       Ho, Wo, (int)stride);
    return y;
  }
  // v4: the 256x256 8-phase deep pipeline serves full 256-column tiles;
  // the v2 128-tile kernel covers the Cout remainder (e.g. 320 = 256+64).
  // SDWD_CONV=v2 forces the round-1 kernel everywhere; =v4 forces the
  // deep pipeline on every legal shape (A/B tooling).
  static const int conv_force = [] {
    const char *e = getenv("SDWD_CONV");
    if (!e) return 0;
    if (strcmp(e, "v2") == 0) return 2;
    if (strcmp(e, "v4") == 0) return 4;
    return 0;
  }();
  const bool v2only = conv_force == 2;
  const long mt4 = (M + V4_BM - 1) / V4_BM;
  // measured winners (profiles/r02_conv_family.md): the 256x256 pipeline
  // wins on the SD-UNet level-0/1 shapes (Cin<=960, Cout<=640, big M) and
  // — since the phase-0 stage-8 schedule — the VAE 512-ch shapes too
  // (+7-12%); the 1280-channel shapes stay on v2 (2 blocks/CU beats the
  // deep pipeline there: 737-765 vs 814-822 TF/s forced).
  // ... and only when the 512-thread 1-block/CU grid actually fills the
  // 256 CUs (small shards at N=8 under-fill 256-row tiles: the 128-row
  // v2 kernel at 2 blocks/CU wins there)
  const bool v4_shape =
      conv_force == 4
          ? (Cin % 64 == 0 && Cout >= 256)
          : (Cin <= 960 && Cout <= 640 && Cout >= 256 &&
             ((M + V4_BM - 1) / V4_BM) * (Cout / 256) >= 256);
  const long nfull = (v2only || !v4_shape) ? 0 : Cout / 256;
  const int rem = (int)(Cout - nfull * 256);
  if (nfull > 0) {
    dim3 g4((unsigned)nfull, (unsigned)std::min<long>(mt4, 32768),
            (unsigned)((mt4 + 32767) / 32768));
    dim3 b4(512);
#define PICK4(B_, R_, C_) conv3x3_v4_kernel<B_, R_, C_>
    auto k4 = has_b ? (has_r ? (has_cb ? PICK4(true, true, true)
                                       : PICK4(true, true, false))
                             : (has_cb ? PICK4(true, false, true)
                                       : PICK4(true, false, false)))
                    : (has_r ? (has_cb ? PICK4(false, true, true)
                                       : PICK4(false, true, false))
                             : (has_cb ? PICK4(false, false, true)
                                       : PICK4(false, false, false)));
#undef PICK4
    hipLaunchKernelGGL(k4, g4, b4, 0, stream,
                       (const __hip_bfloat16 *)x.data_ptr(),
                       (const __hip_bfloat16 *)w_prep.data_ptr(), bptr,
                       rptr, cbptr,
                       (const __hip_bfloat16 *)zero_page.data_ptr(),
                       (__hip_bfloat16 *)y.data_ptr(), N, H, W, Cin,
                       (int)(nfull * 256), Cout, Ho, Wo, (int)stride,
                       gnp_ptr);
  }
  if (rem > 0) {
    const long co0 = nfull * 256;
    // BN=64 when the remainder is one 64-tile OR when the BN=128 grid
    // would under-fill the 256 CUs (small shards: twice the blocks beat
    // the halved B-reuse when occupancy-starved)
    const bool bn64 =
        rem <= 64 || mtiles * ((rem + 127) / 128) < 512;
#define PICK(B_, R_, C_)                                              \
  (bn64 ? conv3x3_nhwc_bf16_kernel<B_, R_, C_, 64>                    \
        : conv3x3_nhwc_bf16_kernel<B_, R_, C_, 128>)
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
    dim3 g2((unsigned)((rem + (bn64 ? 63 : 127)) / (bn64 ? 64 : 128)),
            grid.y, grid.z);
    hipLaunchKernelGGL(
        kern, g2, block, 0, stream, (const __hip_bfloat16 *)x.data_ptr(),
        (const __hip_bfloat16 *)w_prep.data_ptr() + co0 * 9 * Cin,
        bptr ? bptr + co0 : nullptr,
        rptr ? rptr + co0 : nullptr, cbptr ? cbptr + co0 : nullptr,
        (const __hip_bfloat16 *)zero_page.data_ptr(),
        (__hip_bfloat16 *)y.data_ptr() + co0, N, H, W, Cin, rem, Ho, Wo,
        (int)stride, Cout, gnp_ptr ? gnp_ptr + co0 : nullptr);
  }
  return y;
}

torch::Tensor conv3x3_nhwc(torch::Tensor x, torch::Tensor w_prep,
                           c10::optional<torch::Tensor> bias,
                           c10::optional<torch::Tensor> residual,
                           c10::optional<torch::Tensor> chan_bias,
                           long stride) {
  return conv3x3_nhwc_impl(x, w_prep, bias, residual, chan_bias, stride,
                           nullptr);
}

std::vector<torch::Tensor> conv3x3_nhwc_gn(
    torch::Tensor x, torch::Tensor w_prep,
    c10::optional<torch::Tensor> bias,
    c10::optional<torch::Tensor> residual,
    c10::optional<torch::Tensor> chan_bias, long stride) {
  torch::Tensor gnp;
  auto y = conv3x3_nhwc_impl(x, w_prep, bias, residual, chan_bias, stride,
                             &gnp);
  if (!gnp.defined()) gnp = torch::empty({0});
  return {y, gnp};
}

torch::Tensor ups2x_conv3x3_impl(torch::Tensor x, torch::Tensor w_prep,
                                 c10::optional<torch::Tensor> bias,
                                 torch::Tensor *gnp_out) {
  // nearest-2x upsample fused into the 3x3 conv (VAE decoder / UNet
  // Upsample blocks): output is [N, Cout, 2H, 2W]
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_prep.size(0);
  TORCH_CHECK(conv3x3_supported(Cin));
  const int Ho = 2 * H, Wo = 2 * W;
  auto y = torch::empty({N, Cout, Ho, Wo},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  static torch::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = torch::zeros({64}, x.options());
  const long M = (long)N * Ho * Wo;
  const long mtiles = (M + CONV_BM - 1) / CONV_BM;
  dim3 grid((unsigned)((Cout + CONV_BN - 1) / CONV_BN),
            (unsigned)std::min<long>(mtiles, 32768),
            (unsigned)((mtiles + 32767) / 32768));
  dim3 block(256);
  const bool has_b = bias.has_value();
  torch::Tensor bf32;
  const float *bptr = nullptr;
  if (has_b) {
    bf32 = bias->to(torch::kFloat).contiguous();
    bptr = bf32.data_ptr<float>();
  }
  float *gnp_ptr = nullptr;
  if (gnp_out != nullptr && ((long)Ho * Wo) % 128 == 0) {
    *gnp_out = torch::empty(
        {M / 128, 2, (long)Cout}, x.options().dtype(torch::kFloat));
    gnp_ptr = gnp_out->data_ptr<float>();
  }
  auto kern = has_b
                  ? conv3x3_nhwc_bf16_kernel<true, false, false, 128, true>
                  : conv3x3_nhwc_bf16_kernel<false, false, false, 128,
                                             true>;
  hipLaunchKernelGGL(kern, grid, block, 0, cur_stream(),
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (const __hip_bfloat16 *)w_prep.data_ptr(), bptr,
                     nullptr, nullptr,
                     (const __hip_bfloat16 *)zero_page.data_ptr(),
                     (__hip_bfloat16 *)y.data_ptr(), N, H, W, Cin, Cout,
                     Ho, Wo, 1, Cout, gnp_ptr);
  return y;
}

torch::Tensor ups2x_conv3x3(torch::Tensor x, torch::Tensor w_prep,
                            c10::optional<torch::Tensor> bias) {
  return ups2x_conv3x3_impl(x, w_prep, bias, nullptr);
}

std::vector<torch::Tensor> ups2x_conv3x3_gn(
    torch::Tensor x, torch::Tensor w_prep,
    c10::optional<torch::Tensor> bias) {
  torch::Tensor gnp;
  auto y = ups2x_conv3x3_impl(x, w_prep, bias, &gnp);
  if (!gnp.defined()) gnp = torch::empty({0});
  return {y, gnp};
}
#endif
