// Implicit-GEMM 3x3 conv, v4: the guide's 256x256 8-phase deep-pipeline
// structure (cdna_hip_programming.md §5 '256² 8-phase template', T2+T3+T4+T5)
// adapted to the im2col A operand.
//
//  * 512 threads (8 waves as 2M x 4N), BM=BN=256, BK=64; per-wave output
//    128x64 = 8x4 fragments of mfma_f32_16x16x32_bf16, acc = 128 VGPRs.
//  * TWO whole-tile LDS buffers (A 32K + B 32K each, 128 KiB total, one
//    block/CU): while tile t streams out of buf[t&1], tile t+1 is staged
//    into buf[~t&1] by 2 global_load_lds per thread per phase (4 phases
//    per K-tile, one C-quadrant each = 16 MFMAs).
//  * counted `s_waitcnt vmcnt(2)` ONCE per K-tile (phase 3), raw
//    s_barrier everywhere (a __syncthreads would drain the glds queue);
//    hipcc's own lgkm counting orders the ds_read->MFMA chain.
//  * LDS rows are 128 B; the 8 16-B chunks XOR-swizzled by (row>>1)&7 on
//    the glds SOURCE address (rule 21) - b128 fragment reads conflict-free.
//  * bias / per-sample channel bias / residual fused in the epilogue.
//
// Covers full 256-column tiles only; the host launches the v2 kernel on
// the Cout remainder (Cout=320 -> 256 here + 64 there).
#include "common.h"

// XOR-permute the 8 16-byte chunks of a 128-byte LDS row (shared with the
// v2 kernel in conv.hip): b128 fragment reads stay bank-conflict-free.
__device__ __forceinline__ int cswz(int row, int chunk) {
  return chunk ^ ((row >> 1) & 7);
}

#define V4_BM 256
#define V4_BN 256
#define V4_BK 64
#define V4_ATILE (V4_BM * V4_BK)  // elements per A buffer

template <bool HAS_BIAS, bool HAS_RES, bool HAS_CB>
__launch_bounds__(512, 2) __global__ void conv3x3_v4_kernel(
    const __hip_bfloat16 *__restrict__ X,   // [N,H,W,Cin]
    const __hip_bfloat16 *__restrict__ Wt,  // [Cout,3,3,Cin] (+col offset)
    const float *__restrict__ bias,         // [cols] or null (+col offset)
    const __hip_bfloat16 *__restrict__ Res, // [M,ldY] or null (+col offset)
    const __hip_bfloat16 *__restrict__ CB,  // [N,ldY] or null (+col offset)
    const __hip_bfloat16 *__restrict__ Zero,
    __hip_bfloat16 *__restrict__ Y,         // [M,ldY] (+col offset)
    int Nn, int H, int W, int Cin, int Cols, int ldY, int Ho, int Wo,
    int stride, float *__restrict__ GNP = nullptr) {
  // GNP: per-(M-tile, channel) partial (sum, sumsq) of the stored values
  // (see conv.hip - lets GroupNorm skip its full-tensor stats read).
  __shared__ __align__(16) __bf16 smem[4 * V4_ATILE];  // [buf][A|B]

  const long M = (long)Nn * Ho * Wo;
  const long m0 =
      ((long)blockIdx.y + (long)blockIdx.z * 32768) * V4_BM;
  const int n0 = blockIdx.x * V4_BN;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wm = (wid >> 2) * 128;       // wave row base (2 rows of waves)
  const int wn = (wid & 3) * 64;         // wave col base (4 cols of waves)

  // staged-piece bookkeeping: piece p = tid + 512*i covers LDS bytes
  // p*16 of the A (i<4) or B (i-4) tile; row = p/8, chunk = p%8.
  // A rows are t/8 + 64*i; B rows likewise.
  const int prow = tid / 8;
  const int schunk = tid % 8;
  long abase[4];
  int hs[4], ws[4];
  int bco[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const long m = m0 + prow + 64 * i;
    const long mm = (m < M) ? m : (M - 1);
    const int n_img = (int)(mm / ((long)Ho * Wo));
    const int rem = (int)(mm % ((long)Ho * Wo));
    hs[i] = (rem / Wo) * stride;
    ws[i] = (rem % Wo) * stride;
    abase[i] = (((long)n_img * H + hs[i]) * W + ws[i]) * Cin;
    bco[i] = n0 + prow + 64 * i;
  }

  const int kc_per_plane = Cin / V4_BK;
  const int NT = 9 * kc_per_plane;

  // stage ALL 8 pieces of tile t into buffer b at once. Issued at phase 0
  // of the previous tile (that buffer's readers finished a barrier-pair
  // ago), drained by ONE `s_waitcnt vmcnt(0)` at the END of phase 3 —
  // every piece gets ~3 full phases (>1000 cycles) of slack. The earlier
  // schedule spread the issues across the 4 phases with counted waits at
  // ph1/ph3, which left the ph2/ph3-issued pieces only ~1 phase of slack:
  // the waits stalled on just-issued loads (PMC: 34-42% wave-wait).
  // B pieces first (read by every wave at the next tile's phase 0).
  auto stage8 = [&](int t, int b) {
    const int plane = t / kc_per_plane;
    const int kc = t % kc_per_plane;
    const int dy = plane / 3 - 1, dx = plane % 3 - 1;
    const long poff = ((long)dy * W + dx) * Cin + (long)kc * V4_BK;
    __bf16 *abuf = smem + b * (2 * V4_ATILE);
    __bf16 *bbuf = abuf + V4_ATILE;
    const int sc = schunk;
    constexpr int PIECES[8] = {4, 5, 6, 7, 0, 1, 2, 3};
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const int i = PIECES[k];            // piece index 0..7
      if (i < 4) {
        const int row = prow + 64 * i;
        const int swz = cswz(row, sc);
        const bool av = (hs[i] + dy >= 0) && (hs[i] + dy < H) &&
                        (ws[i] + dx >= 0) && (ws[i] + dx < W);
        const __hip_bfloat16 *asrc =
            av ? (X + abase[i] + poff + swz * 8) : Zero;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)asrc,
            (__attribute__((address_space(3))) unsigned int
                 *)(abuf + (long)i * 4096 + (long)tid * 8),
            16, 0, 0);
      } else {
        const int bi = i - 4;
        const int row = prow + 64 * bi;
        const int swz = cswz(row, sc);
        const bool bv = bco[bi] < Cols;
        const __hip_bfloat16 *bsrc =
            bv ? (Wt + (long)bco[bi] * 9 * Cin + plane * Cin +
                  kc * V4_BK + swz * 8)
               : Zero;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)bsrc,
            (__attribute__((address_space(3))) unsigned int
                 *)(bbuf + (long)bi * 4096 + (long)tid * 8),
            16, 0, 0);
      }
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){};

  // prologue: stage tile 0 fully, then begin the loop with tile 1
  // staged at tile 0's phase 0 while tile 0 is computed
  stage8(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int l16 = lane % 16;
  const int kq4 = lane / 16;  // k quarter (8 elems each) of this lane

  for (int t = 0; t < NT; ++t) {
    const __bf16 *abuf = smem + (t & 1) * (2 * V4_ATILE);
    const __bf16 *bbuf = abuf + V4_ATILE;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {  // quadrant (mh, nh)
      const int mh = ph >> 1, nh = ph & 1;
      // register fragments for this quadrant (hipcc counts these reads)
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int s = 0; s < 2; ++s) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int ar = wm + mh * 64 + i * 16 + l16;
          const int ck = cswz(ar, s * 4 + kq4);
          af[i][s] = *(const bf16x8 *)((const char *)(abuf +
                                                      (long)ar * V4_BK) +
                                       ck * 16);
        }
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const int br = wn + nh * 32 + j * 16 + l16;
          const int ck = cswz(br, s * 4 + kq4);
          bf[j][s] = *(const bf16x8 *)((const char *)(bbuf +
                                                      (long)br * V4_BK) +
                                       ck * 16);
        }
      }
      // stage ALL of tile t+1 into the other buffer at phase 0 (its
      // readers in tile t-1 retired before this tile's first barrier)
      if (ph == 0 && t + 1 < NT) stage8(t + 1, (t + 1) & 1);
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
      // ONE drain per K-tile, ~3 phases after the issues; the final
      // barrier below then publishes the staged LDS to every wave
      // before the next tile's first ds_read
      if (ph == 3 && t + 1 < NT)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue
  const int r4 = (lane / 16) * 4;
  float gs[4], gq[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) gs[j] = gq[j] = 0.0f;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int co = n0 + wn + (j >> 1) * 32 + (j & 1) * 16 + l16;
      if (co >= Cols) continue;
      const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m0 + wm + (i >> 2) * 64 + (i & 3) * 16 + r4 + r;
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
          const float vr = bf2f(vb);
          gs[j] += vr;
          gq[j] += vr * vr;
        }
      }
    }
  }
  if (GNP) {
    float *lds = (float *)smem;  // free after the k-loop
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int off = 16; off < 64; off <<= 1) {
        gs[j] += __shfl_down(gs[j], off, WAVE);
        gq[j] += __shfl_down(gq[j], off, WAVE);
      }
      if (lane < 16) {
        float *slot = lds + ((wid * 4 + j) * 16 + lane) * 2;
        slot[0] = gs[j];
        slot[1] = gq[j];
      }
    }
    __syncthreads();
    // GNP granularity is 128 rows (the v2 kernel's BM, so the remainder
    // launch shares the layout): this 256-row tile's two wave M-halves
    // write two consecutive 128-row partial slots
    const long mt128 = m0 / 128;
    for (int c = tid; c < V4_BN; c += 512) {
      const int co = n0 + c;
      if (co >= Cols) continue;
      const int q = c / 64;                      // n-quarter wave
      const int o = c % 64;
      const int jj = (o / 32) * 2 + (o % 32) / 16;
      const int lc = c % 16;
      const float *s0 = lds + ((q * 4 + jj) * 16 + lc) * 2;       // rows 0-127
      const float *s1 = lds + (((q + 4) * 4 + jj) * 16 + lc) * 2; // 128-255
      float *d0 = GNP + (mt128 * 2) * (long)ldY;
      d0[co] = s0[0];
      d0[ldY + co] = s0[1];
      if ((mt128 + 1) * 128 < M) {  // second 128-row half has real rows
        float *d1 = GNP + ((mt128 + 1) * 2) * (long)ldY;
        d1[co] = s1[0];
        d1[ldY + co] = s1[1];
      }
    }
  }
}
