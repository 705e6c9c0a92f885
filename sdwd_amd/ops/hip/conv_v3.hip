// conv v3: deep-pipelined implicit-GEMM 3x3 (guide §5 8-phase-class
// schedule adapted to the conv's shapes).
//
//  * 512 threads (8 waves, 4M x 2N), tile BM=256 x BN=128, BK=64;
//    per-wave output 64x64 as 4x4 mfma_f32_16x16x32_bf16 (x2 k-subs).
//  * THREE LDS buffers (A 32K + B 16K each, 144 KiB total, 1 block/CU):
//    tile T+2 is staged while T is computed and T+1 is in flight, so the
//    tile-boundary wait is a counted `s_waitcnt vmcnt(6)` that never
//    drains the pipeline.
//  * 2 phases per K-tile: {issue 3 of the 6 stage-DMAs for T+2, ds_read
//    the phase's A fragments (B fragments persist in registers from
//    phase 0), raw barrier, setprio(1), 16 MFMA, setprio(0), raw
//    barrier}. Phase split gives the CU scheduler load/compute wave
//    diversity (T5's prerequisite).
//  * same source-side chunk swizzle, zero-page padding and fused
//    bias/chan-bias/residual epilogue as v2 (conv.hip).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4v3;

#define V3_BM 256
#define V3_BN 128
#define V3_BK 64
#define V3_ATILE (V3_BM * V3_BK)              // 16384 elems = 32 KiB
#define V3_BTILE (V3_BN * V3_BK)              // 8192 elems = 16 KiB
#define V3_BUF (V3_ATILE + V3_BTILE)          // per-buffer elems

__device__ __forceinline__ int v3swz(int row, int chunk) {
  return chunk ^ ((row >> 1) & 7);
}

template <bool HAS_BIAS, bool HAS_RES, bool HAS_CB>
__launch_bounds__(512, 2) __global__ void conv3x3_v3_kernel(
    const __hip_bfloat16 *__restrict__ X, const __hip_bfloat16 *__restrict__ Wt,
    const float *__restrict__ bias, const __hip_bfloat16 *__restrict__ Res,
    const __hip_bfloat16 *__restrict__ CB,
    const __hip_bfloat16 *__restrict__ Zero, __hip_bfloat16 *__restrict__ Y,
    int Nn, int H, int W, int Cin, int Cout, int Ho, int Wo, int stride) {
  __shared__ __align__(16) __bf16 smem[3 * V3_BUF];

  const long M = (long)Nn * Ho * Wo;
  const long m0 = ((long)blockIdx.y + (long)blockIdx.z * 32768) * V3_BM;
  if (m0 >= M) return;
  const int n0 = blockIdx.x * V3_BN;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;          // 0..7
  const int wm = (wid >> 1) * 64;      // 4 M-slots
  const int wn = (wid & 1) * 64;       // 2 N-slots

  // stage bookkeeping: A instr a in 0..3 -> row a*64 + wid*8 + lane/8;
  // B instr b in 0..1 -> row b*64 + wid*8 + lane/8
  long abase[4];
  int hs[4], ws[4];
  int bco[2];
#pragma unroll
  for (int a = 0; a < 4; ++a) {
    const int row = a * 64 + wid * 8 + lane / 8;
    const long m = m0 + row;
    const long mm = (m < M) ? m : (M - 1);
    const int n_img = (int)(mm / ((long)Ho * Wo));
    const int rem = (int)(mm % ((long)Ho * Wo));
    hs[a] = (rem / Wo) * stride;
    ws[a] = (rem % Wo) * stride;
    abase[a] = (((long)n_img * H + hs[a]) * W + ws[a]) * Cin;
  }
#pragma unroll
  for (int b = 0; b < 2; ++b) bco[b] = n0 + b * 64 + wid * 8 + lane / 8;
  const int schunk = lane % 8;

  const int kc_per_plane = Cin / V3_BK;
  const int NT = 9 * kc_per_plane;

  // one A-piece (a in 0..3) or B-piece (4+b) of tile t
  auto stage_piece = [&](int t, int piece) {
    const int plane = t / kc_per_plane;
    const int kc = t % kc_per_plane;
    const int dy = plane / 3 - 1, dx = plane % 3 - 1;
    __bf16 *buf = smem + (t % 3) * V3_BUF;
    if (piece < 4) {
      const int a = piece;
      const int row = a * 64 + wid * 8 + lane / 8;
      const int sc = v3swz(row, schunk);
      const bool av = (hs[a] + dy >= 0) && (hs[a] + dy < H) &&
                      (ws[a] + dx >= 0) && (ws[a] + dx < W);
      const __hip_bfloat16 *src =
          av ? (X + abase[a] + ((long)dy * W + dx) * Cin +
                (long)kc * V3_BK + sc * 8)
             : Zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)src,
          (__attribute__((address_space(3))) unsigned int
               *)(buf + a * 4096 + wid * 512),
          16, 0, 0);
    } else {
      const int b = piece - 4;
      const int co = bco[b];
      const int row = b * 64 + wid * 8 + lane / 8;
      const int sc = v3swz(row, schunk);
      const __hip_bfloat16 *src =
          (co < Cout)
              ? (Wt + (long)co * 9 * Cin + plane * Cin + kc * V3_BK + sc * 8)
              : Zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)src,
          (__attribute__((address_space(3))) unsigned int
               *)(buf + V3_ATILE + b * 4096 + wid * 512),
          16, 0, 0);
    }
  };
  auto stage_tile = [&](int t) {
#pragma unroll
    for (int p = 0; p < 6; ++p) stage_piece(t, p);
  };

  f32x4v3 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4v3){};

  stage_tile(0);
  if (NT > 1) stage_tile(1);

  const int l16 = lane % 16;
  const int kq = (lane / 16) * 8;

  for (int t = 0; t < NT; ++t) {
    // T landed; T+1's 6 DMAs may stay in flight
    if (t + 1 < NT)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const __bf16 *abuf = smem + (t % 3) * V3_BUF;
    const __bf16 *bbuf = abuf + V3_ATILE;
    const bool can_stage = (t + 2 < NT);

    // B fragments for the whole tile (persist across both phases)
    bf16x8 bfr[2][4];
#pragma unroll
    for (int s = 0; s < 2; ++s)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int bn = wn + j * 16 + l16;
        const int ck = v3swz(bn, (s * 32 + kq) / 8);
        bfr[s][j] = *(const bf16x8 *)((const char *)(bbuf + bn * V3_BK) +
                                      ck * 16);
      }

#pragma unroll
    for (int ph = 0; ph < 2; ++ph) {
      // phase ph: A fragment rows 2*ph, 2*ph+1; stage 3 pieces of T+2
      bf16x8 afr[2][2];
#pragma unroll
      for (int s = 0; s < 2; ++s)
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          const int am = wm + (ph * 2 + i) * 16 + l16;
          const int ck = v3swz(am, (s * 32 + kq) / 8);
          afr[s][i] = *(const bf16x8 *)((const char *)(abuf + am * V3_BK) +
                                        ck * 16);
        }
      if (can_stage) {
#pragma unroll
        for (int p = 0; p < 3; ++p) stage_piece(t + 2, ph * 3 + p);
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 2; ++s)
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[ph * 2 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[s][i], bfr[s][j], acc[ph * 2 + i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

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
