// Flash-style fused attention forward for gfx950 (MFMA 32x32x16 bf16).
//
// Structure (v2):
//  * workgroup = 4 waves; each wave owns 32 q-rows, the WG shares K/V tiles
//    of 32 keys staged in LDS (K row-major padded, V transposed for
//    contiguous B-fragment reads).
//  * swapped QK^T: S^T = mfma(K_frag, Q_frag) so each lane holds the scores
//    of ONE q-row (lane%32) across 16 of 32 keys — softmax is register-local
//    plus one __shfl_xor(32) to combine the half-wave pair.
//  * P (bf16-packed) is redistributed to the PV A-fragment layout with
//    v_permlane32_swap pairs (guide T12), then PV accumulates fp32 via MFMA.
//  * online softmax with per-tile rescale; the previous tile's PV completes
//    before the rescale decision (textbook order).
//  * STRIDED access: Q/K/V/O are addressed with (batch, head, row) strides,
//    so both [B,H,S,D] and [B,S,H,D] layouts run with NO transpose or pad
//    copies; the head dim D only needs D % 8 == 0 (8-element groups beyond
//    D are masked in-kernel, DPAD = next multiple of 16).
//
// UNet shapes: D 40(->48), 80, 160; SDXL 64; VAE (D=512) takes the composed
// fallback path in ext.hip.
#include "common.h"

__device__ __forceinline__ unsigned pack_bf16(float lo, float hi) {
  union {
    __hip_bfloat162 h2;
    unsigned u;
  } cvt;
  cvt.h2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
  return cvt.u;
}

struct AttnStrides {
  long qb, qh, qr;  // batch, head, row strides (elements)
  long kb, kh, kr;
  long vb, vh, vr;
  long ob, oh, or_;
};

template <int DPAD>
__launch_bounds__(256, 2) __global__ void flash_fwd_bf16_kernel(
    const __hip_bfloat16 *__restrict__ Q, const __hip_bfloat16 *__restrict__ K,
    const __hip_bfloat16 *__restrict__ V, __hip_bfloat16 *__restrict__ O,
    int H, long Sq, long Sk, int D, float scale, AttnStrides st) {
  constexpr int KVB = 64;  // two 32-key S^T sub-tiles per staging phase
  constexpr int PADK = 8;   // bf16 per-row pad: breaks ds_read_b128 conflicts
  constexpr int NC = DPAD / 16;              // QK^T k-chunks
  constexpr int DV = (DPAD + 31) / 32 * 32;  // PV d extent (32-col O tiles)
  constexpr int ND = DV / 32;                // PV d-chunks (O accum tiles)

  __shared__ __align__(16) __bf16 kt[KVB][DPAD + PADK];
  __shared__ __align__(16) __bf16 vt[DV][KVB + PADK];

  // (an XCD-grouping remap — one head's q-blocks pinned to one XCD for
  // K/V L2 reuse — measured -13% on the S=4096 shape: the concentrated
  // staging traffic beats the reuse. Plain mapping kept.)
  const int bh = blockIdx.y;
  const int qblk = blockIdx.x;
  const int bb = bh / H, hh = bh % H;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lq = lane % 32;   // q-row (softmax) / d-col (PV C) index
  const int half = lane / 32; // half-wave id
  const long q0 = (long)qblk * 256 + wid * 64;

  const __hip_bfloat16 *Qb = Q + bb * st.qb + hh * st.qh;
  const __hip_bfloat16 *Kb = K + bb * st.kb + hh * st.kh;
  const __hip_bfloat16 *Vb = V + bb * st.vb + hh * st.vh;
  __hip_bfloat16 *Ob = O + bb * st.ob + hh * st.oh;

  // Q fragments for the whole row-block, read once:
  // B-frag of mfma(K,Q): lane holds Q[q=lq][d = c*16 + 8*half + i]
  bf16x8 qf[2][NC];
#pragma unroll
  for (int qs = 0; qs < 2; ++qs) {
    const long qq = q0 + qs * 32 + lq;
    const long qrow = (qq < Sq) ? qq : (Sq - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int d0 = c * 16 + 8 * half;
      qf[qs][c] = (d0 + 8 <= D)
                      ? *(const bf16x8 *)(Qb + qrow * st.qr + d0)
                      : (bf16x8){};
    }
  }

  f32x16 o[2][ND];
#pragma unroll
  for (int qs = 0; qs < 2; ++qs)
#pragma unroll
    for (int d = 0; d < ND; ++d) o[qs][d] = (f32x16){};
  float m[2] = {-1e30f, -1e30f}, l[2] = {0.f, 0.f};

  // zero vt's pad rows once (D..DV); they are never re-staged
  for (int idx = threadIdx.x + (D / 8) * 8 * (KVB + PADK);
       idx < DV * (KVB + PADK); idx += 256)
    ((__bf16 *)vt)[idx] = (__bf16)0.0f;

  for (long kv = 0; kv < Sk; kv += KVB) {
    __syncthreads();  // previous tile's LDS reads complete
    // cooperative K/V stage: 256 threads, 8 bf16 each per step
    for (int idx = threadIdx.x; idx < KVB * (DPAD / 8); idx += 256) {
      const int r = idx / (DPAD / 8);     // key row in tile
      const int c8 = idx % (DPAD / 8);    // 8-elem column group
      bf16x8 kvec = (bf16x8){};
      bf16x8 vvec = (bf16x8){};
      if (kv + r < Sk && c8 * 8 + 8 <= D) {
        kvec = *(const bf16x8 *)(Kb + (kv + r) * st.kr + c8 * 8);
        vvec = *(const bf16x8 *)(Vb + (kv + r) * st.vr + c8 * 8);
      }
      *(bf16x8 *)&kt[r][c8 * 8] = kvec;
#pragma unroll
      for (int j = 0; j < 8; ++j) vt[c8 * 8 + j][r] = vvec[j];
    }
    __syncthreads();

    // per q-subtile: two S^T[32k, 32q] k-sub-tiles, merged softmax, PV
#pragma unroll
    for (int qs = 0; qs < 2; ++qs) {
      f32x16 stile[2] = {(f32x16){}, (f32x16){}};
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int c = 0; c < NC; ++c) {
          bf16x8 kf =
              *(const bf16x8 *)&kt[sub * 32 + lq][c * 16 + 8 * half];
          stile[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              kf, qf[qs][c], stile[sub], 0, 0, 0);
        }

      // softmax in base-2: scores carry scale*log2(e) once, so the hot
      // exponentials are bare v_exp2 with no per-element multiply
      const float l2scale = scale * 1.4426950408889634f;
      float p[2][16];
      float pmax = -1e30f;
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kk = sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float sv = (kv + kk < Sk) ? stile[sub][r] * l2scale : -1e30f;
          p[sub][r] = sv;
          pmax = fmaxf(pmax, sv);
        }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, WAVE));
      const float mnew = fmaxf(m[qs], pmax);
      const float alpha = __builtin_amdgcn_exp2f(m[qs] - mnew);
      float rowsum = 0.f;
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          p[sub][r] = __builtin_amdgcn_exp2f(p[sub][r] - mnew);
          rowsum += p[sub][r];
        }
      rowsum += __shfl_xor(rowsum, 32, WAVE);
      l[qs] = l[qs] * alpha + rowsum;
      m[qs] = mnew;

#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
        const float a = __shfl(alpha, qrow, WAVE);
#pragma unroll
        for (int d = 0; d < ND; ++d) o[qs][d][r] *= a;
      }

#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        unsigned pk[8];
#pragma unroll
        for (int t = 0; t < 8; ++t)
          pk[t] = pack_bf16(p[sub][2 * t], p[sub][2 * t + 1]);
        bf16x8 pa0, pa1;
        {
          auto r0 =
              __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
          auto r1 =
              __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
          unsigned frag[4] = {r0[0], r1[0], r0[1], r1[1]};
          pa0 = *(bf16x8 *)frag;
          auto r2 =
              __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
          auto r3 =
              __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
          unsigned frag1[4] = {r2[0], r3[0], r2[1], r3[1]};
          pa1 = *(bf16x8 *)frag1;
        }
#pragma unroll
        for (int d = 0; d < ND; ++d) {
          bf16x8 v0 =
              *(const bf16x8 *)&vt[d * 32 + lq][sub * 32 + 8 * half];
          o[qs][d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa0, v0, o[qs][d], 0, 0, 0);
          bf16x8 v1 =
              *(const bf16x8 *)&vt[d * 32 + lq][sub * 32 + 16 + 8 * half];
          o[qs][d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa1, v1, o[qs][d], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: O /= l, store (column-per-lane scatter; widen later, T21)
#pragma unroll
  for (int qs = 0; qs < 2; ++qs) {
    const float linv = 1.0f / fmaxf(l[qs], 1e-30f);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
      const float inv = __shfl(linv, qrow, WAVE);
      const long qq = q0 + qs * 32 + qrow;
      if (qq >= Sq) continue;
#pragma unroll
      for (int d = 0; d < ND; ++d)
        if (d * 32 + lq < D)
          Ob[qq * st.or_ + d * 32 + lq] = f2bf(o[qs][d][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// v3: same wave/tile geometry as v2 (4 waves x 64 q-rows, KVB=64 staged in
// LDS, swapped QK^T, in-register softmax, permlane P->A relayout) with the
// guide's attention-ladder levers applied:
//  * async-STAGE split (T14): next tile's K/V issued to registers BEFORE the
//    compute phase; only the LDS-write latency sits at the tile seam.
//  * scale folded into Q in log2 units: the hot loop's scores are already
//    scale*log2(e)*(q.k) so softmax is bare v_exp2 with no per-score mul.
//  * full-tile fast path: the (kv+kk<Sk) mask VALU only runs on the ragged
//    last tile; staged loads clamp rows/cols instead of branching per lane.
//  * defer-max (T13): the O/l rescale (16 shfl + 32 mul per q-subtile) is
//    skipped while the running max grows < 2^THR; P is bounded by 2^THR
//    which fp32 accumulation absorbs (THR=8 -> P<=256).
//  * s_setprio(1) brackets around the MFMA clusters (T5).
// ---------------------------------------------------------------------------
template <int DPAD, int QS = (DPAD <= 96 ? 2 : 1)>
__launch_bounds__(256, 2) __global__ void flash_fwd_bf16_v3(
    const __hip_bfloat16 *__restrict__ Q, const __hip_bfloat16 *__restrict__ K,
    const __hip_bfloat16 *__restrict__ V, __hip_bfloat16 *__restrict__ O,
    int H, long Sq, long Sk, int D, float scale, AttnStrides st) {
  constexpr int KVB = 64;
  constexpr int PADK = 8;
  constexpr int NC = DPAD / 16;
  constexpr int DV = (DPAD + 31) / 32 * 32;
  constexpr int ND = DV / 32;
  // tr-read conflict pad: the residual ~6% SQ_LDS_BANK_CONFLICT is
  // tr-instruction-intrinsic (a +16 stride variant measured identical;
  // guide T10: addr swizzles don't move tr_read conflict classes)
  constexpr int PADV = (DV == 128) ? 24 : 8;
  constexpr int RSV = DV + PADV;              // vt2 row stride (elements)
  constexpr int NG = KVB * (DPAD / 8);        // bf16x8 pieces per operand
  constexpr int NST = 2 * NG / 256;           // pieces/thread (exact: 16*DPAD%256==0)
  constexpr float THR = 8.0f;                 // defer-max threshold (log2)
  // async register staging costs NST*8 VGPRs; at DPAD>=80 that spills,
  // so big head dims stage synchronously instead
  constexpr bool ASYNC = (DPAD <= 64);

  // BOTH K and V are staged row-major with pure vectorized writes: the PV
  // B-fragments come from ds_read_b64_tr_b16 hardware transpose reads
  // (mapping decoded by tools/tr16_probe.py: with addr(l) =
  // &vt2[k0 + ((l&15)>>2)][d0 + 4*(l&3)], lane l receives V[k0+j][d0+(l&15)]
  // for j=0..3). This removes the 8-scalar-LDS-write V transpose AND makes
  // the staging exactly NST pieces for EVERY thread (wave-balanced barriers;
  // the old 1.5-piece split parked half the waves at each barrier - the
  // round-1 PMC showed 52.6% SQ_WAIT_ANY at D=48).
  // double-buffered tiles: tile i+1 is written into buf i^1 while buf i
  // is computed, so ONE barrier per tile suffices (write(i) only needs
  // compute(i-2) complete, which the barrier of iteration i-1 guarantees)
  __shared__ __align__(16) __bf16 kt[2][KVB][DPAD + PADK];
  __shared__ __align__(16) __bf16 vt2[2][KVB][RSV];

  const int bh = blockIdx.y;
  const int qblk = blockIdx.x;
  const int bb = bh / H, hh = bh % H;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lq = lane % 32;
  const int half = lane / 32;
  const long q0 = (long)qblk * (QS * 128) + wid * (QS * 32);

  const __hip_bfloat16 *Qb = Q + bb * st.qb + hh * st.qh;
  const __hip_bfloat16 *Kb = K + bb * st.kb + hh * st.kh;
  const __hip_bfloat16 *Vb = V + bb * st.vb + hh * st.vh;
  __hip_bfloat16 *Ob = O + bb * st.ob + hh * st.oh;

  // staged-piece coordinates: pieces [0,NG) are K rows, [NG,2NG) V rows;
  // NG is a multiple of 64 so the K/V split is wave-uniform and every
  // piece costs the same (one b128 write). Coordinates are recomputed on
  // the rare clamped paths instead of held live (VGPR budget).
  const auto piece = [&](int s, int &r, int &c, bool &v) {
    const int pc = threadIdx.x + s * 256;
    const int q = (pc < NG) ? pc : pc - NG;
    v = pc >= NG;
    r = q / (DPAD / 8);
    c = q % (DPAD / 8);
  };
  int eoff[NST];
#pragma unroll
  for (int s = 0; s < NST; ++s) {
    int r, c; bool v; piece(s, r, c, v);
    eoff[s] = v ? (r * RSV + c * 8) : (r * (DPAD + PADK) + c * 8);
  }
  // clamp the d-group so the ragged head dim never reads past a row
  const int dmax = D / 8 - 1;  // last full 8-group (D%8==0 guaranteed)

  // Q fragments with scale*log2(e) folded in (one extra bf16 rounding)
  const float l2scale = scale * 1.4426950408889634f;
  bf16x8 qf[QS][NC];
#pragma unroll
  for (int qs = 0; qs < QS; ++qs) {
    const long qq = q0 + qs * 32 + lq;
    const long qrow = (qq < Sq) ? qq : (Sq - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int d0 = c * 16 + 8 * half;
      bf16x8 qv = (bf16x8){};
      if (d0 + 8 <= D) {
        qv = *(const bf16x8 *)(Qb + qrow * st.qr + d0);
#pragma unroll
        for (int i = 0; i < 8; ++i)
          qv[i] = (__bf16)((float)qv[i] * l2scale);
      }
      qf[qs][c] = qv;
    }
  }

  // per-lane tr-read base LDS byte address (probe-decoded mapping): the
  // per-read row/col deltas are added as small unsigned offsets
  const unsigned vbase =
      (unsigned)(unsigned long long)(const char *)&vt2[0][0][0] +
      ((((lane & 15) >> 2) + 8 * half) * RSV + 4 * (lane & 3) +
       ((lane >> 4) & 1) * 16) *
          2;

  f32x16 o[QS][ND];
#pragma unroll
  for (int qs = 0; qs < QS; ++qs)
#pragma unroll
    for (int d = 0; d < ND; ++d) o[qs][d] = (f32x16){};
  float m[QS], l[QS];
#pragma unroll
  for (int qs = 0; qs < QS; ++qs) { m[qs] = -1e30f; l[qs] = 0.f; }

  // per-thread running piece pointers: tile 0 clamped; full-tile advances
  // are one uniform-delta 64-bit add instead of strided re-derivation
  const __hip_bfloat16 *pp[NST];
#pragma unroll
  for (int s = 0; s < NST; ++s) {
    int sr, sc; bool v; piece(s, sr, sc, v);
    const long r = (sr < Sk) ? sr : (Sk - 1);
    const int cg = (sc <= dmax) ? sc : dmax;
    pp[s] = (v ? Vb + r * st.vr : Kb + r * st.kr) + cg * 8;
  }
  const long kdelta = KVB * st.kr, vdelta = KVB * st.vr;

  // prologue: issue tile 0's loads (clamped; OOB keys masked at score time)
  bf16x8 stg[ASYNC ? NST : 1];
  if constexpr (ASYNC) {
#pragma unroll
    for (int s = 0; s < NST; ++s) stg[s] = *(const bf16x8 *)pp[s];
  }

  int cur = 0;
  for (long kv = 0; kv < Sk; kv += KVB, cur ^= 1) {
    if constexpr (ASYNC) {
#pragma unroll
      for (int s = 0; s < NST; ++s) {
        const bool v = threadIdx.x + s * 256 >= NG;
        *(bf16x8 *)((v ? &vt2[cur][0][0] : &kt[cur][0][0]) + eoff[s]) =
            stg[s];
      }
    } else {
      // synchronous cooperative stage (VGPR-tight big head dims)
#pragma unroll
      for (int s = 0; s < NST; ++s) {
        const bool v = threadIdx.x + s * 256 >= NG;
        *(bf16x8 *)((v ? &vt2[cur][0][0] : &kt[cur][0][0]) + eoff[s]) =
            *(const bf16x8 *)pp[s];
      }
      if (kv + 2 * KVB <= Sk) {
#pragma unroll
        for (int s = 0; s < NST; ++s)
          pp[s] += (threadIdx.x + s * 256 >= NG) ? vdelta : kdelta;
      } else if (kv + KVB < Sk) {
#pragma unroll
        for (int s = 0; s < NST; ++s) {
          int sr, sc; bool v; piece(s, sr, sc, v);
          const long rr = kv + KVB + sr;
          const long r = (rr < Sk) ? rr : (Sk - 1);
          const int cg = (sc <= dmax) ? sc : dmax;
          pp[s] = (v ? Vb + r * st.vr : Kb + r * st.kr) + cg * 8;
        }
      }
    }
    __syncthreads();
    // async-STAGE: issue the NEXT tile's loads now; they complete under
    // this tile's compute (the compiler places the vmcnt at first reuse)
    if constexpr (ASYNC) {
      if (kv + KVB < Sk) {
        if (kv + 2 * KVB <= Sk) {
#pragma unroll
          for (int s = 0; s < NST; ++s)
            pp[s] += (threadIdx.x + s * 256 >= NG) ? vdelta : kdelta;
        } else {
#pragma unroll
          for (int s = 0; s < NST; ++s) {
            int sr, sc; bool v; piece(s, sr, sc, v);
            const long rr = kv + KVB + sr;
            const long r = (rr < Sk) ? rr : (Sk - 1);
            const int cg = (sc <= dmax) ? sc : dmax;
            pp[s] = (v ? Vb + r * st.vr : Kb + r * st.kr) + cg * 8;
          }
        }
#pragma unroll
        for (int s = 0; s < NST; ++s) stg[s] = *(const bf16x8 *)pp[s];
      }
    }

    const bool fulltile = (kv + KVB <= Sk);
#pragma unroll
    for (int qs = 0; qs < QS; ++qs) {
      f32x16 stile[2] = {(f32x16){}, (f32x16){}};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int c = 0; c < NC; ++c) {
          bf16x8 kf =
              *(const bf16x8 *)&kt[cur][sub * 32 + lq][c * 16 + 8 * half];
          stile[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              kf, qf[qs][c], stile[sub], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);

      // softmax IN the accumulator registers (no p[] copy: VGPR budget)
      float pmax = -1e30f;
      if (fulltile) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r)
            pmax = fmaxf(pmax, stile[sub][r]);
      } else {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kk = sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
            if (kv + kk >= Sk) stile[sub][r] = -1e30f;
            pmax = fmaxf(pmax, stile[sub][r]);
          }
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, WAVE));

      // defer-max: rescale only when some row's max grew past THR
      if (!__all(pmax <= m[qs] + THR)) {
        const float mnew = fmaxf(m[qs], pmax);
        const float alpha = __builtin_amdgcn_exp2f(m[qs] - mnew);
        l[qs] *= alpha;
        m[qs] = mnew;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
          const float a = __shfl(alpha, qrow, WAVE);
#pragma unroll
          for (int d = 0; d < ND; ++d) o[qs][d][r] *= a;
        }
      }
      float rowsum = 0.f;
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          stile[sub][r] = __builtin_amdgcn_exp2f(stile[sub][r] - m[qs]);
          rowsum += stile[sub][r];
        }
      rowsum += __shfl_xor(rowsum, 32, WAVE);
      l[qs] += rowsum;

#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        unsigned pk[8];
#pragma unroll
        for (int t = 0; t < 8; ++t)
          pk[t] = pack_bf16(stile[sub][2 * t], stile[sub][2 * t + 1]);
        bf16x8 pa[2];
        {
          auto r0 =
              __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
          auto r1 =
              __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
          unsigned frag[4] = {r0[0], r1[0], r0[1], r1[1]};
          pa[0] = *(bf16x8 *)frag;
          auto r2 =
              __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
          auto r3 =
              __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
          unsigned frag1[4] = {r2[0], r3[0], r2[1], r3[1]};
          pa[1] = *(bf16x8 *)frag1;
        }
        // PV with tr-read V fragments: per 16-key chunk, 2 transpose
        // reads per d-tile deliver V[k..k+7][dcol] straight from the
        // row-major tile (guide T10; mapping from tools/tr16_probe.py)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
          for (int db = 0; db < ND; db += 2) {  // <=2 d-tiles per wait:
            constexpr int DB = 2;               // bounds tr live ranges
            unsigned long long tv[DB][2];
#pragma unroll
            for (int dd = 0; dd < DB; ++dd) {
              const int d = db + dd;
              if (d >= ND) break;
              const unsigned a0 =
                  vbase + (unsigned)(cur * (KVB * RSV * 2)) +
                  (unsigned)(((sub * 32 + kc * 16) * RSV + d * 32) * 2);
              asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                           "ds_read_b64_tr_b16 %1, %3"
                           : "=&v"(tv[dd][0]), "=&v"(tv[dd][1])
                           : "v"(a0), "v"(a0 + (unsigned)(8 * RSV)));
            }
            // the wait NAMES the tr destinations (guide §5.7 form ii):
            // the consuming MFMAs then depend on it through registers, so
            // no sched_barrier wall is needed and independent work (the
            // other q-subtile's chain) may schedule into the LDS shadow
            asm volatile("s_waitcnt lgkmcnt(0)"
                         : "+v"(tv[0][0]), "+v"(tv[0][1]), "+v"(tv[1][0]),
                           "+v"(tv[1][1])
                         :
                         : "memory");
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int dd = 0; dd < DB; ++dd) {
              const int d = db + dd;
              if (d >= ND) break;
              union {
                unsigned long long u[2];
                bf16x8 v;
              } cv;
              cv.u[0] = tv[dd][0];
              cv.u[1] = tv[dd][1];
              o[qs][d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  pa[kc], cv.v, o[qs][d], 0, 0, 0);
            }
            __builtin_amdgcn_s_setprio(0);
          }
        }
      }
    }
  }

#pragma unroll
  for (int qs = 0; qs < QS; ++qs) {
    const float linv = 1.0f / fmaxf(l[qs], 1e-30f);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
      const float inv = __shfl(linv, qrow, WAVE);
      const long qq = q0 + qs * 32 + qrow;
      if (qq >= Sq) continue;
#pragma unroll
      for (int d = 0; d < ND; ++d)
        if (d * 32 + lq < D)
          Ob[qq * st.or_ + d * 32 + lq] = f2bf(o[qs][d][r] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// host-side dispatch (torch API lives in ext.hip which includes this file)
// ---------------------------------------------------------------------------
bool flash_supported(long d_head) {
  return d_head > 16 && d_head <= 192 && (d_head % 8) == 0;
}

#if defined(__HIP_PLATFORM_AMD__) && !defined(SDWD_NO_TORCH)
static inline long round16(long d) { return (d + 15) / 16 * 16; }

// qkv layout: "bhsd" (q.size = [B,H,S,D]) or "bshd" ([B,S,H,D]); tensors
// need unit stride in D and 16-byte-aligned row starts, NOT full contiguity.
static torch::Tensor flash_attention_raw(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v, double scale,
                                         bool bshd) {
  const long B = q.size(0);
  const long H = bshd ? q.size(2) : q.size(1);
  const long Sq = bshd ? q.size(1) : q.size(2);
  const long Sk = bshd ? k.size(1) : k.size(2);
  const long D = q.size(3);
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "flash: D must be unit-stride");
  auto out = bshd ? torch::empty({B, Sq, H, D}, q.options())
                  : torch::empty({B, H, Sq, D}, q.options());
  AttnStrides st;
  if (bshd) {
    st.qb = q.stride(0); st.qh = q.stride(2); st.qr = q.stride(1);
    st.kb = k.stride(0); st.kh = k.stride(2); st.kr = k.stride(1);
    st.vb = v.stride(0); st.vh = v.stride(2); st.vr = v.stride(1);
    st.ob = out.stride(0); st.oh = out.stride(2); st.or_ = out.stride(1);
  } else {
    st.qb = q.stride(0); st.qh = q.stride(1); st.qr = q.stride(2);
    st.kb = k.stride(0); st.kh = k.stride(1); st.kr = k.stride(2);
    st.vb = v.stride(0); st.vh = v.stride(1); st.vr = v.stride(2);
    st.ob = out.stride(0); st.oh = out.stride(1); st.or_ = out.stride(2);
  }
  const long DP = round16(D);
  dim3 block(256);
  auto stream = cur_stream();
  static const bool use_v2 = [] {
    const char *e = getenv("SDWD_ATTN");
    return e && strcmp(e, "v2") == 0;
  }();
  static const bool v3all = [] {  // measure v3 past DPAD 64 (spilly)
    const char *e = getenv("SDWD_ATTN");
    return e && strcmp(e, "v3all") == 0;
  }();

#define LAUNCH_FLASH(DP_)                                                   \
  do {                                                                      \
    /* measured dispatch: v3 wins every self-attention shape (at D>64     \
       its prologue/epilogue spills are off the hot loop: +24% at D=80,  \
       +7% at D=160); the short-Sk cross shapes (Sk=77) stay on v2 */    \
    if (use_v2 || (DP_ > 64 && !v3all && Sk < 256)) {                    \
      dim3 grid((unsigned)((Sq + 255) / 256), (unsigned)(B * H));           \
      hipLaunchKernelGGL(flash_fwd_bf16_kernel<DP_>, grid, block, 0,        \
                         stream, (const __hip_bfloat16 *)q.data_ptr(),      \
                         (const __hip_bfloat16 *)k.data_ptr(),              \
                         (const __hip_bfloat16 *)v.data_ptr(),              \
                         (__hip_bfloat16 *)out.data_ptr(), (int)H, Sq, Sk,  \
                         (int)D, (float)scale, st);                         \
    } else {                                                                \
      constexpr long ROWS_ = (DP_ <= 96 ? 256 : 128); /* QS*128 */          \
      dim3 grid((unsigned)((Sq + ROWS_ - 1) / ROWS_), (unsigned)(B * H));   \
      hipLaunchKernelGGL(flash_fwd_bf16_v3<DP_>, grid, block, 0, stream,    \
                         (const __hip_bfloat16 *)q.data_ptr(),              \
                         (const __hip_bfloat16 *)k.data_ptr(),              \
                         (const __hip_bfloat16 *)v.data_ptr(),              \
                         (__hip_bfloat16 *)out.data_ptr(), (int)H, Sq, Sk,  \
                         (int)D, (float)scale, st);                         \
    }                                                                       \
  } while (0)

  switch (DP) {
    case 32: LAUNCH_FLASH(32); break;
    case 48: LAUNCH_FLASH(48); break;
    case 64: LAUNCH_FLASH(64); break;
    case 80: LAUNCH_FLASH(80); break;
    case 96: LAUNCH_FLASH(96); break;
    case 112: LAUNCH_FLASH(112); break;
    case 128: LAUNCH_FLASH(128); break;
    case 144: LAUNCH_FLASH(144); break;
    case 160: LAUNCH_FLASH(160); break;
    case 176: LAUNCH_FLASH(176); break;
    case 192: LAUNCH_FLASH(192); break;
    default: TORCH_CHECK(false, "flash: unsupported padded head dim ", DP);
  }
#undef LAUNCH_FLASH
  return out;
}

torch::Tensor flash_attention_bf16(torch::Tensor q, torch::Tensor k,
                                   torch::Tensor v, double scale) {
  return flash_attention_raw(q, k, v, scale, /*bshd=*/false);
}
#endif
