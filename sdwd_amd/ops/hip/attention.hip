// Flash-style fused attention forward for gfx950 (MFMA 32x32x16 bf16).
//
// Structure (v1, correctness-first with the known-good idioms):
//  * workgroup = 4 waves; each wave owns 32 q-rows, the WG shares K/V tiles
//    of 32 keys staged in LDS (K row-major padded, V transposed for
//    contiguous B-fragment reads).
//  * swapped QK^T: S^T = mfma(K_frag, Q_frag) so each lane holds the scores
//    of ONE q-row (lane%32) across 16 of 32 keys — softmax is register-local
//    plus one __shfl_xor(32) to combine the half-wave pair.
//  * P (bf16-packed) is redistributed to the PV A-fragment layout with
//    v_permlane32_swap pairs (guide T12), then PV accumulates fp32 via MFMA.
//  * online softmax with per-tile rescale; the previous tile's PV completes
//    before the rescale decision (textbook order, no defer threshold yet).
//
// Contract: D == DPAD (multiple of 16, <= 192); callers pad the head dim.
// Q,K,V,O: [BH, S, DPAD] bf16 contiguous. UNet shapes: D 40->48, 80, 160;
// SDXL 64; VAE (D=512) takes the composed fallback path in ext.hip.
#include "common.h"

__device__ __forceinline__ unsigned pack_bf16(float lo, float hi) {
  union {
    __hip_bfloat162 h2;
    unsigned u;
  } cvt;
  cvt.h2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
  return cvt.u;
}

template <int DPAD>
__launch_bounds__(256, 2) __global__ void flash_fwd_bf16_kernel(
    const __hip_bfloat16 *__restrict__ Q, const __hip_bfloat16 *__restrict__ K,
    const __hip_bfloat16 *__restrict__ V, __hip_bfloat16 *__restrict__ O,
    long Sq, long Sk, float scale) {
  constexpr int KVB = 32;
  constexpr int PADK = 8;   // bf16 per-row pad: breaks ds_read_b128 conflicts
  constexpr int NC = DPAD / 16;          // QK^T k-chunks
  constexpr int DV = (DPAD + 31) / 32 * 32;  // PV d extent (32-col O tiles)
  constexpr int ND = DV / 32;            // PV d-chunks (O accum tiles)

  __shared__ __align__(16) __bf16 kt[KVB][DPAD + PADK];
  __shared__ __align__(16) __bf16 vt[DV][KVB + PADK];

  const int bh = blockIdx.y;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lq = lane % 32;   // q-row (softmax) / d-col (PV C) index
  const int half = lane / 32; // half-wave id
  const long q0 = (long)blockIdx.x * 128 + wid * 32;

  const __hip_bfloat16 *Qb = Q + (long)bh * Sq * DPAD;
  const __hip_bfloat16 *Kb = K + (long)bh * Sk * DPAD;
  const __hip_bfloat16 *Vb = V + (long)bh * Sk * DPAD;
  __hip_bfloat16 *Ob = O + (long)bh * Sq * DPAD;

  // Q fragments for the whole row-block, read once:
  // B-frag of mfma(K,Q): lane holds Q[q=lq][d = c*16 + 8*half + i]
  bf16x8 qf[NC];
  {
    const long qrow = (q0 + lq < Sq) ? (q0 + lq) : (Sq - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c)
      qf[c] = *(const bf16x8 *)(Qb + qrow * DPAD + c * 16 + 8 * half);
  }

  f32x16 o[ND];
#pragma unroll
  for (int d = 0; d < ND; ++d) o[d] = (f32x16){};
  float m = -1e30f, l = 0.f;

  // zero vt's pad rows once (DPAD..DV); they are never re-staged
  if (DV != DPAD) {
    for (int idx = threadIdx.x; idx < (DV - DPAD) * (KVB + PADK); idx += 256)
      vt[DPAD + idx / (KVB + PADK)][idx % (KVB + PADK)] = (__bf16)0.0f;
  }

  for (long kv = 0; kv < Sk; kv += KVB) {
    __syncthreads();  // previous tile's LDS reads complete
    // cooperative K/V stage: 256 threads, 8 bf16 each per step
    for (int idx = threadIdx.x; idx < KVB * NC * 2; idx += 256) {
      const int r = idx / (NC * 2);       // key row in tile
      const int c8 = idx % (NC * 2);      // 8-elem column group
      bf16x8 kvec = (bf16x8){};
      bf16x8 vvec = (bf16x8){};
      if (kv + r < Sk) {
        kvec = *(const bf16x8 *)(Kb + (kv + r) * DPAD + c8 * 8);
        vvec = *(const bf16x8 *)(Vb + (kv + r) * DPAD + c8 * 8);
      }
      *(bf16x8 *)&kt[r][c8 * 8] = kvec;
#pragma unroll
      for (int j = 0; j < 8; ++j) vt[c8 * 8 + j][r] = vvec[j];
      // (vt rows c8*8+j < DPAD only; pad rows stay zero)
    }
    __syncthreads();

    // S^T[32k, 32q] = sum_c K[.,c] x Q^T[c,.]
    f32x16 st = (f32x16){};
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 kf = *(const bf16x8 *)&kt[lq][c * 16 + 8 * half];
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[c], st, 0, 0, 0);
    }

    // online softmax for q-row lq; lane has k = (r&3)+8*(r>>2)+4*half
    float p[16];
    float pmax = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kk = (r & 3) + 8 * (r >> 2) + 4 * half;
      float s = (kv + kk < Sk) ? st[r] * scale : -1e30f;
      p[r] = s;
      pmax = fmaxf(pmax, s);
    }
    pmax = fmaxf(pmax, __shfl_xor(pmax, 32, WAVE));
    const float mnew = fmaxf(m, pmax);
    const float alpha = __expf(m - mnew);
    float rowsum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = __expf(p[r] - mnew);
      rowsum += p[r];
    }
    rowsum += __shfl_xor(rowsum, 32, WAVE);
    l = l * alpha + rowsum;
    m = mnew;

    // O *= alpha: O's q-row layout differs from P's (lane-local) layout, so
    // broadcast alpha[qrow] from the lane that owns that q-row (lane==qrow,
    // both halves hold identical alpha after the shfl_xor above).
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
      const float a = __shfl(alpha, qrow, WAVE);
#pragma unroll
      for (int d = 0; d < ND; ++d) o[d][r] *= a;
    }

    // pack P -> PV A-fragments via permlane32_swap (chunk k0..15, k16..31)
    unsigned pk[8];
#pragma unroll
    for (int t = 0; t < 8; ++t) pk[t] = pack_bf16(p[2 * t], p[2 * t + 1]);
    bf16x8 pa0, pa1;
    {
      auto r0 = __builtin_amdgcn_permlane32_swap(pk[0], pk[2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(pk[1], pk[3], false, false);
      unsigned w0 = r0[0], w1 = r1[0], w2 = r0[1], w3 = r1[1];
      unsigned frag[4] = {w0, w1, w2, w3};
      pa0 = *(bf16x8 *)frag;
      auto r2 = __builtin_amdgcn_permlane32_swap(pk[4], pk[6], false, false);
      auto r3 = __builtin_amdgcn_permlane32_swap(pk[5], pk[7], false, false);
      unsigned frag1[4] = {r2[0], r3[0], r2[1], r3[1]};
      pa1 = *(bf16x8 *)frag1;
    }

    // PV: O[32q, 32d] += P[32q,16k] x V[16k,32d] per d-block
#pragma unroll
    for (int d = 0; d < ND; ++d) {
      bf16x8 v0 = *(const bf16x8 *)&vt[d * 32 + lq][8 * half];
      o[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa0, v0, o[d], 0, 0, 0);
      bf16x8 v1 = *(const bf16x8 *)&vt[d * 32 + lq][16 + 8 * half];
      o[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1, v1, o[d], 0, 0, 0);
    }
  }

  // epilogue: O /= l, store (column-per-lane scatter; widen later, T21)
  const float linv = 1.0f / fmaxf(l, 1e-30f);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * half;
    const float inv = __shfl(linv, qrow, WAVE);
    if (q0 + qrow >= Sq) continue;
#pragma unroll
    for (int d = 0; d < ND; ++d)
      if (d * 32 + lq < DPAD)
        Ob[(q0 + qrow) * DPAD + d * 32 + lq] = f2bf(o[d][r] * inv);
  }
}

// ---------------------------------------------------------------------------
// host-side dispatch (torch API lives in ext.hip which includes this file)
// ---------------------------------------------------------------------------
bool flash_supported(long d_head) { return d_head > 16 && d_head <= 192; }

#ifdef __HIP_PLATFORM_AMD__
static inline long round16(long d) { return (d + 15) / 16 * 16; }

torch::Tensor flash_attention_bf16(torch::Tensor q, torch::Tensor k,
                                   torch::Tensor v, double scale) {
  const long B = q.size(0), H = q.size(1);
  const long Sq = q.size(2), Sk = k.size(2), D = q.size(3);
  const long DP = round16(D);
  auto qp = q, kp = k, vp = v;
  if (DP != D) {
    namespace F = torch::nn::functional;
    auto opts = F::PadFuncOptions({0, DP - D});
    qp = F::pad(q, opts).contiguous();
    kp = F::pad(k, opts).contiguous();
    vp = F::pad(v, opts).contiguous();
  }
  auto q3 = qp.view({B * H, Sq, DP});
  auto k3 = kp.view({B * H, Sk, DP});
  auto v3 = vp.view({B * H, Sk, DP});
  auto out = torch::empty_like(q3);
  dim3 grid((unsigned)((Sq + 127) / 128), (unsigned)(B * H));
  dim3 block(256);
  auto stream = cur_stream();

#define LAUNCH_FLASH(DP_)                                                   \
  hipLaunchKernelGGL(flash_fwd_bf16_kernel<DP_>, grid, block, 0, stream,    \
                     (const __hip_bfloat16 *)q3.data_ptr(),                 \
                     (const __hip_bfloat16 *)k3.data_ptr(),                 \
                     (const __hip_bfloat16 *)v3.data_ptr(),                 \
                     (__hip_bfloat16 *)out.data_ptr(), Sq, Sk, (float)scale)

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
  auto o4 = out.view({B, H, Sq, DP});
  if (DP != D) o4 = o4.slice(3, 0, D).contiguous();
  return o4;
}
#endif
