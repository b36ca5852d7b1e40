// Flash attention backward v4 for gfx950 — 32×32 MFMA structure with
// in-register P/dS redistribution (no per-wave LDS round trips), mirroring
// attention_fwd32.hip. D ∈ {64, 128} and S % 256 == 0 (v3 fallback else);
// the D=128 dkv build trades K/V register residency and T14 staging for
// accumulator headroom (see KV_RES) — 256 VGPRs, 6 spilled, occupancy 2.
//
//   dq kernel : grid over 256-row Q blocks (8 waves × 32 q rows);
//               per 64-kv tile recompute S^T = K·Q^T and dP^T = V·dO^T
//               (swapped: C col = q), dS^T in registers, dQ += dS·K with
//               dS redistributed C→A by two v_permlane32_swap per K-step.
//   dkv kernel: grid over 256-row KV blocks (8 waves × 32 kv rows);
//               per 64-q tile S = Q·K^T, dP = dO·V^T (C col = kv),
//               dV += P^T·dO and dK += dS^T·Q via the same redistribution.

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int KT = 64;            // staged tile rows (kv in dq, q in dkv)
constexpr int PAD = 8;
constexpr int LST = KT + PAD;     // 72
constexpr int QW = 32;            // rows per wave
constexpr int NW = 8;
constexpr int BT = QW * NW;       // 256 rows per workgroup

#define MFMA32(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((a), (b), (c), 0, 0, 0)

ACCO_DEV unsigned pack_bf16_(float lo, float hi) {
  // v_cvt_pk_bf16_f32 (RNE, no builtin on gfx950 — guide T12): one
  // instruction replaces ~9 VALU of manual round-to-nearest-even
  // bit-twiddling per packed dword. The trailing s_nop 1 covers the
  // VALU-write → v_permlane32_swap hazard window for the consumer
  // (guide T21 hazard note; hipcc pads nothing inside asm).
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
      : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// C-layout floats (16 regs of one 32-row sub-tile) → A-fragment for K-step
// kk (16 of the 32 C rows): su_j = pack(c[8kk+2j], c[8kk+2j+1]) carries row
// pairs {2hi, 2hi+1, 4+2hi, 5+2hi}; swapping su0↔su2 and su1↔su3 across
// half-waves yields dword j = row pair 4hi+j (derivation in
// attention_fwd32.hip).
ACCO_DEV short8 c_to_afrag(const float* c, int kk) {
  const float* pp = c + kk * 8;
  unsigned su0 = pack_bf16_(pp[0], pp[1]);
  unsigned su1 = pack_bf16_(pp[2], pp[3]);
  unsigned su2 = pack_bf16_(pp[4], pp[5]);
  unsigned su3 = pack_bf16_(pp[6], pp[7]);
  auto r02 = __builtin_amdgcn_permlane32_swap(su0, su2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(su1, su3, false, false);
  short8 pa;
  reinterpret_cast<unsigned*>(&pa)[0] = r02[0];
  reinterpret_cast<unsigned*>(&pa)[1] = r13[0];
  reinterpret_cast<unsigned*>(&pa)[2] = r02[1];
  reinterpret_cast<unsigned*>(&pa)[3] = r13[1];
  return pa;
}

// same redistribution over an f32x16 C fragment kept in vector registers
// (constant kk from an unrolled loop — element reads stay in VGPRs)
ACCO_DEV short8 c_to_afrag_v(const f32x16& c, int kk) {
  const int b = kk * 8;
  unsigned su0 = pack_bf16_(c[b + 0], c[b + 1]);
  unsigned su1 = pack_bf16_(c[b + 2], c[b + 3]);
  unsigned su2 = pack_bf16_(c[b + 4], c[b + 5]);
  unsigned su3 = pack_bf16_(c[b + 6], c[b + 7]);
  auto r02 = __builtin_amdgcn_permlane32_swap(su0, su2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(su1, su3, false, false);
  short8 pa;
  reinterpret_cast<unsigned*>(&pa)[0] = r02[0];
  reinterpret_cast<unsigned*>(&pa)[1] = r13[0];
  reinterpret_cast<unsigned*>(&pa)[2] = r02[1];
  reinterpret_cast<unsigned*>(&pa)[3] = r13[1];
  return pa;
}

// stage a [KT × D] tile TRANSPOSED into [D][LST] (as attention_bwd.hip,
// 512-thread block version)
template <int D>
ACCO_DEV void stage_T(const u16* src, long long stride, u16* dst) {
  const int r2 = (threadIdx.x & 31) * 2;
  for (int dg = threadIdx.x >> 5; dg < D / 8; dg += 16) {
    ushort4 a0 = reinterpret_cast<const ushort4*>(src + (long long)r2 * stride + dg * 8)[0];
    ushort4 a1 = reinterpret_cast<const ushort4*>(src + (long long)r2 * stride + dg * 8)[1];
    ushort4 b0 = reinterpret_cast<const ushort4*>(src + (long long)(r2 + 1) * stride + dg * 8)[0];
    ushort4 b1 = reinterpret_cast<const ushort4*>(src + (long long)(r2 + 1) * stride + dg * 8)[1];
    u16 av[8] = {a0.x, a0.y, a0.z, a0.w, a1.x, a1.y, a1.z, a1.w};
    u16 bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<ushort2*>(dst + (dg * 8 + i) * LST + r2) =
          make_ushort2(av[i], bv[i]);
  }
}

// stage a [KT × D] tile ROW-MAJOR into [KT][D+8] (512-thread block)
template <int D>
ACCO_DEV void stage_R(const u16* src, long long stride, u16* dst) {
  for (int c = threadIdx.x; c < KT * (D / 8); c += 512) {
    const int r = c / (D / 8), dc = c % (D / 8);
    reinterpret_cast<uint4*>(dst + r * (D + 8))[dc] =
        *reinterpret_cast<const uint4*>(src + (long long)r * stride + dc * 8);
  }
}

// ------------------------------------------------------------------- dQ
// D=64 capped at 128 VGPR: 2 co-resident 8-wave blocks per CU, matching
// the forward kernel's occupancy choice.
template <int D>
__global__ __launch_bounds__(512, D == 64 ? 4 : 2)
void attn_bwd32_dq_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                          const u16* __restrict__ v, const u16* __restrict__ dO,
                          const float* __restrict__ lse,
                          const float* __restrict__ delta,
                          u16* __restrict__ dq,
                          int S, int H, int Hkv, float scale, int window,
                          long long q_rs, long long kv_rs, long long do_rs,
                          long long dq_rs) {
  constexpr int KS = D / 16;
  constexpr int DT = D / 32;
  constexpr int KROW = D + 8;
  const int qt = blockIdx.x, bh = blockIdx.y;
  const int b = bh / H, h = bh % H, hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lq = lane & 31, hi = lane >> 5;

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* k_row = smem;                       // [KT][KROW]
  u16* v_row = k_row + KT * KROW;          // [KT][KROW]
  u16* kT_lds = v_row + KT * KROW;         // [D][LST]

  const long long qs = q_rs, ks = kv_rs;
  const int q0 = qt * BT + wave * QW;
  const u16* Qp = q + ((long long)b * S + q0) * qs + (long long)h * D;
  const u16* dOp = dO + ((long long)b * S + q0) * do_rs + (long long)h * D;
  const u16* Kb = k + (long long)b * S * ks + (long long)hkv * D;
  const u16* Vb = v + (long long)b * S * ks + (long long)hkv * D;

  short8 qf[KS], dof[KS];
#pragma unroll
  for (int s = 0; s < KS; ++s) {
    qf[s] = *reinterpret_cast<const short8*>(
        Qp + (long long)lq * qs + s * 16 + hi * 8);
    dof[s] = *reinterpret_cast<const short8*>(
        dOp + (long long)lq * do_rs + s * 16 + hi * 8);
  }
  // log2-domain P recompute: lse pre-scaled by log2e, delta by ·scale so
  // the per-element work is one mul, one exp2 and one fma
  const float lse2_c = lse[(long long)bh * S + q0 + lq] * 1.4426950408889634f;
  const float delta_s = delta[(long long)bh * S + q0 + lq] * scale;
  const float scale2 = scale * 1.4426950408889634f;

  f32x16 acc_dq[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc_dq[t][r] = 0.0f;

  int j_lo = 0;
  if (window > 0) {
    int kv_min = qt * BT - window + 1;
    if (kv_min > 0) j_lo = kv_min / KT;
  }
  const int j_hi = (qt * BT + BT - 1) / KT;
  const int q_wave_max = q0 + QW - 1;

  for (int j = j_lo; j <= j_hi; ++j) {
    __syncthreads();
    stage_R<D>(Kb + (long long)(j * KT) * ks, ks, k_row);
    stage_R<D>(Vb + (long long)(j * KT) * ks, ks, v_row);
    stage_T<D>(Kb + (long long)(j * KT) * ks, ks, kT_lds);
    __syncthreads();
    if (j * KT > q_wave_max) continue;

#pragma unroll
    for (int m32 = 0; m32 < 2; ++m32) {
      f32x16 st, dpt;
#pragma unroll
      for (int r = 0; r < 16; ++r) { st[r] = 0.0f; dpt[r] = 0.0f; }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        short8 kf = *reinterpret_cast<const short8*>(
            k_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
        short8 vf = *reinterpret_cast<const short8*>(
            v_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
        st = MFMA32(kf, qf[s], st);
        dpt = MFMA32(vf, dof[s], dpt);
      }
      __builtin_amdgcn_s_setprio(0);

      // dS^T (C: col=q=lq, row=kv spread); interior tiles (all kv ≤ all q,
      // all within the window) skip the per-element mask predicates
      const int q_g = q0 + lq;
      const bool need_mask = (j * KT + KT - 1 > q0) ||
                             (window > 0 && j * KT <= q_wave_max - window);
      float ds16[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float x = st[r] * scale2 - lse2_c;
        if (need_mask) {
          const int kv_g =
              j * KT + m32 * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          bool valid = (kv_g <= q_g);
          if (window > 0) valid = valid && (kv_g > q_g - window);
          x = valid ? x : -1e30f;
        }
        const float pval = __builtin_amdgcn_exp2f(x);
        ds16[r] = pval * __builtin_fmaf(dpt[r], scale, -delta_s);
      }

      // dQ += dS·K (A = dS[row=q][k=kv] via swap; B = K^T from kT_lds)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        short8 dsa = c_to_afrag(ds16, kk);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 kb = *reinterpret_cast<const short8*>(
              kT_lds + (t * 32 + lq) * LST + m32 * 32 + kk * 16 + hi * 8);
          acc_dq[t] = MFMA32(dsa, kb, acc_dq[t]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  u16* dQp = dq + ((long long)b * S + q0) * dq_rs + (long long)h * D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
#pragma unroll
    for (int t = 0; t < DT; ++t)
      dQp[(long long)qrow * dq_rs + t * 32 + lq] = f32_to_bf16(acc_dq[t][r]);
  }
}

// ---------------------------------------------------------------- dK, dV
template <int D>
__global__ __launch_bounds__(512)
void attn_bwd32_dkv_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                           const u16* __restrict__ v,
                           const u16* __restrict__ dO,
                           const float* __restrict__ lse,
                           const float* __restrict__ delta,
                           u16* __restrict__ dk, u16* __restrict__ dv,
                           int S, int H, int Hkv, float scale, int window,
                           long long q_rs, long long kv_rs, long long do_rs) {
  constexpr int KS = D / 16;
  constexpr int DT = D / 32;
  constexpr int KROW = D + 8;
  const int jb = blockIdx.x, bh = blockIdx.y;
  const int b = bh / H, h = bh % H, hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lq = lane & 31, hi = lane >> 5;

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* q_row = smem;                        // [KT][KROW]
  u16* do_row = q_row + KT * KROW;          // [KT][KROW]
  u16* qT_lds = do_row + KT * KROW;         // [D][LST]
  u16* doT_lds = qT_lds + D * LST;          // [D][LST]

  const long long qs = q_rs, ks = kv_rs;
  const long long od = (long long)H * D;   // dk/dv temps: contiguous [B,S,H,D]
  const int kv0 = jb * BT + wave * QW;
  const u16* Kp = k + ((long long)b * S + kv0) * ks + (long long)hkv * D;
  const u16* Vp = v + ((long long)b * S + kv0) * ks + (long long)hkv * D;
  const u16* Qb = q + (long long)b * S * qs + (long long)h * D;
  const u16* dOb = dO + (long long)b * S * do_rs + (long long)h * D;

  // K^T / V^T as B operands: lane = K[kv=lq][d=hi*8+i+16s].
  // Register-resident at D=64 (64 VGPRs); at D=128 that residency pushes
  // the kernel to 91 spilled VGPRs, so the fragments are re-read from the
  // L2-hot global rows inside the K-step loop instead.
  constexpr bool KV_RES = (D == 64);
  short8 kTf[KV_RES ? KS : 1], vTf[KV_RES ? KS : 1];
  if constexpr (KV_RES) {
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      kTf[s] = *reinterpret_cast<const short8*>(
          Kp + (long long)lq * ks + s * 16 + hi * 8);
      vTf[s] = *reinterpret_cast<const short8*>(
          Vp + (long long)lq * ks + s * 16 + hi * 8);
    }
  }

  f32x16 acc_dk[DT], acc_dv[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) { acc_dk[t][r] = 0.f; acc_dv[t][r] = 0.f; }

  int qt_hi = S / KT - 1;
  if (window > 0) {
    const int q_lim = jb * BT + BT - 1 + window;
    qt_hi = min(qt_hi, q_lim / KT);
  }
  const int qt_lo = (jb * BT) / KT;
  const int kv_wave_min = kv0;

  // T14 async staging: hold the next q tile's Q/dO loads in registers
  // across the barrier; the loads stay in flight under the MFMA work.
  // Transpose staging uses the (d-pair, kv-oct) item scheme of the
  // forward's V staging: 8 coalesced uint row loads → two ds_write_b128
  // (replaces 8 narrow ushort2 writes per tensor — 2.4× fewer LDS write
  // cycles). Threads < half stage Q^T, the other half dO^T.
  const bool t_q = (threadIdx.x < 256);   // D=64 only: 256 items per tensor
  const int t_id = threadIdx.x & 255;
  const int t_d0 = (t_id % 32) * 2;
  const int t_kv8 = (t_id / 32) * 8;
  unsigned streg[8];
  // row-major staging piece: one uint4 per thread per tensor (KT*D/8/512)
  static_assert(KT * (D / 8) % 512 == 0, "rowmajor chunks");
  constexpr int RCH = KT * (D / 8) / 512;
  uint4 rq[RCH], rdo[RCH];

  auto load_qtile = [&](int qt) {
    const u16* Qs = Qb + (long long)(qt * KT) * qs;
    const u16* Ds = dOb + (long long)(qt * KT) * do_rs;
    if (D == 64) {
      // half the threads stage Q^T items, half dO^T items
      const u16* Ts = t_q ? Qs : Ds;
      const long long tstride = t_q ? qs : do_rs;
#pragma unroll
      for (int r = 0; r < 8; ++r)
        streg[r] = *reinterpret_cast<const unsigned*>(
            Ts + (long long)(t_kv8 + r) * tstride + t_d0);
    }
#pragma unroll
    for (int c = 0; c < RCH; ++c) {
      const int cc = threadIdx.x + c * 512;
      const int r = cc / (D / 8), dc = cc % (D / 8);
      rq[c] = *reinterpret_cast<const uint4*>(Qs + (long long)r * qs + dc * 8);
      rdo[c] = *reinterpret_cast<const uint4*>(Ds + (long long)r * do_rs + dc * 8);
    }
  };
  auto write_qtile = [&]() {
    if (D == 64) {
      uint4 lo, hi4;
      lo.x = (streg[0] & 0xffffu) | (streg[1] << 16);
      lo.y = (streg[2] & 0xffffu) | (streg[3] << 16);
      lo.z = (streg[4] & 0xffffu) | (streg[5] << 16);
      lo.w = (streg[6] & 0xffffu) | (streg[7] << 16);
      hi4.x = (streg[0] >> 16) | (streg[1] & 0xffff0000u);
      hi4.y = (streg[2] >> 16) | (streg[3] & 0xffff0000u);
      hi4.z = (streg[4] >> 16) | (streg[5] & 0xffff0000u);
      hi4.w = (streg[6] >> 16) | (streg[7] & 0xffff0000u);
      u16* dst = t_q ? qT_lds : doT_lds;
      *reinterpret_cast<uint4*>(dst + t_d0 * LST + t_kv8) = lo;
      *reinterpret_cast<uint4*>(dst + (t_d0 + 1) * LST + t_kv8) = hi4;
    }
#pragma unroll
    for (int c = 0; c < RCH; ++c) {
      const int cc = threadIdx.x + c * 512;
      const int r = cc / (D / 8), dc = cc % (D / 8);
      reinterpret_cast<uint4*>(q_row + r * KROW)[dc] = rq[c];
      reinterpret_cast<uint4*>(do_row + r * KROW)[dc] = rdo[c];
    }
  };

  // direct (non-T14) staging for the !KV_RES (D=128) build: the
  // cross-barrier register staging holds 32 VGPRs for the whole loop and
  // tips the 128-VGPR accumulator build into spilling
  auto stage_qtile_direct = [&](int qt) {
    const u16* Qs = Qb + (long long)(qt * KT) * qs;
    const u16* Ds = dOb + (long long)(qt * KT) * do_rs;
    stage_T<D>(Qs, qs, qT_lds);
    stage_T<D>(Ds, do_rs, doT_lds);
#pragma unroll
    for (int c = 0; c < RCH; ++c) {
      const int cc = threadIdx.x + c * 512;
      const int r = cc / (D / 8), dc = cc % (D / 8);
      reinterpret_cast<uint4*>(q_row + r * KROW)[dc] =
          *reinterpret_cast<const uint4*>(Qs + (long long)r * qs + dc * 8);
      reinterpret_cast<uint4*>(do_row + r * KROW)[dc] =
          *reinterpret_cast<const uint4*>(Ds + (long long)r * do_rs + dc * 8);
    }
  };

  if constexpr (KV_RES) load_qtile(qt_lo);
  for (int qt = qt_lo; qt <= qt_hi; ++qt) {
    __syncthreads();
    if constexpr (KV_RES) {
      write_qtile();
    } else {
      stage_qtile_direct(qt);
    }
    __syncthreads();
    if constexpr (KV_RES)
      if (qt < qt_hi) load_qtile(qt + 1);  // in flight under the MFMAs
    // tile fully before this wave's kv rows → all masked: skip compute
    if (qt * KT + KT - 1 < kv_wave_min) continue;

#pragma unroll
    for (int m32 = 0; m32 < 2; ++m32) {
      // per-lane lse/delta for q = qt*KT + m32*32 + lq (broadcast by row),
      // pre-scaled: lse → log2 domain, delta → ·scale (fma fold below)
      const float lse_l =
          lse[(long long)bh * S + qt * KT + m32 * 32 + lq] * 1.4426950408889634f;
      const float del_l =
          delta[(long long)bh * S + qt * KT + m32 * 32 + lq] * scale;

      f32x16 st, dpt;
#pragma unroll
      for (int r = 0; r < 16; ++r) { st[r] = 0.0f; dpt[r] = 0.0f; }
      __builtin_amdgcn_s_setprio(1);
      if constexpr (KV_RES) {
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          short8 qfr = *reinterpret_cast<const short8*>(
              q_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
          short8 dofr = *reinterpret_cast<const short8*>(
              do_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
          st = MFMA32(qfr, kTf[s], st);
          dpt = MFMA32(dofr, vTf[s], dpt);
        }
      } else {
        // non-resident K/V (D=128): bounded unroll keeps the in-flight
        // global loads from ballooning the register file (full unroll of
        // 8 K-steps spills)
#pragma unroll 2
        for (int s = 0; s < KS; ++s) {
          short8 qfr = *reinterpret_cast<const short8*>(
              q_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
          short8 dofr = *reinterpret_cast<const short8*>(
              do_row + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
          short8 kf = *reinterpret_cast<const short8*>(
              Kp + (long long)lq * ks + s * 16 + hi * 8);
          short8 vf = *reinterpret_cast<const short8*>(
              Vp + (long long)lq * ks + s * 16 + hi * 8);
          st = MFMA32(qfr, kf, st);
          dpt = MFMA32(dofr, vf, dpt);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // P and dS (C: col = kv = lq, row = q spread) — written back into
      // st/dpt in place: separate p16/ds16 arrays cost 32 VGPRs and spill
      // the D=128 instantiation
      const int kv_g = kv0 + lq;
      const float scale2_ = scale * 1.4426950408889634f;
      // interior q tiles (all q ≥ all kv of this wave, all within window)
      // skip the per-element mask predicates; (q_g < S) holds by S % KT == 0
      const bool need_mask =
          (qt * KT < kv0 + QW - 1) ||
          (window > 0 && kv0 + window <= qt * KT + KT - 1);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int rr = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float lse_q = __shfl(lse_l, rr, 64);
        const float del_q = __shfl(del_l, rr, 64);
        float x = st[r] * scale2_ - lse_q;
        if (need_mask) {
          const int q_g = qt * KT + m32 * 32 + rr;
          bool valid = (kv_g <= q_g);
          if (window > 0) valid = valid && (kv_g > q_g - window);
          x = valid ? x : -1e30f;
        }
        const float pval = __builtin_amdgcn_exp2f(x);
        st[r] = pval;
        dpt[r] = pval * __builtin_fmaf(dpt[r], scale, -del_q);
      }

      // dV += P^T·dO ; dK += dS^T·Q   (A rows = kv via swap; B from LDS^T)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        short8 pa = c_to_afrag_v(st, kk);
        short8 dsa = c_to_afrag_v(dpt, kk);
        // NOTE: must stay fully unrolled — t indexes the accumulator
        // register arrays (guide rule 20: runtime indexing demotes them
        // to scratch; a partial-unroll experiment cost 576 B/lane)
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 dob = *reinterpret_cast<const short8*>(
              doT_lds + (t * 32 + lq) * LST + m32 * 32 + kk * 16 + hi * 8);
          short8 qb = *reinterpret_cast<const short8*>(
              qT_lds + (t * 32 + lq) * LST + m32 * 32 + kk * 16 + hi * 8);
          acc_dv[t] = MFMA32(pa, dob, acc_dv[t]);
          acc_dk[t] = MFMA32(dsa, qb, acc_dk[t]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  u16* dKp = dk + ((long long)b * S + kv0) * od + (long long)h * D;
  u16* dVp = dv + ((long long)b * S + kv0) * od + (long long)h * D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int krow = (r & 3) + 8 * (r >> 2) + 4 * hi;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      dKp[(long long)krow * od + t * 32 + lq] = f32_to_bf16(acc_dk[t][r]);
      dVp[(long long)krow * od + t * 32 + lq] = f32_to_bf16(acc_dv[t][r]);
    }
  }
}

}  // namespace

extern "C" {

void acco_attn_bwd32_dq(const void* q, const void* k, const void* v,
                        const void* dO, const float* lse, const float* delta,
                        void* dq, int B, int S, int H, int Hkv, int D,
                        float scale, int window, long long q_rs,
                        long long kv_rs, long long do_rs, long long dq_rs,
                        hipStream_t stream) {
  dim3 grid(S / BT, B * H);
  const int lds = (2 * KT * (D + 8) + D * LST) * sizeof(u16);
  if (D == 64)
    hipLaunchKernelGGL(attn_bwd32_dq_kernel<64>, grid, dim3(512), lds, stream,
                       (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dO, lse, delta, (u16*)dq, S, H, Hkv,
                       scale, window, q_rs, kv_rs, do_rs, dq_rs);
  else
    hipLaunchKernelGGL(attn_bwd32_dq_kernel<128>, grid, dim3(512), lds,
                       stream, (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dO, lse, delta, (u16*)dq, S, H, Hkv,
                       scale, window, q_rs, kv_rs, do_rs, dq_rs);
}

void acco_attn_bwd32_dkv(const void* q, const void* k, const void* v,
                         const void* dO, const float* lse,
                         const float* delta, void* dk, void* dv, int B,
                         int S, int H, int Hkv, int D, float scale,
                         int window, long long q_rs, long long kv_rs,
                         long long do_rs, hipStream_t stream) {
  dim3 grid(S / BT, B * H);
  const int lds = (2 * KT * (D + 8) + 2 * D * LST) * sizeof(u16);
  if (D == 64)
    hipLaunchKernelGGL(attn_bwd32_dkv_kernel<64>, grid, dim3(512), lds,
                       stream, (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dO, lse, delta, (u16*)dk, (u16*)dv, S, H,
                       Hkv, scale, window, q_rs, kv_rs, do_rs);
  else
    hipLaunchKernelGGL(attn_bwd32_dkv_kernel<128>, grid, dim3(512), lds,
                       stream, (const u16*)q, (const u16*)k, (const u16*)v,
                       (const u16*)dO, lse, delta, (u16*)dk, (u16*)dv, S, H,
                       Hkv, scale, window, q_rs, kv_rs, do_rs);
}

}  // extern "C"
