#include "hip/hip_runtime.h"
// Flash-style causal attention BACKWARD for gfx950 (CDNA4 MFMA) —
// SURVEY.md §2.5 K2 for the attention block, §7 "hard parts".
//
// FlashAttention-2 style two-kernel recompute scheme (no S×S
// materialization, no atomics):
//   dq kernel : grid over Q tiles; for each kv tile j<=qt recompute
//               P = exp(S - lse), dP = dO·V^T, dS = P∘(dP - Delta),
//               dQ += dS·K.
//   dkv kernel: grid over KV tiles; for each q tile qt>=j accumulate
//               dV += P^T·dO and dK += dS^T·Q.
//   Delta[b,h,q] = rowsum(dO ∘ O) is computed by the Python wrapper.
//
// Fragment-layout playbook as in the forward: MFMA operand order chosen so
// every A/B fragment is a contiguous 16-byte per-lane load; tiles used by
// all 4 waves are staged ONCE in LDS (row-major straight copies for A
// fragments; transposed with paired-row b32 writes for B fragments);
// C-layout products feeding the next MFMA's A side (P, dS) take a
// per-wave LDS round trip.

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int TILE = 64;
constexpr int PAD = 8;
constexpr int LST = TILE + PAD;       // 72: transposed-tile & P row stride

#define MFMA(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// stage a [TILE × D] global tile TRANSPOSED into LDS [D][LST]
// (paired-row ushort2 writes: half the instructions of scalar b16)
template <int D>
ACCO_DEV void stage_transposed(const u16* src, long long row_stride,
                               u16* dst) {
  const int r2 = (threadIdx.x & 31) * 2;            // tile row pair
  for (int dg = threadIdx.x >> 5; dg < D / 8; dg += 8) {
    ushort4 a0 = reinterpret_cast<const ushort4*>(
        src + (long long)r2 * row_stride + dg * 8)[0];
    ushort4 a1 = reinterpret_cast<const ushort4*>(
        src + (long long)r2 * row_stride + dg * 8)[1];
    ushort4 b0 = reinterpret_cast<const ushort4*>(
        src + (long long)(r2 + 1) * row_stride + dg * 8)[0];
    ushort4 b1 = reinterpret_cast<const ushort4*>(
        src + (long long)(r2 + 1) * row_stride + dg * 8)[1];
    u16 av[8] = {a0.x, a0.y, a0.z, a0.w, a1.x, a1.y, a1.z, a1.w};
    u16 bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<ushort2*>(dst + (dg * 8 + i) * LST + r2) =
          make_ushort2(av[i], bv[i]);
  }
}

// stage a [TILE × D] global tile ROW-MAJOR into LDS [TILE][D+8]
template <int D>
ACCO_DEV void stage_rowmajor(const u16* src, long long row_stride, u16* dst) {
  for (int c = threadIdx.x; c < TILE * (D / 8); c += 256) {
    const int r = c / (D / 8), dc = c % (D / 8);
    reinterpret_cast<uint4*>(dst + r * (D + 8))[dc] =
        *reinterpret_cast<const uint4*>(src + (long long)r * row_stride + dc * 8);
  }
}

// ------------------------------------------------------------------- dQ
template <int D, int QW>
__global__ __launch_bounds__(256)
void attn_bwd_dq_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                        const u16* __restrict__ v, const u16* __restrict__ dO,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,   // [B,H,S]
                        u16* __restrict__ dq,
                        int S, int H, int Hkv, float scale, int window) {
  constexpr int KS = D / 32;
  constexpr int DT = D / 16;
  constexpr int KROW = D + 8;
  constexpr int M2 = QW / 16;        // q sub-tiles per wave
  constexpr int QT = 4 * QW;         // q rows per workgroup
  const int qt = blockIdx.x, bh = blockIdx.y;
  const int b = bh / H, h = bh % H, hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lg = lane >> 4, lc = lane & 15;

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* kT_lds = smem;                             // [D][LST]
  u16* k_row = smem + D * LST;                    // [TILE][KROW]
  u16* v_row = k_row + TILE * KROW;               // [TILE][KROW]
  u16* ds_lds = v_row + TILE * KROW + wave * 16 * LST;   // per-wave dS^T

  const long long qs = (long long)H * D, ks = (long long)Hkv * D;
  const int q0 = qt * QT + wave * QW;
  const u16* Qp = q + ((long long)b * S + q0) * qs + (long long)h * D;
  const u16* dOp = dO + ((long long)b * S + q0) * qs + (long long)h * D;
  const u16* Kb = k + (long long)b * S * ks + (long long)hkv * D;
  const u16* Vb = v + (long long)b * S * ks + (long long)hkv * D;

  // Q / dO as B operands (swapped MFMAs): lane = [row m*16+lc][d lg*8+i]
  short8 qf[M2][KS], dof[M2][KS];
  float lse_c[M2], delta_c[M2];
#pragma unroll
  for (int m = 0; m < M2; ++m) {
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      qf[m][s] = *reinterpret_cast<const short8*>(
          Qp + (long long)(m * 16 + lc) * qs + s * 32 + lg * 8);
      dof[m][s] = *reinterpret_cast<const short8*>(
          dOp + (long long)(m * 16 + lc) * qs + s * 32 + lg * 8);
    }
    lse_c[m] = lse[(long long)bh * S + q0 + m * 16 + lc];
    delta_c[m] = delta[(long long)bh * S + q0 + m * 16 + lc];
  }

  f32x4 acc_dq[M2][DT];
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int t = 0; t < DT; ++t) acc_dq[m][t] = {0.f, 0.f, 0.f, 0.f};

  int j_lo = 0;
  if (window > 0) {
    int kv_min = qt * QT - window + 1;
    if (kv_min > 0) j_lo = kv_min / TILE;
  }
  const int j_hi = (qt * QT + QT - 1) / TILE;

  for (int j = j_lo; j <= j_hi; ++j) {
    __syncthreads();
    stage_transposed<D>(Kb + (long long)(j * TILE) * ks, ks, kT_lds);
    stage_rowmajor<D>(Kb + (long long)(j * TILE) * ks, ks, k_row);
    stage_rowmajor<D>(Vb + (long long)(j * TILE) * ks, ks, v_row);
    __syncthreads();

#pragma unroll
    for (int m = 0; m < M2; ++m) {
      // S^T = K·Q^T and dP^T = V·dO^T (C: col=q=lc, row=kv spread)
      f32x4 st[4], dpt[4];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        f32x4 a1 = {0.f, 0.f, 0.f, 0.f}, a2 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          short8 kf = *reinterpret_cast<const short8*>(
              k_row + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
          short8 vf = *reinterpret_cast<const short8*>(
              v_row + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
          a1 = MFMA(kf, qf[m][s], a1);
          a2 = MFMA(vf, dof[m][s], a2);
        }
        st[m16] = a1;
        dpt[m16] = a2;
      }
      __builtin_amdgcn_s_setprio(0);

      // dS^T = P^T ∘ (dP^T - Delta) * scale, P = exp(S*scale - lse)
      const int q_g = q0 + m * 16 + lc;
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        u16 pk[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kv_g = j * TILE + m16 * 16 + lg * 4 + r;
          bool valid = (kv_g <= q_g);
          if (window > 0) valid = valid && (kv_g > q_g - window);
          const float pval =
              valid ? __expf(st[m16][r] * scale - lse_c[m]) : 0.0f;
          const float dsv = pval * (dpt[m16][r] - delta_c[m]) * scale;
          pk[r] = f32_to_bf16(dsv);
        }
        *reinterpret_cast<ushort4*>(ds_lds + lc * LST + m16 * 16 + lg * 4) =
            make_ushort4(pk[0], pk[1], pk[2], pk[3]);
      }
      __builtin_amdgcn_s_waitcnt(0);   // lgkm: own-wave LDS writes

      // dQ += dS·K : A = dS[q=lc][kv], B = K^T[kv][d] from kT_lds
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        short8 dsa = *reinterpret_cast<const short8*>(
            ds_lds + lc * LST + s * 32 + lg * 8);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 kb = *reinterpret_cast<const short8*>(
              kT_lds + (t * 16 + lc) * LST + s * 32 + lg * 8);
          acc_dq[m][t] = MFMA(dsa, kb, acc_dq[m][t]);
        }
      }
    }
  }

  // store dQ rows q = m*16 + lg*4+r, col d=t*16+lc
  u16* dQp = dq + ((long long)b * S + q0) * qs + (long long)h * D;
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
      for (int t = 0; t < DT; ++t)
        dQp[(long long)(m * 16 + lg * 4 + r) * qs + t * 16 + lc] =
            f32_to_bf16(acc_dq[m][t][r]);
}

// ---------------------------------------------------------------- dK, dV
template <int D, int QW>
__global__ __launch_bounds__(256)
void attn_bwd_dkv_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                         const u16* __restrict__ v, const u16* __restrict__ dO,
                         const float* __restrict__ lse,
                         const float* __restrict__ delta,
                         u16* __restrict__ dk, u16* __restrict__ dv,
                         int S, int H, int Hkv, float scale, int window) {
  constexpr int KS = D / 32;
  constexpr int DT = D / 16;
  constexpr int KROW = D + 8;
  constexpr int M2 = QW / 16;       // kv sub-tiles per wave
  constexpr int KB = 4 * QW;        // kv rows per workgroup
  const int jb = blockIdx.x, bh = blockIdx.y;
  const int b = bh / H, h = bh % H, hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lg = lane >> 4, lc = lane & 15;

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* qT_lds = smem;                               // [D][LST]
  u16* doT_lds = smem + D * LST;                    // [D][LST]
  u16* q_row = doT_lds + D * LST;                   // [TILE][KROW]
  u16* do_row = q_row + TILE * KROW;                // [TILE][KROW]
  u16* p_lds = do_row + TILE * KROW + wave * 16 * LST;
  u16* ds_lds = do_row + TILE * KROW + (4 + wave) * 16 * LST;

  const long long qs = (long long)H * D, ks = (long long)Hkv * D;
  const int kv0 = jb * KB + wave * QW;              // wave's QW kv rows
  const u16* Kp = k + ((long long)b * S + kv0) * ks + (long long)hkv * D;
  const u16* Vp = v + ((long long)b * S + kv0) * ks + (long long)hkv * D;
  const u16* Qb = q + (long long)b * S * qs + (long long)h * D;
  const u16* dOb = dO + (long long)b * S * qs + (long long)h * D;

  // K, V as B operands: lane = [row kv = m*16+lc][d lg*8+i]
  short8 kTf[M2][KS], vTf[M2][KS];
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int s = 0; s < KS; ++s) {
      kTf[m][s] = *reinterpret_cast<const short8*>(
          Kp + (long long)(m * 16 + lc) * ks + s * 32 + lg * 8);
      vTf[m][s] = *reinterpret_cast<const short8*>(
          Vp + (long long)(m * 16 + lc) * ks + s * 32 + lg * 8);
    }

  f32x4 acc_dk[M2][DT], acc_dv[M2][DT];
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int t = 0; t < DT; ++t) {
      acc_dk[m][t] = {0.f, 0.f, 0.f, 0.f};
      acc_dv[m][t] = {0.f, 0.f, 0.f, 0.f};
    }

  int qt_hi = S / TILE - 1;
  if (window > 0) {
    const int q_lim = jb * KB + KB - 1 + window;
    qt_hi = min(qt_hi, q_lim / TILE);
  }
  const int qt_lo = (jb * KB) / TILE;

  for (int qt = qt_lo; qt <= qt_hi; ++qt) {
    __syncthreads();
    stage_transposed<D>(Qb + (long long)(qt * TILE) * qs, qs, qT_lds);
    stage_transposed<D>(dOb + (long long)(qt * TILE) * qs, qs, doT_lds);
    stage_rowmajor<D>(Qb + (long long)(qt * TILE) * qs, qs, q_row);
    stage_rowmajor<D>(dOb + (long long)(qt * TILE) * qs, qs, do_row);
    __syncthreads();

#pragma unroll
    for (int m = 0; m < M2; ++m) {
      // S = Q·K^T, dP = dO·V^T (C: col = kv = lc, row = q spread)
      f32x4 st[4], dpt[4];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        f32x4 a1 = {0.f, 0.f, 0.f, 0.f}, a2 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          short8 qfr = *reinterpret_cast<const short8*>(
              q_row + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
          short8 dofr = *reinterpret_cast<const short8*>(
              do_row + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
          a1 = MFMA(qfr, kTf[m][s], a1);
          a2 = MFMA(dofr, vTf[m][s], a2);
        }
        st[m16] = a1;
        dpt[m16] = a2;
      }
      __builtin_amdgcn_s_setprio(0);

      const int kv_g = kv0 + m * 16 + lc;
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        u16 ppk[4], dsk[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int q_g = qt * TILE + m16 * 16 + lg * 4 + r;
          bool valid = (kv_g <= q_g) && (q_g < S);
          if (window > 0) valid = valid && (kv_g > q_g - window);
          const float lse_q = lse[(long long)bh * S + min(q_g, S - 1)];
          const float del_q = delta[(long long)bh * S + min(q_g, S - 1)];
          const float pval = valid ? __expf(st[m16][r] * scale - lse_q) : 0.0f;
          const float dsv = pval * (dpt[m16][r] - del_q) * scale;
          ppk[r] = f32_to_bf16(pval);
          dsk[r] = f32_to_bf16(dsv);
        }
        // store P^T and dS^T: row kv=lc, col q = m16*16 + lg*4 + r
        *reinterpret_cast<ushort4*>(p_lds + lc * LST + m16 * 16 + lg * 4) =
            make_ushort4(ppk[0], ppk[1], ppk[2], ppk[3]);
        *reinterpret_cast<ushort4*>(ds_lds + lc * LST + m16 * 16 + lg * 4) =
            make_ushort4(dsk[0], dsk[1], dsk[2], dsk[3]);
      }
      __builtin_amdgcn_s_waitcnt(0);

      // dV += P^T·dO (A: P^T[kv=lc][q], B: dO^T from doT_lds)
      // dK += dS^T·Q (A: dS^T[kv=lc][q], B: Q^T from qT_lds)
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        short8 pa = *reinterpret_cast<const short8*>(
            p_lds + lc * LST + s * 32 + lg * 8);
        short8 dsa = *reinterpret_cast<const short8*>(
            ds_lds + lc * LST + s * 32 + lg * 8);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 dob = *reinterpret_cast<const short8*>(
              doT_lds + (t * 16 + lc) * LST + s * 32 + lg * 8);
          short8 qb = *reinterpret_cast<const short8*>(
              qT_lds + (t * 16 + lc) * LST + s * 32 + lg * 8);
          acc_dv[m][t] = MFMA(pa, dob, acc_dv[m][t]);
          acc_dk[m][t] = MFMA(dsa, qb, acc_dk[m][t]);
        }
      }
    }
  }

  // store dK/dV rows kv = kv0 + m*16 + lg*4 + r, col d = t*16+lc.
  // GQA: emitted per QUERY head; the wrapper group-reduces to kv heads.
  u16* dKp = dk + ((long long)b * S + kv0) * qs + (long long)h * D;
  u16* dVp = dv + ((long long)b * S + kv0) * qs + (long long)h * D;
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
      for (int t = 0; t < DT; ++t) {
        dKp[(long long)(m * 16 + lg * 4 + r) * qs + t * 16 + lc] =
            f32_to_bf16(acc_dk[m][t][r]);
        dVp[(long long)(m * 16 + lg * 4 + r) * qs + t * 16 + lc] =
            f32_to_bf16(acc_dv[m][t][r]);
      }
}

}  // namespace

extern "C" void acco_attn_bwd32_dq(const void*, const void*, const void*,
                                   const void*, const float*, const float*,
                                   void*, int, int, int, int, int, float,
                                   int, long long, long long, long long,
                                   long long, hipStream_t);
extern "C" void acco_attn_bwd32_dkv(const void*, const void*, const void*,
                                    const void*, const float*, const float*,
                                    void*, void*, int, int, int, int, int,
                                    float, int, long long, long long,
                                    long long, hipStream_t);

extern "C" {

void acco_attn_bwd_dq(const void* q, const void* k, const void* v,
                      const void* dO, const float* lse, const float* delta,
                      void* dq, int B, int S, int H, int Hkv, int D,
                      float scale, int window, hipStream_t stream) {
  if ((D == 64 || D == 128) && S % 256 == 0) {
    acco_attn_bwd32_dq(q, k, v, dO, lse, delta, dq, B, S, H, Hkv, D, scale,
                       window, (long long)H * D, (long long)Hkv * D,
                       (long long)H * D, (long long)H * D, stream);
    return;
  }
  const bool wide = (S % 128 == 0) && (D == 64);
  dim3 grid(S / (wide ? 128 : 64), B * H);
  const int lds = (D * LST + 2 * TILE * (D + 8) + 4 * 16 * LST) * sizeof(u16);
#define LQ(DD, QQ) hipLaunchKernelGGL((attn_bwd_dq_kernel<DD, QQ>), grid,     dim3(256), lds, stream, (const u16*)q, (const u16*)k, (const u16*)v,     (const u16*)dO, lse, delta, (u16*)dq, S, H, Hkv, scale, window)
  if (D == 64) { if (wide) LQ(64, 32); else LQ(64, 16); }
  else         LQ(128, 16);
#undef LQ
}

void acco_attn_bwd_dkv(const void* q, const void* k, const void* v,
                       const void* dO, const float* lse, const float* delta,
                       void* dk, void* dv, int B, int S, int H, int Hkv,
                       int D, float scale, int window, hipStream_t stream) {
  if ((D == 64 || D == 128) && S % 256 == 0) {
    acco_attn_bwd32_dkv(q, k, v, dO, lse, delta, dk, dv, B, S, H, Hkv, D,
                        scale, window, (long long)H * D, (long long)Hkv * D,
                        (long long)H * D, stream);
    return;
  }
  const bool wide = (S % 128 == 0) && (D == 64);
  dim3 grid(S / (wide ? 128 : 64), B * H);
  const int lds =
      (2 * D * LST + 2 * TILE * (D + 8) + 8 * 16 * LST) * sizeof(u16);
#define LK(DD, QQ) hipLaunchKernelGGL((attn_bwd_dkv_kernel<DD, QQ>), grid,     dim3(256), lds, stream, (const u16*)q, (const u16*)k, (const u16*)v,     (const u16*)dO, lse, delta, (u16*)dk, (u16*)dv, S, H, Hkv, scale, window)
  if (D == 64) { if (wide) LK(64, 32); else LK(64, 16); }
  else         LK(128, 16);
#undef LK
}

}  // extern "C"
