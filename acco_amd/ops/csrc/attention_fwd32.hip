// Flash attention forward v4 for gfx950 — 32×32 MFMA structure with fully
// in-register softmax (guide §B "8-warp 32×32 ladder" + T12).
//
// Differences vs attention_fwd.hip (the 16×16 v3, kept as fallback):
// - mfma_f32_32x32x16_bf16: the swapped QK^T puts a q row's 64 scores in
//   TWO half-waves (col = lane&31, row = (reg&3)+8*(reg>>2)+4*(lane>>5)),
//   so the softmax row reduce is 31 in-lane ops + ONE shfl_xor(32);
// - P never touches LDS: the C-layout → A-fragment redistribution is two
//   v_permlane32_swap per 16-kv K-step (pairs su0↔su2, su1↔su3 — see the
//   pair-index derivation in the comment at pv_afrag below);
// - 8 waves per workgroup (256 q rows): K/V staged once per kv tile for
//   all 8 waves; fully-causal-masked tiles skip compute per wave (barriers
//   kept);
// - D = 64 and S % 256 == 0 only (the dispatch falls back to v3 else).

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int KT = 64;             // kv tile rows
constexpr int PAD = 8;
constexpr int LST = KT + PAD;      // 72
constexpr int QW = 32;             // q rows per wave
constexpr int NW = 8;              // waves per workgroup
constexpr int QT = QW * NW;        // 256 q rows per workgroup

#define MFMA32(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((a), (b), (c), 0, 0, 0)

ACCO_DEV unsigned pack_bf16(float lo, float hi) {
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

// D=64 is capped at 128 VGPR: 2 co-resident 8-wave blocks per CU (4
// waves/SIMD) out-weigh scheduling freedom — measured faster than the
// uncapped 159-VGPR build and than a 204-VGPR T14 double-buffered variant.
template <int D>
__global__ __launch_bounds__(512, D == 64 ? 4 : 2)
void attn_fwd32_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                       const u16* __restrict__ v, u16* __restrict__ o,
                       float* __restrict__ lse,        // [B, H, S]
                       int S, int H, int Hkv, float scale, int window,
                       long long q_rs, long long kv_rs, long long o_rs) {
  constexpr int KS = D / 16;       // QK^T K-steps over head dim (16 each)
  constexpr int DT = D / 32;       // 32-wide d tiles of the output
  const int qt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lq = lane & 31;        // q col (swapped QK^T) / d col (PV out)
  const int hi = lane >> 5;        // half-wave

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* k_lds = smem;                        // [KT][D+8]
  u16* v_lds = smem + KT * (D + 8);         // V^T: [D][LST]
  constexpr int KROW = D + 8;

  const int q0 = qt * QT + wave * QW;
  const long long qs = q_rs;           // token-row strides (packed-qkv aware)
  const long long ks = kv_rs;
  const u16* Qp = q + ((long long)b * S + q0) * qs + (long long)h * D;
  const u16* Kb = k + (long long)b * S * ks + (long long)hkv * D;
  const u16* Vb = v + (long long)b * S * ks + (long long)hkv * D;

  // Q as B operand of the swapped QK^T: lane holds Q[q=lq][d=hi*8+i+16s]
  short8 qf[KS];
#pragma unroll
  for (int s = 0; s < KS; ++s)
    qf[s] = *reinterpret_cast<const short8*>(
        Qp + (long long)lq * qs + s * 16 + hi * 8);

  float m_c = -1e30f, l_c = 0.0f;      // per q col = lq (dup over hi)
  f32x16 acc_o[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc_o[t][r] = 0.0f;

  int j_lo = 0;
  if (window > 0) {
    int kv_min = qt * QT - window + 1;
    if (kv_min > 0) j_lo = kv_min / KT;
  }
  const int j_hi = (qt * QT + QT - 1) / KT;
  const int q_wave_max = q0 + QW - 1;
  const float scale2 = scale * 1.4426950408889634f;   // log2 domain
  constexpr float THR2 = 8.0f * 1.4426950408889634f;  // defer-max (T13)

  for (int j = j_lo; j <= j_hi; ++j) {
    // ---- stage K (row-major copy) + V (transposed) for all 8 waves
    __syncthreads();
    {
      const u16* Kt = Kb + (long long)(j * KT) * ks;
      for (int c = threadIdx.x; c < KT * (D / 8); c += 512) {
        const int kv = c / (D / 8), dc = c % (D / 8);
        reinterpret_cast<uint4*>(k_lds + kv * KROW)[dc] =
            *reinterpret_cast<const uint4*>(Kt + (long long)kv * ks + dc * 8);
      }
      const u16* Vt = Vb + (long long)(j * KT) * ks;
      const int kv2 = (threadIdx.x & 31) * 2;
      for (int dg = threadIdx.x >> 5; dg < D / 8; dg += 16) {
        ushort4 a0 = reinterpret_cast<const ushort4*>(
            Vt + (long long)kv2 * ks + dg * 8)[0];
        ushort4 a1 = reinterpret_cast<const ushort4*>(
            Vt + (long long)kv2 * ks + dg * 8)[1];
        ushort4 b0 = reinterpret_cast<const ushort4*>(
            Vt + (long long)(kv2 + 1) * ks + dg * 8)[0];
        ushort4 b1 = reinterpret_cast<const ushort4*>(
            Vt + (long long)(kv2 + 1) * ks + dg * 8)[1];
        u16 av[8] = {a0.x, a0.y, a0.z, a0.w, a1.x, a1.y, a1.z, a1.w};
        u16 bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
#pragma unroll
        for (int i = 0; i < 8; ++i)
          *reinterpret_cast<ushort2*>(v_lds + (dg * 8 + i) * LST + kv2) =
              make_ushort2(av[i], bv[i]);
      }
    }
    __syncthreads();

    // fully-masked tile for this wave (kv all future, or all outside the
    // local window): skip compute
    if (j * KT > q_wave_max ||
        (window > 0 && j * KT + KT - 1 <= q0 - window))
      continue;

    // ---- S^T: st[m32] per 32-kv sub-tile (C: col=q=lq, row=kv spread)
    f32x16 st[2];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m32 = 0; m32 < 2; ++m32) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.0f;
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        short8 kf = *reinterpret_cast<const short8*>(
            k_lds + (m32 * 32 + lq) * KROW + s * 16 + hi * 8);
        acc = MFMA32(kf, qf[s], acc);
      }
      st[m32] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + online softmax (per q col = lq), log2 domain: scores are
    // s·scale·log2e and raw v_exp_f32 (= 2^x) replaces mul+exp per element.
    // WAVE-level mask class: interior tiles (all kv ≤ all q, all in window)
    // skip the per-element predicates entirely.
    const int q_g = q0 + lq;
    const bool need_mask = (j * KT + KT - 1 > q0) ||
                           (window > 0 && j * KT <= q_wave_max - window);
    float p[32];
    float tmax = -1e30f;
#pragma unroll
    for (int m32 = 0; m32 < 2; ++m32)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        // row = (r&3) + 8*(r>>2) + 4*hi within the 32-kv sub-tile
        const int kv_g = j * KT + m32 * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float x = st[m32][r] * scale2;
        if (need_mask) {
          bool valid = (kv_g <= q_g);
          if (window > 0) valid = valid && (kv_g > q_g - window);
          x = valid ? x : -1e30f;
        }
        p[m32 * 16 + r] = x;
        tmax = fmaxf(tmax, x);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));

    // ---- T13 defer-max: skip the O-rescale pass (16 shfl + 32 mul) while
    // the wave's max grows ≤ 8 natural units (P bounded by e^8, fine in
    // f32 accumulation). Decision precedes this tile's exponentials and
    // the previous tile's PV is complete — the safe textbook order.
    float m_new = m_c;
    if (!__all(tmax <= m_c + THR2)) {
      m_new = fmaxf(m_c, tmax);
      const float alpha = __builtin_amdgcn_exp2f(m_c - m_new);
      l_c *= alpha;
      float alpha_row[16];
#pragma unroll
      for (int r = 0; r < 16; ++r)
        alpha_row[r] = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
      for (int t = 0; t < DT; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_o[t][r] *= alpha_row[r];
      m_c = m_new;
    }
    float rsum = 0.0f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      const float e =
          (p[i] > -9e29f) ? __builtin_amdgcn_exp2f(p[i] - m_new) : 0.0f;
      p[i] = e;
      rsum += e;
    }
    rsum += __shfl_xor(rsum, 32, 64);
    l_c += rsum;

    // ---- P (C layout) → A fragments in-register, then PV
    // Per 16-kv K-step kk the A fragment's 4 dwords are kv pairs
    // (4*hi + j), j=0..3 (pair = kv/2 within the 16). The lane's own
    // packed dwords su_j = pack(p[8kk+2j], p[8kk+2j+1]) carry pairs
    // {2hi, 2hi+1, 4+2hi, 5+2hi}; swapping su0↔su2 and su1↔su3 across
    // half-waves (v_permlane32_swap) yields exactly pairs 4hi+{0,2} and
    // 4hi+{1,3}.  (m32 sub-tile kk = 0,1 within each 32-kv block.)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m32 = 0; m32 < 2; ++m32) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const float* pp = p + m32 * 16 + kk * 8;
        unsigned su0 = pack_bf16(pp[0], pp[1]);
        unsigned su1 = pack_bf16(pp[2], pp[3]);
        unsigned su2 = pack_bf16(pp[4], pp[5]);
        unsigned su3 = pack_bf16(pp[6], pp[7]);
        auto r02 = __builtin_amdgcn_permlane32_swap(su0, su2, false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(su1, su3, false, false);
        unsigned a0 = r02[0], a2 = r02[1];
        unsigned a1 = r13[0], a3 = r13[1];
        short8 pa;
        *reinterpret_cast<unsigned*>(&pa) = a0;
        reinterpret_cast<unsigned*>(&pa)[1] = a1;
        reinterpret_cast<unsigned*>(&pa)[2] = a2;
        reinterpret_cast<unsigned*>(&pa)[3] = a3;
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 vb = *reinterpret_cast<const short8*>(
              v_lds + (t * 32 + lq) * LST + m32 * 32 + kk * 16 + hi * 8);
          acc_o[t] = MFMA32(pa, vb, acc_o[t]);
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // ---- epilogue: O rows q = (r&3)+8*(r>>2)+4*hi, col d = t*32 + lq
  float l_row[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    l_row[r] = __shfl(l_c, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
  u16* Op = o + ((long long)b * S + q0) * o_rs + (long long)h * D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const float inv_l = (l_row[r] > 0.f) ? 1.0f / l_row[r] : 0.0f;
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
#pragma unroll
    for (int t = 0; t < DT; ++t)
      Op[(long long)qrow * o_rs + t * 32 + lq] =
          f32_to_bf16(acc_o[t][r] * inv_l);
  }
  // lse stays in NATURAL units (the backward consumes exp(s·scale − lse));
  // the running m_c/l_c are log2-domain, so lse = ln2·(m2 + log2 l)
  if (hi == 0)
    lse[((long long)bh) * S + q0 + lq] =
        (m_c + __builtin_amdgcn_logf(fmaxf(l_c, 1e-30f))) *
        0.6931471805599453f;
}

}  // namespace

extern "C" void acco_attn_fwd32(const void* q, const void* k, const void* v,
                                void* o, float* lse, int B, int S, int H,
                                int Hkv, int D, float scale, int window,
                                long long q_rs, long long kv_rs,
                                long long o_rs, hipStream_t stream) {
  dim3 grid(S / QT, B * H);
  const int lds = (KT * (D + 8) + D * LST) * sizeof(u16);
  if (D == 64)
    hipLaunchKernelGGL(attn_fwd32_kernel<64>, grid, dim3(512), lds, stream,
                       (const u16*)q, (const u16*)k, (const u16*)v, (u16*)o,
                       lse, S, H, Hkv, scale, window, q_rs, kv_rs, o_rs);
  else
    hipLaunchKernelGGL(attn_fwd32_kernel<128>, grid, dim3(512), lds, stream,
                       (const u16*)q, (const u16*)k, (const u16*)v, (u16*)o,
                       lse, S, H, Hkv, scale, window, q_rs, kv_rs, o_rs);
}
