#include "hip/hip_runtime.h"
// Flash-style causal attention FORWARD for gfx950 (CDNA4 MFMA).
//
// Replaces the reference's implicit HF eager attention,
// which materializes the S×S fp32 score matrix; here scores never leave
// registers/LDS (online softmax, running m/l per q row).
//
// v3 structure:
// - layout [B, S, H, D] bf16, D ∈ {64, 128};
// - workgroup = 4 waves; each wave owns QW q rows (QW=32 when S%128==0,
//   else 16) as QW/16 MFMA sub-tiles — doubling the MFMA work amortized
//   over each K/V staging + barrier pair was worth ~2× on the v1 shape;
// - swapped QK^T (guide §B: mfma(K, Q) puts a softmax row across a
//   16-lane group: in-lane max/sum over 16 + two shfl_xor hops);
// - K tile staged row-major in LDS (vector copy, shared by 4 waves);
//   V tile staged TRANSPOSED (paired-kv ushort2 writes) for the PV
//   B-fragments; both padded to a 144 B row stride (16 rows hit 16
//   distinct bank groups for ds_read_b128 — no conflicts);
// - P redistributed q-col→A-fragment layout through a per-wave LDS tile;
// - causal + optional local window (GPT-Neo 256) + GQA; saves lse.

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int KT = 64;           // kv rows per tile
constexpr int VPAD = 8;
constexpr int LSTRIDE = KT + VPAD;   // 72

#define MFMA(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

template <int D, int QW>         // QW = q rows per wave (16 or 32)
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                     const u16* __restrict__ v, u16* __restrict__ o,
                     float* __restrict__ lse,          // [B, H, S]
                     int S, int H, int Hkv, float scale, int window) {
  constexpr int KS = D / 32;     // MFMA K-steps over head dim
  constexpr int DT = D / 16;     // d tiles of the output
  constexpr int M2 = QW / 16;    // q sub-tiles per wave
  constexpr int QT = 4 * QW;     // q rows per workgroup
  constexpr int KROW = D + 8;
  const int qt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lc = lane & 15;

  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* v_lds = smem;                                   // [D][LSTRIDE]
  u16* k_lds = smem + D * LSTRIDE;                     // [KT][KROW]
  u16* p_lds = k_lds + KT * KROW + wave * 16 * LSTRIDE;  // per wave [16][LSTRIDE]

  const int q0 = qt * QT + wave * QW;
  const long long qs = (long long)H * D;
  const long long ks = (long long)Hkv * D;
  const u16* Qp = q + ((long long)b * S + q0) * qs + (long long)h * D;
  const u16* Kb = k + (long long)b * S * ks + (long long)hkv * D;
  const u16* Vb = v + (long long)b * S * ks + (long long)hkv * D;

  // Q fragments per sub-tile: lane holds Q[q = m*16+lc][d = s*32 + lg*8 + i]
  short8 qf[M2][KS];
#pragma unroll
  for (int m = 0; m < M2; ++m)
#pragma unroll
    for (int s = 0; s < KS; ++s)
      qf[m][s] = *reinterpret_cast<const short8*>(
          Qp + (long long)(m * 16 + lc) * qs + s * 32 + lg * 8);

  float m_c[M2], l_c[M2];
  f32x4 acc_o[M2][DT];
#pragma unroll
  for (int m = 0; m < M2; ++m) {
    m_c[m] = -1e30f;
    l_c[m] = 0.0f;
#pragma unroll
    for (int t = 0; t < DT; ++t) acc_o[m][t] = {0.f, 0.f, 0.f, 0.f};
  }

  int j_lo = 0;
  if (window > 0) {
    int kv_min = qt * QT - window + 1;
    if (kv_min > 0) j_lo = kv_min / KT;
  }
  const int j_hi = (qt * QT + QT - 1) / KT;            // causal

  for (int j = j_lo; j <= j_hi; ++j) {
    // ---- stage V^T (paired-kv ushort2) and K (vector copy); 4 waves share
    __syncthreads();
    {
      const u16* Vt = Vb + (long long)(j * KT) * ks;
      const int kv2 = (threadIdx.x & 31) * 2;
      for (int dg = threadIdx.x >> 5; dg < D / 8; dg += 8) {
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
          *reinterpret_cast<ushort2*>(v_lds + (dg * 8 + i) * LSTRIDE + kv2) =
              make_ushort2(av[i], bv[i]);
      }
      const u16* Kt = Kb + (long long)(j * KT) * ks;
      for (int c = threadIdx.x; c < KT * (D / 8); c += 256) {
        const int kv = c / (D / 8), dc = c % (D / 8);
        reinterpret_cast<uint4*>(k_lds + kv * KROW)[dc] =
            *reinterpret_cast<const uint4*>(Kt + (long long)kv * ks + dc * 8);
      }
    }
    __syncthreads();

    const bool diag = (j * KT + KT - 1) > (qt * QT);   // any masking possible

#pragma unroll
    for (int m = 0; m < M2; ++m) {
      // ---- S^T: st[m16] = K_sub · Q^T (C: col=q=lc, row=kv spread)
      f32x4 st[4];
      __builtin_amdgcn_s_setprio(1);   // guide T5: favor the MFMA cluster
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int s = 0; s < KS; ++s) {
          short8 kf = *reinterpret_cast<const short8*>(
              k_lds + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
          acc = MFMA(kf, qf[m][s], acc);
        }
        st[m16] = acc;
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- mask + online softmax (per col q=lc)
      const int q_g = q0 + m * 16 + lc;
      float p[16];
      float tmax = -1e30f;
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kv_g = j * KT + m16 * 16 + lg * 4 + r;
          float x = st[m16][r] * scale;
          if (diag || window > 0) {
            bool valid = (kv_g <= q_g);
            if (window > 0) valid = valid && (kv_g > q_g - window);
            x = valid ? x : -1e30f;
          }
          p[m16 * 4 + r] = x;
          tmax = fmaxf(tmax, x);
        }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
      const float m_new = fmaxf(m_c[m], tmax);
      const float alpha = __expf(m_c[m] - m_new);
      float rsum = 0.0f;
#pragma unroll
      for (int i = 0; i < 16; ++i) {
        float e = (p[i] > -9e29f) ? __expf(p[i] - m_new) : 0.0f;
        p[i] = e;
        rsum += e;
      }
      rsum += __shfl_xor(rsum, 16, 64);
      rsum += __shfl_xor(rsum, 32, 64);
      l_c[m] = l_c[m] * alpha + rsum;
      m_c[m] = m_new;

      // ---- P → per-wave LDS (row q=lc, col kv = m16*16 + lg*4 + r)
#pragma unroll
      for (int m16 = 0; m16 < 4; ++m16) {
        u16 pk[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) pk[r] = f32_to_bf16(p[m16 * 4 + r]);
        *reinterpret_cast<ushort4*>(p_lds + lc * LSTRIDE + m16 * 16 + lg * 4) =
            make_ushort4(pk[0], pk[1], pk[2], pk[3]);
      }

      // ---- rescale O by alpha (row layout q = lg*4 + r)
      float alpha_row[4];
#pragma unroll
      for (int r = 0; r < 4; ++r)
        alpha_row[r] = __shfl(alpha, lg * 4 + r, 64);
#pragma unroll
      for (int t = 0; t < DT; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r) acc_o[m][t][r] *= alpha_row[r];

      // (plain LDS accesses: hipcc's counted lgkmcnt orders the P
      // write->read pair; no full-counter drain needed)
      // ---- PV: A = P[q=lc][kv], B = V^T from LDS
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        short8 pa = *reinterpret_cast<const short8*>(
            p_lds + lc * LSTRIDE + s * 32 + lg * 8);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
          short8 vb = *reinterpret_cast<const short8*>(
              v_lds + (t * 16 + lc) * LSTRIDE + s * 32 + lg * 8);
          acc_o[m][t] = MFMA(pa, vb, acc_o[m][t]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  // ---- epilogue: O rows q = m*16 + lg*4 + r, col d = t*16 + lc
  u16* Op = o + ((long long)b * S + q0) * qs + (long long)h * D;
#pragma unroll
  for (int m = 0; m < M2; ++m) {
    float l_row[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      l_row[r] = __shfl(l_c[m], lg * 4 + r, 64);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float inv_l = (l_row[r] > 0.f) ? 1.0f / l_row[r] : 0.0f;
#pragma unroll
      for (int t = 0; t < DT; ++t)
        Op[(long long)(m * 16 + lg * 4 + r) * qs + t * 16 + lc] =
            f32_to_bf16(acc_o[m][t][r] * inv_l);
    }
    if (lg == 0)
      lse[((long long)bh) * S + q0 + m * 16 + lc] =
          m_c[m] + __logf(fmaxf(l_c[m], 1e-30f));
  }
}

}  // namespace

extern "C" void acco_attn_fwd32(const void* q, const void* k, const void* v,
                                void* o, float* lse, int B, int S, int H,
                                int Hkv, int D, float scale, int window,
                                long long q_rs, long long kv_rs,
                                long long o_rs, hipStream_t stream);

extern "C" void acco_attn_fwd(const void* q, const void* k, const void* v,
                              void* o, float* lse, int B, int S, int H,
                              int Hkv, int D, float scale, int window,
                              hipStream_t stream) {
  if ((D == 64 || D == 128) && S % 256 == 0) {
    // v4: 32x32 MFMA + in-register softmax (attention_fwd32.hip)
    acco_attn_fwd32(q, k, v, o, lse, B, S, H, Hkv, D, scale, window,
                    (long long)H * D, (long long)Hkv * D, (long long)H * D,
                    stream);
    return;
  }
  const int lds_bytes =
      (D * LSTRIDE + KT * (D + 8) + 4 * 16 * LSTRIDE) * sizeof(u16);
  // D=128 at QW=32 needs ~197 VGPR -> 1 wave/SIMD; keep QW=16 there
  const bool wide = (S % 128 == 0) && (D == 64);
  const int qt_rows = wide ? 128 : 64;
  dim3 grid(S / qt_rows, B * H);
#define LA(DD, QQ) hipLaunchKernelGGL((attn_fwd_kernel<DD, QQ>), grid, \
    dim3(256), lds_bytes, stream, (const u16*)q, (const u16*)k, \
    (const u16*)v, (u16*)o, lse, S, H, Hkv, scale, window)
  if (D == 64) { if (wide) LA(64, 32); else LA(64, 16); }
  else         LA(128, 16);
#undef LA
}
