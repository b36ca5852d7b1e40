#include "hip/hip_runtime.h"
// Flash-style causal attention FORWARD for gfx950 (CDNA4 MFMA).
//
// Replaces the reference's implicit HF eager attention,
// which materializes the S×S fp32 score matrix; here scores never leave
// registers/LDS (online softmax, running m/l per q row).
//
// Design (v1, correctness-first with the cheap CDNA4 idioms):
// - layout [B, S, H, D] bf16 (projections' natural layout), D ∈ {64, 128};
// - workgroup = 4 waves = one 64-row Q tile of one (b, h); each wave owns
//   16 q rows; grid = (S/64, B*H);
// - swapped QK^T (guide §B: compute mfma(K, Q) so the C fragment's col
//   index = q row → the whole softmax row lives across a 16-lane group:
//   in-lane max/sum over 16 + two shfl_xor hops, no LDS for the row);
// - mfma_f32_16x16x32_bf16 everywhere; A/B fragments are 16-byte
//   contiguous per-lane loads straight from global (K/Q) or LDS (P/V);
// - V tile staged TRANSPOSED in LDS once per kv tile, shared by the 4
//   waves (row stride padded 64→72 bf16: conflict-free ds_read_b128);
// - P redistributed q-col→A-fragment layout through a per-wave LDS tile
//   (the layout mismatch between MFMA C and A fragments — guide §B);
// - causal + optional local window (GPT-Neo 256) by masking the diagonal
//   tile and clamping the kv-tile loop;
// - GQA: kv head = h / (H / Hkv);
// - saves lse = m + log(l) per q row for the backward recompute.

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int QT = 64;           // q rows per workgroup
constexpr int KT = 64;           // kv rows per tile
constexpr int VPAD = 8;          // LDS row pad (bf16) → stride 72
constexpr int LSTRIDE = KT + VPAD;

#define MFMA(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

template <int D>
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const u16* __restrict__ q, const u16* __restrict__ k,
                     const u16* __restrict__ v, u16* __restrict__ o,
                     float* __restrict__ lse,          // [B, H, S]
                     int S, int H, int Hkv, float scale, int window) {
  constexpr int KS = D / 32;     // MFMA K-steps over head dim
  constexpr int DT = D / 16;     // d tiles of the output
  const int qt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hkv = h / (H / Hkv);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;      // lane group 0..3
  const int lc = lane & 15;      // col / row-in-16 index

  // LDS: V^T tile [D][72] + K tile [64][D+8] shared; P tiles per wave
  extern __shared__ __attribute__((aligned(16))) u16 smem[];
  u16* v_lds = smem;                                   // D * LSTRIDE
  u16* k_lds = smem + D * LSTRIDE;                     // KT * (D + 8)
  u16* p_lds = k_lds + KT * (D + 8) + wave * 16 * LSTRIDE;
  constexpr int KROW = D + 8;                          // K tile row stride

  const int q0 = qt * QT + wave * 16;                  // wave's first q row
  const long long qrow_stride = (long long)H * D;
  const long long krow_stride = (long long)Hkv * D;
  const u16* Qp = q + ((long long)b * S + q0) * qrow_stride + (long long)h * D;
  const u16* Kb = k + (long long)b * S * krow_stride + (long long)hkv * D;
  const u16* Vb = v + (long long)b * S * krow_stride + (long long)hkv * D;

  // Q fragments: lane holds Q[q=lc][d = s*32 + lg*8 + i]
  short8 qf[KS];
#pragma unroll
  for (int s = 0; s < KS; ++s)
    qf[s] = *reinterpret_cast<const short8*>(
        Qp + (long long)lc * qrow_stride + s * 32 + lg * 8);

  float m_c = -1e30f;            // running max for col q=lc (dup ×4 groups)
  float l_c = 0.0f;              // running denom
  f32x4 acc_o[DT];
#pragma unroll
  for (int t = 0; t < DT; ++t) acc_o[t] = {0.f, 0.f, 0.f, 0.f};

  const int q_max = qt * QT + QT - 1;
  int j_lo = 0;
  if (window > 0) {
    int kv_min = qt * QT - window + 1;
    if (kv_min > 0) j_lo = kv_min / KT;
  }
  const int j_hi = qt;           // causal

  for (int j = j_lo; j <= j_hi; ++j) {
    // ---- stage V^T (transposed, b32 paired-kv writes) and K (row-major
    // straight copy) cooperatively — shared by the 4 waves, so each K/V
    // element crosses HBM/L2 once instead of once per wave
    __syncthreads();             // previous tile's reads done
    {
      const u16* Vt = Vb + (long long)(j * KT) * krow_stride;
      // V: thread handles kv pair (2*(tid%32), +1), d-group of 8 = tid/32
      const int kv2 = (threadIdx.x & 31) * 2;
      for (int dg = threadIdx.x >> 5; dg < D / 8; dg += 8) {
        ushort4 a0 = reinterpret_cast<const ushort4*>(
            Vt + (long long)kv2 * krow_stride + dg * 8)[0];
        ushort4 a1 = reinterpret_cast<const ushort4*>(
            Vt + (long long)kv2 * krow_stride + dg * 8)[1];
        ushort4 b0 = reinterpret_cast<const ushort4*>(
            Vt + (long long)(kv2 + 1) * krow_stride + dg * 8)[0];
        ushort4 b1 = reinterpret_cast<const ushort4*>(
            Vt + (long long)(kv2 + 1) * krow_stride + dg * 8)[1];
        u16 av[8] = {a0.x, a0.y, a0.z, a0.w, a1.x, a1.y, a1.z, a1.w};
        u16 bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
#pragma unroll
        for (int i = 0; i < 8; ++i)
          *reinterpret_cast<ushort2*>(v_lds + (dg * 8 + i) * LSTRIDE + kv2) =
              make_ushort2(av[i], bv[i]);
      }
      // K: straight vector copy into [64][KROW]
      const u16* Kt = Kb + (long long)(j * KT) * krow_stride;
      for (int c = threadIdx.x; c < KT * (D / 8); c += 256) {
        const int kv = c / (D / 8), dc = c % (D / 8);
        reinterpret_cast<uint4*>(k_lds + kv * KROW)[dc] =
            *reinterpret_cast<const uint4*>(
                Kt + (long long)kv * krow_stride + dc * 8);
      }
    }
    __syncthreads();

    // ---- S^T tile: st[m16] = K_sub · Q^T  (C: col=q=lc, row=kv spread)
    f32x4 st[4];
#pragma unroll
    for (int m16 = 0; m16 < 4; ++m16) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int s = 0; s < KS; ++s) {
        short8 kf = *reinterpret_cast<const short8*>(
            k_lds + (m16 * 16 + lc) * KROW + s * 32 + lg * 8);
        acc = MFMA(kf, qf[s], acc);
      }
      st[m16] = acc;
    }

    // ---- mask + online softmax (per col q=lc)
    const int q_g = qt * QT + wave * 16 + lc;
    float p[16];
    float tmax = -1e30f;
#pragma unroll
    for (int m16 = 0; m16 < 4; ++m16)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kv_g = j * KT + m16 * 16 + lg * 4 + r;
        float x = st[m16][r] * scale;
        bool valid = (kv_g <= q_g);
        if (window > 0) valid = valid && (kv_g > q_g - window);
        x = valid ? x : -1e30f;
        p[m16 * 4 + r] = x;
        tmax = fmaxf(tmax, x);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_c, tmax);
    const float alpha = __expf(m_c - m_new);
    float rsum = 0.0f;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      float e = (p[i] > -9e29f) ? __expf(p[i] - m_new) : 0.0f;
      p[i] = e;
      rsum += e;
    }
    rsum += __shfl_xor(rsum, 16, 64);
    rsum += __shfl_xor(rsum, 32, 64);
    l_c = l_c * alpha + rsum;
    m_c = m_new;

    // ---- write P to per-wave LDS: row q=lc, col kv = m16*16 + lg*4 + r
#pragma unroll
    for (int m16 = 0; m16 < 4; ++m16) {
      u16 pk[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) pk[r] = f32_to_bf16(p[m16 * 4 + r]);
      *reinterpret_cast<ushort4*>(
          p_lds + lc * LSTRIDE + m16 * 16 + lg * 4) =
          make_ushort4(pk[0], pk[1], pk[2], pk[3]);
    }

    // ---- rescale O by alpha (row layout: q row = lg*4 + r)
    float alpha_row[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      alpha_row[r] = __shfl(alpha, lg * 4 + r, 64);
#pragma unroll
    for (int t = 0; t < DT; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[t][r] *= alpha_row[r];

    __builtin_amdgcn_s_waitcnt(0);   // lgkm: P writes visible to own wave

    // ---- PV: A = P (lane: P[q=lc][kv=s*32+lg*8+i]), B = V^T from LDS
#pragma unroll
    for (int s = 0; s < 2; ++s) {    // kv K-steps: 64/32
      short8 pa = *reinterpret_cast<const short8*>(
          p_lds + lc * LSTRIDE + s * 32 + lg * 8);
#pragma unroll
      for (int t = 0; t < DT; ++t) {
        short8 vb = *reinterpret_cast<const short8*>(
            v_lds + (t * 16 + lc) * LSTRIDE + s * 32 + lg * 8);
        acc_o[t] = MFMA(pa, vb, acc_o[t]);
      }
    }
  }

  // ---- epilogue: O rows q = lg*4 + r, col d = t*16 + lc
  float l_row[4], m_row[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    l_row[r] = __shfl(l_c, lg * 4 + r, 64);
    m_row[r] = __shfl(m_c, lg * 4 + r, 64);
  }
  u16* Op = o + ((long long)b * S + q0) * qrow_stride + (long long)h * D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv_l = (l_row[r] > 0.f) ? 1.0f / l_row[r] : 0.0f;
#pragma unroll
    for (int t = 0; t < DT; ++t)
      Op[(long long)(lg * 4 + r) * qrow_stride + t * 16 + lc] =
          f32_to_bf16(acc_o[t][r] * inv_l);
  }
  if (wave < 4 && lane < 16) {
    // one lane per q row writes lse (lane lc of group 0 covers row lc)
    if (lg == 0)
      lse[((long long)bh) * S + q0 + lc] = m_c + __logf(fmaxf(l_c, 1e-30f));
  }
  (void)q_max;
}

}  // namespace

extern "C" void acco_attn_fwd(const void* q, const void* k, const void* v,
                              void* o, float* lse, int B, int S, int H,
                              int Hkv, int D, float scale, int window,
                              hipStream_t stream) {
  dim3 grid(S / QT, B * H);
  const int lds_bytes =
      (D * LSTRIDE + KT * (D + 8) + 4 * 16 * LSTRIDE) * sizeof(u16);
  if (D == 64)
    hipLaunchKernelGGL(attn_fwd_kernel<64>, grid, dim3(256), lds_bytes,
                       stream, (const u16*)q, (const u16*)k, (const u16*)v,
                       (u16*)o, lse, S, H, Hkv, scale, window);
  else
    hipLaunchKernelGGL(attn_fwd_kernel<128>, grid, dim3(256), lds_bytes,
                       stream, (const u16*)q, (const u16*)k, (const u16*)v,
                       (u16*)o, lse, S, H, Hkv, scale, window);
}
