// Attention-backward Delta preprocessing for gfx950:
//   delta[b, h, s] = sum_d dO[b,s,h,:] * O[b,s,h,:]   (fp32 out, [B,H,S])
// One wave per (token, head) row; bf16x8 loads; replaces a 4-kernel ATen
// chain (two bf16->f32 casts, a multiply, a reduce) in the flash-attention
// backward wrapper.

#include "common.h"

namespace {

using u16 = unsigned short;

// LPR = lanes per row = D/8 (power of two). A wave covers 64/LPR rows, so
// every lane issues one 16-byte load per input — full coalescing even at
// D=64 (the old one-wave-per-row layout left 56/64 lanes idle there).
template <int LPR>
__global__ void attn_delta_kernel(const u16* __restrict__ dO,
                                  const u16* __restrict__ O,
                                  float* __restrict__ delta,
                                  long long TH,   // B*S*H rows
                                  int S, int H, int D) {
  constexpr int RPW = 64 / LPR;                 // rows per wave
  const int lane = threadIdx.x & 63;
  const int sub = lane / LPR;                   // row slot within the wave
  const int cl = (lane % LPR) * 8;              // column of this lane
  const long long row0 =
      ((long long)blockIdx.x * 4 + (threadIdx.x >> 6)) * RPW + sub;
  const long long stride = (long long)gridDim.x * 4 * RPW;
  for (long long row = row0; row < TH; row += stride) {
    const u16* a = dO + row * D + cl;
    const u16* b = O + row * D + cl;
    ushort4 a0 = reinterpret_cast<const ushort4*>(a)[0];
    ushort4 a1 = reinterpret_cast<const ushort4*>(a)[1];
    ushort4 b0 = reinterpret_cast<const ushort4*>(b)[0];
    ushort4 b1 = reinterpret_cast<const ushort4*>(b)[1];
    float acc = bf16_to_f32(a0.x) * bf16_to_f32(b0.x)
              + bf16_to_f32(a0.y) * bf16_to_f32(b0.y)
              + bf16_to_f32(a0.z) * bf16_to_f32(b0.z)
              + bf16_to_f32(a0.w) * bf16_to_f32(b0.w)
              + bf16_to_f32(a1.x) * bf16_to_f32(b1.x)
              + bf16_to_f32(a1.y) * bf16_to_f32(b1.y)
              + bf16_to_f32(a1.z) * bf16_to_f32(b1.z)
              + bf16_to_f32(a1.w) * bf16_to_f32(b1.w);
    for (int off = LPR / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, 64);
    if (cl == 0) {
      // row = (b*S + s)*H + h  →  delta index (b*H + h)*S + s
      const long long bs = row / H;
      const int h = (int)(row % H);
      const long long bidx = bs / S;
      const int s = (int)(bs % S);
      delta[(bidx * H + h) * (long long)S + s] = acc;
    }
  }
}

// generic fallback for D not in {64, 128}: one wave per row
__global__ void attn_delta_kernel_any(const u16* __restrict__ dO,
                                      const u16* __restrict__ O,
                                      float* __restrict__ delta,
                                      long long TH, int S, int H, int D) {
  const long long row0 = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const long long stride = (long long)gridDim.x * 4;
  for (long long row = row0; row < TH; row += stride) {
    const u16* a = dO + row * D;
    const u16* b = O + row * D;
    float acc = 0.0f;
    for (int c = lane * 8; c < D; c += 64 * 8) {
      ushort4 a0 = reinterpret_cast<const ushort4*>(a + c)[0];
      ushort4 a1 = reinterpret_cast<const ushort4*>(a + c)[1];
      ushort4 b0 = reinterpret_cast<const ushort4*>(b + c)[0];
      ushort4 b1 = reinterpret_cast<const ushort4*>(b + c)[1];
      acc += bf16_to_f32(a0.x) * bf16_to_f32(b0.x)
           + bf16_to_f32(a0.y) * bf16_to_f32(b0.y)
           + bf16_to_f32(a0.z) * bf16_to_f32(b0.z)
           + bf16_to_f32(a0.w) * bf16_to_f32(b0.w)
           + bf16_to_f32(a1.x) * bf16_to_f32(b1.x)
           + bf16_to_f32(a1.y) * bf16_to_f32(b1.y)
           + bf16_to_f32(a1.z) * bf16_to_f32(b1.z)
           + bf16_to_f32(a1.w) * bf16_to_f32(b1.w);
    }
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, 64);
    if (lane == 0) {
      const long long bs = row / H;
      const int h = (int)(row % H);
      const long long bidx = bs / S;
      const int s = (int)(bs % S);
      delta[(bidx * H + h) * (long long)S + s] = acc;
    }
  }
}

// GQA group-reduce of the per-query-head dK/dV into the packed grad's
// k|v sections in ONE pass:
//   dqkv[b, s, k_off + hkv·D + d] = bf16(Σ_r dkq[b, s, (hkv·rep + r)·D + d])
// (likewise dvq → v_off). Replaces the ATen chain the wrapper ran per
// layer: two dim-3 fp32 sums + two bf16 casts + two strided copies
// (~47 µs → one roofline pass).
__global__ void gqa_reduce_kernel(const u16* __restrict__ dkq,
                                  const u16* __restrict__ dvq,
                                  u16* __restrict__ dqkv,
                                  long long T,       // B*S token rows
                                  int Hkv, int rep, int D,
                                  long long W,       // dqkv row stride
                                  long long k_off, long long v_off) {
  const int kvD = Hkv * D;                    // elems per token per section
  const long long total = T * (long long)kvD / 8;   // vec8 work items
  for (long long it = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       it < total; it += (long long)gridDim.x * blockDim.x) {
    const long long t = it / (kvD / 8);
    const int c8 = (int)(it % (kvD / 8)) * 8;       // elem within section
    const int hkv = c8 / D, d = c8 % D;
    const long long in_base = (t * (long long)Hkv * rep + hkv * rep) * D + d;
    float ak[8] = {0, 0, 0, 0, 0, 0, 0, 0}, av[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < rep; ++r) {
      const u16* pk = dkq + in_base + (long long)r * D;
      const u16* pv = dvq + in_base + (long long)r * D;
      ushort4 k0 = reinterpret_cast<const ushort4*>(pk)[0];
      ushort4 k1 = reinterpret_cast<const ushort4*>(pk)[1];
      ushort4 v0 = reinterpret_cast<const ushort4*>(pv)[0];
      ushort4 v1 = reinterpret_cast<const ushort4*>(pv)[1];
      ak[0] += bf16_to_f32(k0.x); ak[1] += bf16_to_f32(k0.y);
      ak[2] += bf16_to_f32(k0.z); ak[3] += bf16_to_f32(k0.w);
      ak[4] += bf16_to_f32(k1.x); ak[5] += bf16_to_f32(k1.y);
      ak[6] += bf16_to_f32(k1.z); ak[7] += bf16_to_f32(k1.w);
      av[0] += bf16_to_f32(v0.x); av[1] += bf16_to_f32(v0.y);
      av[2] += bf16_to_f32(v0.z); av[3] += bf16_to_f32(v0.w);
      av[4] += bf16_to_f32(v1.x); av[5] += bf16_to_f32(v1.y);
      av[6] += bf16_to_f32(v1.z); av[7] += bf16_to_f32(v1.w);
    }
    u16* ok = dqkv + t * W + k_off + c8;
    u16* ov = dqkv + t * W + v_off + c8;
    reinterpret_cast<ushort4*>(ok)[0] =
        make_ushort4(f32_to_bf16(ak[0]), f32_to_bf16(ak[1]),
                     f32_to_bf16(ak[2]), f32_to_bf16(ak[3]));
    reinterpret_cast<ushort4*>(ok)[1] =
        make_ushort4(f32_to_bf16(ak[4]), f32_to_bf16(ak[5]),
                     f32_to_bf16(ak[6]), f32_to_bf16(ak[7]));
    reinterpret_cast<ushort4*>(ov)[0] =
        make_ushort4(f32_to_bf16(av[0]), f32_to_bf16(av[1]),
                     f32_to_bf16(av[2]), f32_to_bf16(av[3]));
    reinterpret_cast<ushort4*>(ov)[1] =
        make_ushort4(f32_to_bf16(av[4]), f32_to_bf16(av[5]),
                     f32_to_bf16(av[6]), f32_to_bf16(av[7]));
  }
}

}  // namespace

extern "C" void acco_attn_gqa_reduce(const void* dkq, const void* dvq,
                                     void* dqkv, long long T, int Hkv,
                                     int rep, int D, long long W,
                                     long long k_off, long long v_off,
                                     hipStream_t stream) {
  const long long total = T * (long long)Hkv * D / 8;
  int grid = (int)((total + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(gqa_reduce_kernel, dim3(grid), dim3(256), 0, stream,
                     (const u16*)dkq, (const u16*)dvq, (u16*)dqkv, T, Hkv,
                     rep, D, W, k_off, v_off);
}

extern "C" void acco_attn_delta(const void* dO, const void* O, float* delta,
                                long long B, int S, int H, int D,
                                hipStream_t stream) {
  const long long TH = B * (long long)S * H;
  if (D == 64 || D == 128) {
    const int rpw = 64 / (D / 8);
    const long long waves = (TH + rpw - 1) / rpw;
    int grid = (int)(((waves + 3) / 4 < 4096) ? (waves + 3) / 4 : 4096);
    if (grid < 1) grid = 1;
    if (D == 64)
      hipLaunchKernelGGL(attn_delta_kernel<8>, dim3(grid), dim3(256), 0,
                         stream, (const u16*)dO, (const u16*)O, delta, TH, S,
                         H, D);
    else
      hipLaunchKernelGGL(attn_delta_kernel<16>, dim3(grid), dim3(256), 0,
                         stream, (const u16*)dO, (const u16*)O, delta, TH, S,
                         H, D);
    return;
  }
  int grid = (int)(((TH + 3) / 4 < 4096) ? (TH + 3) / 4 : 4096);
  hipLaunchKernelGGL(attn_delta_kernel_any, dim3(grid), dim3(256), 0, stream,
                     (const u16*)dO, (const u16*)O, delta, TH, S, H, D);
}
