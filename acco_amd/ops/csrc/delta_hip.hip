#include "hip/hip_runtime.h"
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

}  // namespace

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
