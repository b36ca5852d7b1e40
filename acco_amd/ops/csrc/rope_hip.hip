#include "hip/hip_runtime.h"
// Rotary position embedding (HF-Llama rotate_half convention) fwd + bwd for
// gfx950. Operates on the projections' natural [B, S, H, D] contiguous
// layout (no transpose copies); cos/sin are HOST-precomputed fp32 tables
// [S, D] (guide Appendix B: no on-device sinf/cosf on the hot path).
// Forward:  y1 = x1*cos - x2*sin ; y2 = x2*cos + x1*sin   (x2 = x[d+D/2])
// Backward: rotation by -theta:  dx1 = dy1*cos + dy2*sin ;
//           dx2 = dy2*cos - dy1*sin.

#include "common.h"

namespace {

using u16 = unsigned short;

template <bool BWD>
__global__ void rope_kernel(const u16* __restrict__ x, u16* __restrict__ y,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            long long total_q,   // B*S*H*(D/8)
                            int H, int D, int S,
                            long long row_stride) {  // elements per token row
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const int qph = D / 8;                 // 4-pair quads per half-head
  for (long long i = i0; i < total_q; i += stride) {
    const int q = (int)(i % qph);        // which 4-pair group in the head
    long long th = i / qph;              // token*H + h
    const long long tok = th / H;
    const int h = (int)(th % H);
    const int s = (int)(tok % S);
    const long long base = tok * row_stride + (long long)h * D + q * 4;
    const long long base2 = base + D / 2;           // x2 offset
    ushort4 x1 = *reinterpret_cast<const ushort4*>(x + base);
    ushort4 x2 = *reinterpret_cast<const ushort4*>(x + base2);
    float4 c = *reinterpret_cast<const float4*>(cos_t + (long long)s * D + q * 4);
    float4 sn = *reinterpret_cast<const float4*>(sin_t + (long long)s * D + q * 4);
    float a[4] = {bf16_to_f32(x1.x), bf16_to_f32(x1.y), bf16_to_f32(x1.z),
                  bf16_to_f32(x1.w)};
    float b[4] = {bf16_to_f32(x2.x), bf16_to_f32(x2.y), bf16_to_f32(x2.z),
                  bf16_to_f32(x2.w)};
    float cc[4] = {c.x, c.y, c.z, c.w};
    float ss[4] = {sn.x, sn.y, sn.z, sn.w};
    u16 o1[4], o2[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float y1, y2;
      if constexpr (!BWD) {
        y1 = a[k] * cc[k] - b[k] * ss[k];
        y2 = b[k] * cc[k] + a[k] * ss[k];
      } else {
        y1 = a[k] * cc[k] + b[k] * ss[k];
        y2 = b[k] * cc[k] - a[k] * ss[k];
      }
      o1[k] = f32_to_bf16(y1);
      o2[k] = f32_to_bf16(y2);
    }
    *reinterpret_cast<ushort4*>(y + base) = make_ushort4(o1[0], o1[1], o1[2], o1[3]);
    *reinterpret_cast<ushort4*>(y + base2) = make_ushort4(o2[0], o2[1], o2[2], o2[3]);
  }
}

}  // namespace

extern "C" void acco_rope(const void* x, void* y, const float* cos_t,
                          const float* sin_t, long long B, int S, int H,
                          int D, bool bwd, long long row_stride,
                          hipStream_t stream) {
  const long long total_q = B * (long long)S * H * (D / 8);
  const int grid = elementwise_grid(total_q, 256);
  if (bwd)
    hipLaunchKernelGGL(rope_kernel<true>, dim3(grid), dim3(256), 0, stream,
                       (const u16*)x, (u16*)y, cos_t, sin_t, total_q, H, D, S,
                       row_stride);
  else
    hipLaunchKernelGGL(rope_kernel<false>, dim3(grid), dim3(256), 0, stream,
                       (const u16*)x, (u16*)y, cos_t, sin_t, total_q, H, D, S,
                       row_stride);
}
