#include "hip/hip_runtime.h"
// Fused ZeRO-1 sharded AdamW for gfx950 — K3+K4+K5+K7 of SURVEY.md §2.5 in
// one pass over memory:
//   - reads the reduce-scattered gradient segment (bf16 or fp32) in place,
//   - scales by 1/global_grad_count (count read from a device int tensor,
//     so the comm thread never syncs to divide — reference
//     trainer_decoupled.py:97-98 does mul_(1/count) as a separate kernel),
//   - fp32 AdamW math identical to torch.optim.AdamW,
//   - writes updated bf16/fp32 params back into the com-buffer segment for
//     the all-gather,
//   - COMMIT=false is the ACCO tentative step (even com rounds): emits
//     updated params without touching p/m/v — replacing the reference's
//     snapshot->step->rollback (trainer_decoupled.py:79-84,113-125) at zero
//     state-traffic cost.
//
// Memory-bound: 28 B/element (commit) — HBM3E-roofline limited; vectorized
// float4 / bf16x4 accesses per guide Guideline 13.

#include "common.h"

namespace {

template <typename Tbuf, bool COMMIT>
__global__ void fused_adamw_kernel(
    float* __restrict__ p, const Tbuf* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    Tbuf* __restrict__ out,
    const float* __restrict__ scale_dev,  // nullptr -> use scale directly
    long long n,
    float scale, float lr, float beta1, float beta2, float eps,
    float wd_factor,                 // (1 - lr*weight_decay)
    float inv_bc1, float inv_bc2_sqrt)
{
  const float s = (scale_dev != nullptr) ? (scale * *scale_dev) : scale;
  const long long i0 = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const long long n4 = n >> 2;

  for (long long i = i0; i < n4; i += stride) {
    float4 pf = reinterpret_cast<const float4*>(p)[i];
    float4 mf = reinterpret_cast<const float4*>(m)[i];
    float4 vf = reinterpret_cast<const float4*>(v)[i];
    float gf[4];
    if constexpr (sizeof(Tbuf) == 2) {
      ushort4 gu = reinterpret_cast<const ushort4*>(g)[i];
      gf[0] = bf16_to_f32(gu.x); gf[1] = bf16_to_f32(gu.y);
      gf[2] = bf16_to_f32(gu.z); gf[3] = bf16_to_f32(gu.w);
    } else {
      float4 gv = reinterpret_cast<const float4*>(g)[i];
      gf[0] = gv.x; gf[1] = gv.y; gf[2] = gv.z; gf[3] = gv.w;
    }
    float po[4], mo[4], vo[4];
    float pp[4] = {pf.x, pf.y, pf.z, pf.w};
    float mm[4] = {mf.x, mf.y, mf.z, mf.w};
    float vv[4] = {vf.x, vf.y, vf.z, vf.w};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = gf[k] * s;
      float pk = pp[k] * wd_factor;
      float mk = beta1 * mm[k] + (1.0f - beta1) * gk;
      float vk = beta2 * vv[k] + (1.0f - beta2) * gk * gk;
      float denom = sqrtf(vk) * inv_bc2_sqrt + eps;
      pk -= lr * inv_bc1 * mk / denom;
      po[k] = pk; mo[k] = mk; vo[k] = vk;
    }
    if constexpr (COMMIT) {
      reinterpret_cast<float4*>(p)[i] = make_float4(po[0], po[1], po[2], po[3]);
      reinterpret_cast<float4*>(m)[i] = make_float4(mo[0], mo[1], mo[2], mo[3]);
      reinterpret_cast<float4*>(v)[i] = make_float4(vo[0], vo[1], vo[2], vo[3]);
    }
    if (out != nullptr) {
      if constexpr (sizeof(Tbuf) == 2) {
        ushort4 ou;
        ou.x = f32_to_bf16(po[0]); ou.y = f32_to_bf16(po[1]);
        ou.z = f32_to_bf16(po[2]); ou.w = f32_to_bf16(po[3]);
        reinterpret_cast<ushort4*>(out)[i] = ou;
      } else {
        reinterpret_cast<float4*>(out)[i] =
            make_float4(po[0], po[1], po[2], po[3]);
      }
    }
  }

  // scalar tail (segments are 256-aligned so this is normally empty)
  for (long long i = (n4 << 2) + i0; i < n; i += stride) {
    float gk;
    if constexpr (sizeof(Tbuf) == 2) gk = bf16_to_f32(((const unsigned short*)g)[i]);
    else                             gk = ((const float*)g)[i];
    gk *= s;
    float pk = p[i] * wd_factor;
    float mk = beta1 * m[i] + (1.0f - beta1) * gk;
    float vk = beta2 * v[i] + (1.0f - beta2) * gk * gk;
    float denom = sqrtf(vk) * inv_bc2_sqrt + eps;
    pk -= lr * inv_bc1 * mk / denom;
    if constexpr (COMMIT) { p[i] = pk; m[i] = mk; v[i] = vk; }
    if (out != nullptr) {
      if constexpr (sizeof(Tbuf) == 2) ((unsigned short*)out)[i] = f32_to_bf16(pk);
      else                             ((float*)out)[i] = pk;
    }
  }
}

}  // namespace

extern "C" void acco_fused_adamw_launch(
    void* p, const void* g, void* m, void* v, void* out,
    const float* scale_dev,
    long long n, bool buf_is_bf16, bool commit,
    float scale, float lr, float beta1, float beta2, float eps,
    float weight_decay, long long step_plus_1, hipStream_t stream)
{
  const int block = 256;
  const int grid = elementwise_grid((n + 3) / 4, block);
  const float wd_factor = 1.0f - lr * weight_decay;
  const double t = (double)step_plus_1;
  const float inv_bc1 = (float)(1.0 / (1.0 - pow((double)beta1, t)));
  const float inv_bc2_sqrt = (float)(1.0 / sqrt(1.0 - pow((double)beta2, t)));

#define LAUNCH(TB, CM)                                                        \
  hipLaunchKernelGGL((fused_adamw_kernel<TB, CM>), dim3(grid), dim3(block),   \
                     0, stream, (float*)p, (const TB*)g, (float*)m,           \
                     (float*)v, (TB*)out, scale_dev, n, scale, lr, beta1,     \
                     beta2, eps, wd_factor, inv_bc1, inv_bc2_sqrt)

  if (buf_is_bf16) {
    if (commit) LAUNCH(unsigned short, true); else LAUNCH(unsigned short, false);
  } else {
    if (commit) LAUNCH(float, true); else LAUNCH(float, false);
  }
#undef LAUNCH
}
