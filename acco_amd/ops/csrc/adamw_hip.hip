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

typedef float f4_ __attribute__((ext_vector_type(4)));
typedef unsigned short u4_ __attribute__((ext_vector_type(4)));

// p/m/v/g/out are each streamed exactly once per round: nontemporal
// loads/stores keep the ~3.4 GB working set out of L2 (guide: streaming
// kernels should not thrash caches shared with the concurrently running
// compute stream).
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

  // two grid-stride chunks per iteration: 8 independent loads in flight
  // before any math retires (latency hiding for the HBM round trips)
  auto body = [&](long long i, const f4_& pf, const f4_& mf, const f4_& vf,
                  const float* gf) {
    float po[4], mo[4], vo[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = gf[k] * s;
      float pk = pf[k] * wd_factor;
      float mk = beta1 * mf[k] + (1.0f - beta1) * gk;
      float vk = beta2 * vf[k] + (1.0f - beta2) * gk * gk;
      float denom = sqrtf(vk) * inv_bc2_sqrt + eps;
      pk -= lr * inv_bc1 * mk / denom;
      po[k] = pk; mo[k] = mk; vo[k] = vk;
    }
    if constexpr (COMMIT) {
      __builtin_nontemporal_store(f4_{po[0], po[1], po[2], po[3]},
                                  reinterpret_cast<f4_*>(p) + i);
      __builtin_nontemporal_store(f4_{mo[0], mo[1], mo[2], mo[3]},
                                  reinterpret_cast<f4_*>(m) + i);
      __builtin_nontemporal_store(f4_{vo[0], vo[1], vo[2], vo[3]},
                                  reinterpret_cast<f4_*>(v) + i);
    }
    if (out != nullptr) {
      if constexpr (sizeof(Tbuf) == 2) {
        u4_ ou = {f32_to_bf16(po[0]), f32_to_bf16(po[1]),
                  f32_to_bf16(po[2]), f32_to_bf16(po[3])};
        __builtin_nontemporal_store(ou, reinterpret_cast<u4_*>(out) + i);
      } else {
        __builtin_nontemporal_store(f4_{po[0], po[1], po[2], po[3]},
                                    reinterpret_cast<f4_*>(out) + i);
      }
    }
  };
  auto load_g = [&](long long i, float* gf) {
    if constexpr (sizeof(Tbuf) == 2) {
      u4_ gu = __builtin_nontemporal_load(
          reinterpret_cast<const u4_*>(g) + i);
      gf[0] = bf16_to_f32(gu[0]); gf[1] = bf16_to_f32(gu[1]);
      gf[2] = bf16_to_f32(gu[2]); gf[3] = bf16_to_f32(gu[3]);
    } else {
      f4_ gv = __builtin_nontemporal_load(
          reinterpret_cast<const f4_*>(g) + i);
      gf[0] = gv[0]; gf[1] = gv[1]; gf[2] = gv[2]; gf[3] = gv[3];
    }
  };

  long long i = i0;
  for (; i + stride < n4; i += 2 * stride) {
    const long long i2 = i + stride;
    f4_ pf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(p) + i);
    f4_ mf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(m) + i);
    f4_ vf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(v) + i);
    float gf[4];
    load_g(i, gf);
    f4_ pf2 = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(p) + i2);
    f4_ mf2 = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(m) + i2);
    f4_ vf2 = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(v) + i2);
    float gf2[4];
    load_g(i2, gf2);
    body(i, pf, mf, vf, gf);
    body(i2, pf2, mf2, vf2, gf2);
  }
  for (; i < n4; i += stride) {
    f4_ pf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(p) + i);
    f4_ mf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(m) + i);
    f4_ vf = __builtin_nontemporal_load(reinterpret_cast<const f4_*>(v) + i);
    float gf[4];
    load_g(i, gf);
    body(i, pf, mf, vf, gf);
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
