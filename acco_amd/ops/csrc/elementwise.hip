// Elementwise model ops for gfx950: SwiGLU and gelu_new, forward + backward.
// Memory-bound; bf16 vectorized ×8 per lane (16 B loads — guide G13),
// grid-stride with capped grid (guide G11). Replaces the HF SwiGLU / GELU
// ATen chains of SURVEY.md §2.5 K1/K2 with one kernel per direction.

#include "common.h"

namespace {

ACCO_DEV float sigmoidf_(float x) { return 1.0f / (1.0f + __expf(-x)); }

// hardware-exp tanh: one v_exp_f32 + a few VALU ops instead of libm's
// polynomial tanhf (the kernels are VALU-bound at 99% busy with tanhf —
// profiles/r01_kernels_pmc.txt). exp overflow at large |x| gives
// 2/inf = 0 → ±1 exactly; bf16-accurate.
ACCO_DEV float tanh_fast(float x) {
  float a = __builtin_fabsf(x);
  float t = 1.0f - 2.0f / (__expf(2.0f * a) + 1.0f);
  return __builtin_copysignf(t, x);
}

// gelu_new: 0.5x(1+tanh(sqrt(2/pi)(x+0.044715x^3)))
ACCO_DEV float gelu_new_f(float x) {
  const float c = 0.7978845608028654f;    // sqrt(2/pi)
  float u = c * (x + 0.044715f * x * x * x);
  return 0.5f * x * (1.0f + tanh_fast(u));
}

ACCO_DEV float gelu_new_grad_f(float x) {
  const float c = 0.7978845608028654f;
  float x2 = x * x;
  float u = c * (x + 0.044715f * x * x2);
  float t = tanh_fast(u);
  float sech2 = 1.0f - t * t;
  float du = c * (1.0f + 3.0f * 0.044715f * x2);
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * du;
}

using u16 = unsigned short;

// ---- SwiGLU: out = silu(g) * u
// g/u may be row-strided sections of one packed [rows][2I] tensor
// (the fused gate_up projection output): row = i8 / (I/8) vec8 groups.
__global__ void swiglu_fwd_kernel(const u16* __restrict__ g,
                                  const u16* __restrict__ u,
                                  u16* __restrict__ out, long long n8,
                                  int i8_per_row, long long row_stride) {
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n8; i += stride) {
    const long long row = i / i8_per_row;
    const long long co = (i % i8_per_row) + row * row_stride;
    ushort4 gv0 = reinterpret_cast<const ushort4*>(g)[2 * co];
    ushort4 gv1 = reinterpret_cast<const ushort4*>(g)[2 * co + 1];
    ushort4 uv0 = reinterpret_cast<const ushort4*>(u)[2 * co];
    ushort4 uv1 = reinterpret_cast<const ushort4*>(u)[2 * co + 1];
    u16 gs[8] = {gv0.x, gv0.y, gv0.z, gv0.w, gv1.x, gv1.y, gv1.z, gv1.w};
    u16 us[8] = {uv0.x, uv0.y, uv0.z, uv0.w, uv1.x, uv1.y, uv1.z, uv1.w};
    u16 os[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float gf = bf16_to_f32(gs[k]);
      float uf = bf16_to_f32(us[k]);
      os[k] = f32_to_bf16(gf * sigmoidf_(gf) * uf);
    }
    reinterpret_cast<ushort4*>(out)[2 * i] = make_ushort4(os[0], os[1], os[2], os[3]);
    reinterpret_cast<ushort4*>(out)[2 * i + 1] = make_ushort4(os[4], os[5], os[6], os[7]);
  }
}

// dgate = dout * u * d/dg[g*sig(g)]; dup = dout * silu(g)
// (dout contiguous [rows][I]; g/u/dg/du row-strided packed sections)
__global__ void swiglu_bwd_kernel(const u16* __restrict__ dout,
                                  const u16* __restrict__ g,
                                  const u16* __restrict__ u,
                                  u16* __restrict__ dg,
                                  u16* __restrict__ du, long long n8,
                                  int i8_per_row, long long row_stride) {
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n8; i += stride) {
    const long long co = (i % i8_per_row) + (i / i8_per_row) * row_stride;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      ushort4 dv = reinterpret_cast<const ushort4*>(dout)[2 * i + h];
      ushort4 gv = reinterpret_cast<const ushort4*>(g)[2 * co + h];
      ushort4 uv = reinterpret_cast<const ushort4*>(u)[2 * co + h];
      u16 ds[4] = {dv.x, dv.y, dv.z, dv.w};
      u16 gs[4] = {gv.x, gv.y, gv.z, gv.w};
      u16 us[4] = {uv.x, uv.y, uv.z, uv.w};
      u16 dgo[4], duo[4];
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float df = bf16_to_f32(ds[k]);
        float gf = bf16_to_f32(gs[k]);
        float uf = bf16_to_f32(us[k]);
        float s = sigmoidf_(gf);
        float silu = gf * s;
        float dsilu = s * (1.0f + gf * (1.0f - s));
        dgo[k] = f32_to_bf16(df * uf * dsilu);
        duo[k] = f32_to_bf16(df * silu);
      }
      reinterpret_cast<ushort4*>(dg)[2 * co + h] = make_ushort4(dgo[0], dgo[1], dgo[2], dgo[3]);
      reinterpret_cast<ushort4*>(du)[2 * co + h] = make_ushort4(duo[0], duo[1], duo[2], duo[3]);
    }
  }
}

// ---- gelu_new
__global__ void gelu_fwd_kernel(const u16* __restrict__ x,
                                u16* __restrict__ out, long long n8) {
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n8; i += stride) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      ushort4 xv = reinterpret_cast<const ushort4*>(x)[2 * i + h];
      u16 xs[4] = {xv.x, xv.y, xv.z, xv.w};
      u16 os[4];
#pragma unroll
      for (int k = 0; k < 4; ++k)
        os[k] = f32_to_bf16(gelu_new_f(bf16_to_f32(xs[k])));
      reinterpret_cast<ushort4*>(out)[2 * i + h] = make_ushort4(os[0], os[1], os[2], os[3]);
    }
  }
}

__global__ void gelu_bwd_kernel(const u16* __restrict__ dout,
                                const u16* __restrict__ x,
                                u16* __restrict__ dx, long long n8) {
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < n8; i += stride) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      ushort4 dv = reinterpret_cast<const ushort4*>(dout)[2 * i + h];
      ushort4 xv = reinterpret_cast<const ushort4*>(x)[2 * i + h];
      u16 ds[4] = {dv.x, dv.y, dv.z, dv.w};
      u16 xs[4] = {xv.x, xv.y, xv.z, xv.w};
      u16 os[4];
#pragma unroll
      for (int k = 0; k < 4; ++k)
        os[k] = f32_to_bf16(bf16_to_f32(ds[k]) *
                            gelu_new_grad_f(bf16_to_f32(xs[k])));
      reinterpret_cast<ushort4*>(dx)[2 * i + h] = make_ushort4(os[0], os[1], os[2], os[3]);
    }
  }
}

}  // namespace

extern "C" {

void acco_swiglu_fwd(const void* g, const void* u, void* out, long long n,
                     int cols, long long row_stride8, hipStream_t s) {
  long long n8 = n / 8;
  int grid = elementwise_grid(n8, 256);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const u16*)g, (const u16*)u, (u16*)out, n8, cols / 8,
                     row_stride8);
}

void acco_swiglu_bwd(const void* dout, const void* g, const void* u, void* dg,
                     void* du, long long n, int cols, long long row_stride8,
                     hipStream_t s) {
  long long n8 = n / 8;
  int grid = elementwise_grid(n8, 256);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const u16*)dout, (const u16*)g, (const u16*)u, (u16*)dg,
                     (u16*)du, n8, cols / 8, row_stride8);
}

void acco_gelu_fwd(const void* x, void* out, long long n, hipStream_t s) {
  long long n8 = n / 8;
  int grid = elementwise_grid(n8, 256);
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const u16*)x, (u16*)out, n8);
}

void acco_gelu_bwd(const void* dout, const void* x, void* dx, long long n,
                   hipStream_t s) {
  long long n8 = n / 8;
  int grid = elementwise_grid(n8, 256);
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const u16*)dout, (const u16*)x, (u16*)dx, n8);
}

}  // extern "C"
