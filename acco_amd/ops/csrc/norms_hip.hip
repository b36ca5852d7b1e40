#include "hip/hip_runtime.h"
// RMSNorm (Llama) and LayerNorm (GPT-Neo) forward + backward for gfx950.
// One 256-thread block (4 waves) per row, grid-striding rows; bf16 loads
// vectorized ×8 (guide G13: scalar bf16 ≈ 2× slower); row statistics by
// wave shuffle + LDS cross-wave reduce; weight/bias staged in LDS once per
// block. dW/dB accumulate per-block partials in registers and land with one
// fp32 atomicAdd per column per block (guide G12).
// Replaces the HF RMSNorm / nn.LayerNorm ATen chains.

#include "common.h"

namespace {

using u16 = unsigned short;

constexpr int BLOCK = 256;
constexpr int VEC = 8;                  // bf16 per thread per chunk
constexpr int MAX_CHUNKS = 8;           // D <= 256*8*8 = 16384

// block-reduce a single float (sum) over 4 waves
ACCO_DEV float block_reduce_sum(float x, float* lds) {
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, 64);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) lds[wave] = x;
  __syncthreads();
  float r = (threadIdx.x < BLOCK / 64) ? lds[threadIdx.x] : 0.0f;
  if (threadIdx.x == 0) {
    for (int w = 1; w < BLOCK / 64; ++w) r += lds[w];
    lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

ACCO_DEV void load8(const u16* p, float* f) {
  ushort4 a = reinterpret_cast<const ushort4*>(p)[0];
  ushort4 b = reinterpret_cast<const ushort4*>(p)[1];
  f[0] = bf16_to_f32(a.x); f[1] = bf16_to_f32(a.y);
  f[2] = bf16_to_f32(a.z); f[3] = bf16_to_f32(a.w);
  f[4] = bf16_to_f32(b.x); f[5] = bf16_to_f32(b.y);
  f[6] = bf16_to_f32(b.z); f[7] = bf16_to_f32(b.w);
}

ACCO_DEV void store8(u16* p, const float* f) {
  reinterpret_cast<ushort4*>(p)[0] =
      make_ushort4(f32_to_bf16(f[0]), f32_to_bf16(f[1]),
                   f32_to_bf16(f[2]), f32_to_bf16(f[3]));
  reinterpret_cast<ushort4*>(p)[1] =
      make_ushort4(f32_to_bf16(f[4]), f32_to_bf16(f[5]),
                   f32_to_bf16(f[6]), f32_to_bf16(f[7]));
}

// ------------------------------------------------------------ RMSNorm fwd
__global__ void rmsnorm_fwd_kernel(const u16* __restrict__ x,
                                   const u16* __restrict__ w,
                                   u16* __restrict__ y,
                                   float* __restrict__ rstd,
                                   long long R, int D, float eps) {
  __shared__ float lds[8];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[MAX_CHUNKS][VEC];
    float ssq = 0.0f;
    int j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      load8(xr + c * VEC, xs[j]);
#pragma unroll
      for (int k = 0; k < VEC; ++k) ssq += xs[j][k] * xs[j][k];
    }
    ssq = block_reduce_sum(ssq, lds);
    const float r = rsqrtf(ssq / (float)D + eps);
    if (threadIdx.x == 0 && rstd != nullptr) rstd[row] = r;
    j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      float wf[VEC];
      load8(w_lds + c * VEC, wf);
      float o[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) o[k] = xs[j][k] * r * wf[k];
      store8(yr + c * VEC, o);
    }
  }
}

// ------------------------------------------------------------ RMSNorm bwd
// dx = r*(dy*w) - x * r^3/D * sum(dy*w*x);  dw_col = sum_rows dy*x*r
__global__ void rmsnorm_bwd_kernel(const u16* __restrict__ dy,
                                   const u16* __restrict__ x,
                                   const u16* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   u16* __restrict__ dx,
                                   float* __restrict__ dw,   // fp32, zeroed
                                   long long R, int D) {
  __shared__ float lds[8];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  float dwacc[MAX_CHUNKS][VEC];
  const int my_chunks = (nv - (int)threadIdx.x + BLOCK - 1) / BLOCK;
  for (int j = 0; j < MAX_CHUNKS; ++j)
#pragma unroll
    for (int k = 0; k < VEC; ++k) dwacc[j][k] = 0.0f;

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float r = rstd[row];
    float xs[MAX_CHUNKS][VEC], ds[MAX_CHUNKS][VEC], ws[MAX_CHUNKS][VEC];
    float dot = 0.0f;
    int j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      load8(xr + c * VEC, xs[j]);
      load8(dyr + c * VEC, ds[j]);
      load8(w_lds + c * VEC, ws[j]);
#pragma unroll
      for (int k = 0; k < VEC; ++k) dot += ds[j][k] * ws[j][k] * xs[j][k];
    }
    dot = block_reduce_sum(dot, lds);
    const float coef = r * r * r * dot / (float)D;
    j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      float o[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        o[k] = r * ds[j][k] * ws[j][k] - xs[j][k] * coef;
        dwacc[j][k] += ds[j][k] * xs[j][k] * r;
      }
      store8(dxr + c * VEC, o);
    }
  }
  // one atomicAdd per column per block
  int j = 0;
  for (int c = threadIdx.x; c < nv; c += BLOCK, ++j)
#pragma unroll
    for (int k = 0; k < VEC; ++k)
      atomicAdd(dw + c * VEC + k, dwacc[j][k]);
  (void)my_chunks;
}

// ---------------------------------------------------------- LayerNorm fwd
__global__ void layernorm_fwd_kernel(const u16* __restrict__ x,
                                     const u16* __restrict__ w,
                                     const u16* __restrict__ b,
                                     u16* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     long long R, int D, float eps) {
  __shared__ float lds[8];
  extern __shared__ __attribute__((aligned(16))) u16 wb_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK) {
    reinterpret_cast<uint4*>(wb_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
    reinterpret_cast<uint4*>(wb_lds + D)[c] = reinterpret_cast<const uint4*>(b)[c];
  }
  __syncthreads();

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[MAX_CHUNKS][VEC];
    float sum = 0.0f;
    int j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      load8(xr + c * VEC, xs[j]);
#pragma unroll
      for (int k = 0; k < VEC; ++k) sum += xs[j][k];
    }
    const float mean = block_reduce_sum(sum, lds) / (float)D;
    float var = 0.0f;
    j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j)
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float d = xs[j][k] - mean;
        var += d * d;
      }
    var = block_reduce_sum(var, lds) / (float)D;
    const float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = r;
    }
    j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      float wf[VEC], bf[VEC], o[VEC];
      load8(wb_lds + c * VEC, wf);
      load8(wb_lds + D + c * VEC, bf);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        o[k] = (xs[j][k] - mean) * r * wf[k] + bf[k];
      store8(yr + c * VEC, o);
    }
  }
}

// ---------------------------------------------------------- LayerNorm bwd
// xhat=(x-mean)*r; dyw=dy*w
// dx = r*(dyw - mean(dyw) - xhat*mean(dyw*xhat)); dw=Σ dy*xhat; db=Σ dy
__global__ void layernorm_bwd_kernel(const u16* __restrict__ dy,
                                     const u16* __restrict__ x,
                                     const u16* __restrict__ w,
                                     const float* __restrict__ mean_in,
                                     const float* __restrict__ rstd_in,
                                     u16* __restrict__ dx,
                                     float* __restrict__ dw,
                                     float* __restrict__ db,
                                     long long R, int D) {
  __shared__ float lds[8];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  float dwacc[MAX_CHUNKS][VEC], dbacc[MAX_CHUNKS][VEC];
  for (int j = 0; j < MAX_CHUNKS; ++j)
#pragma unroll
    for (int k = 0; k < VEC; ++k) { dwacc[j][k] = 0.0f; dbacc[j][k] = 0.0f; }

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float mean = mean_in[row];
    const float r = rstd_in[row];
    float xh[MAX_CHUNKS][VEC], ds[MAX_CHUNKS][VEC], dyw[MAX_CHUNKS][VEC];
    float s1 = 0.0f, s2 = 0.0f;
    int j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      float xs[VEC], wf[VEC];
      load8(xr + c * VEC, xs);
      load8(dyr + c * VEC, ds[j]);
      load8(w_lds + c * VEC, wf);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        xh[j][k] = (xs[k] - mean) * r;
        dyw[j][k] = ds[j][k] * wf[k];
        s1 += dyw[j][k];
        s2 += dyw[j][k] * xh[j][k];
      }
    }
    s1 = block_reduce_sum(s1, lds) / (float)D;
    s2 = block_reduce_sum(s2, lds) / (float)D;
    j = 0;
    for (int c = threadIdx.x; c < nv; c += BLOCK, ++j) {
      float o[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        o[k] = r * (dyw[j][k] - s1 - xh[j][k] * s2);
        dwacc[j][k] += ds[j][k] * xh[j][k];
        dbacc[j][k] += ds[j][k];
      }
      store8(dxr + c * VEC, o);
    }
  }
  int j = 0;
  for (int c = threadIdx.x; c < nv; c += BLOCK, ++j)
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      atomicAdd(dw + c * VEC + k, dwacc[j][k]);
      atomicAdd(db + c * VEC + k, dbacc[j][k]);
    }
}

}  // namespace

extern "C" {

void acco_rmsnorm_fwd(const void* x, const void* w, void* y, void* rstd,
                      long long R, int D, float eps, hipStream_t s) {
  int grid = (int)((R < 8192) ? R : 8192);
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(BLOCK),
                     D * sizeof(u16), s, (const u16*)x, (const u16*)w,
                     (u16*)y, (float*)rstd, R, D, eps);
}

void acco_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                      const void* rstd, void* dx, void* dw_fp32,
                      long long R, int D, hipStream_t s) {
  int grid = (int)((R < 1024) ? R : 1024);
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(BLOCK),
                     D * sizeof(u16), s, (const u16*)dy, (const u16*)x,
                     (const u16*)w, (const float*)rstd, (u16*)dx,
                     (float*)dw_fp32, R, D);
}

void acco_layernorm_fwd(const void* x, const void* w, const void* b, void* y,
                        void* mean, void* rstd, long long R, int D, float eps,
                        hipStream_t s) {
  int grid = (int)((R < 8192) ? R : 8192);
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3(grid), dim3(BLOCK),
                     2 * D * sizeof(u16), s, (const u16*)x, (const u16*)w,
                     (const u16*)b, (u16*)y, (float*)mean, (float*)rstd, R, D,
                     eps);
}

void acco_layernorm_bwd(const void* dy, const void* x, const void* w,
                        const void* mean, const void* rstd, void* dx,
                        void* dw_fp32, void* db_fp32, long long R, int D,
                        hipStream_t s) {
  int grid = (int)((R < 1024) ? R : 1024);
  hipLaunchKernelGGL(layernorm_bwd_kernel, dim3(grid), dim3(BLOCK),
                     D * sizeof(u16), s, (const u16*)dy, (const u16*)x,
                     (const u16*)w, (const float*)mean, (const float*)rstd,
                     (u16*)dx, (float*)dw_fp32, (float*)db_fp32, R, D);
}

}  // extern "C"
