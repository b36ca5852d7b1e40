// RMSNorm (Llama) and LayerNorm (GPT-Neo) forward + backward for gfx950.
// One 256-thread block (4 waves) per row, grid-striding rows; bf16 loads
// vectorized ×8 (guide G13); row statistics by wave shuffle + LDS
// cross-wave reduce; weight/bias staged in LDS once per block. dW/dB
// accumulate per-block partials in registers and land with one fp32
// atomicAdd per column per block (guide G12).
//
// The per-thread value arrays are templated on the compile-time chunk
// count (guide §5.4 rule 20: runtime-indexed ext-vector arrays go to
// scratch — a first version with a runtime chunk loop ran 8-14× off
// roofline; see profiles/r01_bench_llama1b_acco_1gpu_kernels.txt).
// Replaces the HF RMSNorm / nn.LayerNorm ATen chains.

#include "common.h"

namespace {

using u16 = unsigned short;

constexpr int BLOCK = 256;
constexpr int VEC = 8;                  // bf16 per thread per chunk

// block-reduce a single float (sum) over 4 waves
ACCO_DEV float block_reduce_sum(float x, float* lds) {
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, 64);
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) lds[wave] = x;
  __syncthreads();
  float r = lds[0] + lds[1] + lds[2] + lds[3];
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
template <int CH>
__global__ __launch_bounds__(BLOCK)
void rmsnorm_fwd_kernel(const u16* __restrict__ x, const u16* __restrict__ w,
                        u16* __restrict__ y, float* __restrict__ rstd,
                        long long R, int D, float eps) {
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  int cid[CH];
  bool act[CH];
  float wf[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = threadIdx.x + j * BLOCK;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
  }

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[CH][VEC];
    float ssq = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) ssq += xs[j][kk] * xs[j][kk];
      }
    ssq = block_reduce_sum(ssq, lds);
    const float r = rsqrtf(ssq / (float)D + eps);
    if (threadIdx.x == 0 && rstd != nullptr) rstd[row] = r;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) o[kk] = xs[j][kk] * r * wf[j][kk];
        store8(yr + cid[j] * VEC, o);
      }
  }
}

// ------------------------------------------------------------ RMSNorm bwd
// dx = r*(dy*w) - x * r^3/D * sum(dy*w*x);  dw_col = sum_rows dy*x*r
template <int CH>
__global__ __launch_bounds__(BLOCK)
void rmsnorm_bwd_kernel(const u16* __restrict__ dy, const u16* __restrict__ x,
                        const u16* __restrict__ w,
                        const float* __restrict__ rstd, u16* __restrict__ dx,
                        float* __restrict__ dw_part,  // [gridDim.x, D] fp32
                        long long R, int D) {
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = threadIdx.x + j * BLOCK;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) dwacc[j][kk] = 0.0f;
  }

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float r = rstd[row];
    float xs[CH][VEC], ds[CH][VEC];
    float dot = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
        load8(dyr + cid[j] * VEC, ds[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk)
          dot += ds[j][kk] * wf[j][kk] * xs[j][kk];
      }
    dot = block_reduce_sum(dot, lds);
    const float coef = r * r * r * dot / (float)D;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          o[kk] = r * ds[j][kk] * wf[j][kk] - xs[j][kk] * coef;
          dwacc[j][kk] += ds[j][kk] * xs[j][kk] * r;
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }
  // one coalesced partial row per block (atomicAdd on 2k fp32 addresses
  // from 4k blocks serialized ~3x worse than the pre-fix kernel; the
  // wrapper reduces the [grid, D] scratch with one tiny torch sum)
  float* out_row = dw_part + (long long)blockIdx.x * D;
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk)
        out_row[cid[j] * VEC + kk] = dwacc[j][kk];
}

// ---------------------------------------------------------- LayerNorm fwd
template <int CH>
__global__ __launch_bounds__(BLOCK)
void layernorm_fwd_kernel(const u16* __restrict__ x, const u16* __restrict__ w,
                          const u16* __restrict__ b, u16* __restrict__ y,
                          float* __restrict__ mean_out,
                          float* __restrict__ rstd_out,
                          long long R, int D, float eps) {
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 wb_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK) {
    reinterpret_cast<uint4*>(wb_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
    reinterpret_cast<uint4*>(wb_lds + D)[c] = reinterpret_cast<const uint4*>(b)[c];
  }
  __syncthreads();

  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], bf[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = threadIdx.x + j * BLOCK;
    act[j] = cid[j] < nv;
    if (act[j]) {
      load8(wb_lds + cid[j] * VEC, wf[j]);
      load8(wb_lds + D + cid[j] * VEC, bf[j]);
    }
  }

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[CH][VEC];
    float sum = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) sum += xs[j][kk];
      }
    const float mean = block_reduce_sum(sum, lds) / (float)D;
    float var = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j])
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          float d = xs[j][kk] - mean;
          var += d * d;
        }
    var = block_reduce_sum(var, lds) / (float)D;
    const float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = r;
    }
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk)
          o[kk] = (xs[j][kk] - mean) * r * wf[j][kk] + bf[j][kk];
        store8(yr + cid[j] * VEC, o);
      }
  }
}

// ---------------------------------------------------------- LayerNorm bwd
// xhat=(x-mean)*r; dyw=dy*w
// dx = r*(dyw - mean(dyw) - xhat*mean(dyw*xhat)); dw=Σ dy*xhat; db=Σ dy
template <int CH>
__global__ __launch_bounds__(BLOCK)
void layernorm_bwd_kernel(const u16* __restrict__ dy, const u16* __restrict__ x,
                          const u16* __restrict__ w,
                          const float* __restrict__ mean_in,
                          const float* __restrict__ rstd_in,
                          u16* __restrict__ dx,
                          float* __restrict__ dw_part,  // [grid, D]
                          float* __restrict__ db_part,  // [grid, D]
                          long long R, int D) {
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC], dbacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = threadIdx.x + j * BLOCK;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) { dwacc[j][kk] = 0.f; dbacc[j][kk] = 0.f; }
  }

  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float mean = mean_in[row];
    const float r = rstd_in[row];
    float xh[CH][VEC], ds[CH][VEC];
    float s1 = 0.0f, s2 = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float xs[VEC];
        load8(xr + cid[j] * VEC, xs);
        load8(dyr + cid[j] * VEC, ds[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          xh[j][kk] = (xs[kk] - mean) * r;
          const float dyw = ds[j][kk] * wf[j][kk];
          s1 += dyw;
          s2 += dyw * xh[j][kk];
        }
      }
    s1 = block_reduce_sum(s1, lds) / (float)D;
    __syncthreads();
    s2 = block_reduce_sum(s2, lds) / (float)D;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          o[kk] = r * (ds[j][kk] * wf[j][kk] - s1 - xh[j][kk] * s2);
          dwacc[j][kk] += ds[j][kk] * xh[j][kk];
          dbacc[j][kk] += ds[j][kk];
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }
  float* wrow = dw_part + (long long)blockIdx.x * D;
  float* brow = db_part + (long long)blockIdx.x * D;
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk) {
        wrow[cid[j] * VEC + kk] = dwacc[j][kk];
        brow[cid[j] * VEC + kk] = dbacc[j][kk];
      }
}

template <template <int> class K>
struct ChDispatch {};

int chunks_for(int D) {
  const int nv = D / VEC;
  if (nv <= BLOCK) return 1;
  if (nv <= 2 * BLOCK) return 2;
  if (nv <= 4 * BLOCK) return 4;
  return 8;
}

}  // namespace

extern "C" {

int acco_norm_bwd_grid(long long R) { return (int)((R < 2048) ? R : 2048); }

void acco_rmsnorm_fwd(const void* x, const void* w, void* y, void* rstd,
                      long long R, int D, float eps, hipStream_t s) {
  int grid = (int)((R < 8192) ? R : 8192);
  const int lds = D * sizeof(u16);
#define L(CH) hipLaunchKernelGGL(rmsnorm_fwd_kernel<CH>, dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)x, (const u16*)w, (u16*)y, \
    (float*)rstd, R, D, eps)
  switch (chunks_for(D)) {
    case 1: L(1); break; case 2: L(2); break;
    case 4: L(4); break; default: L(8); break;
  }
#undef L
}

void acco_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                      const void* rstd, void* dx, void* dw_fp32,
                      long long R, int D, hipStream_t s) {
  int grid = acco_norm_bwd_grid(R);
  const int lds = D * sizeof(u16);
#define L(CH) hipLaunchKernelGGL(rmsnorm_bwd_kernel<CH>, dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)dy, (const u16*)x, (const u16*)w, \
    (const float*)rstd, (u16*)dx, (float*)dw_fp32, R, D)
  switch (chunks_for(D)) {
    case 1: L(1); break; case 2: L(2); break;
    case 4: L(4); break; default: L(8); break;
  }
#undef L
}

void acco_layernorm_fwd(const void* x, const void* w, const void* b, void* y,
                        void* mean, void* rstd, long long R, int D, float eps,
                        hipStream_t s) {
  int grid = (int)((R < 8192) ? R : 8192);
  const int lds = 2 * D * sizeof(u16);
#define L(CH) hipLaunchKernelGGL(layernorm_fwd_kernel<CH>, dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)x, (const u16*)w, (const u16*)b, \
    (u16*)y, (float*)mean, (float*)rstd, R, D, eps)
  switch (chunks_for(D)) {
    case 1: L(1); break; case 2: L(2); break;
    case 4: L(4); break; default: L(8); break;
  }
#undef L
}

void acco_layernorm_bwd(const void* dy, const void* x, const void* w,
                        const void* mean, const void* rstd, void* dx,
                        void* dw_fp32, void* db_fp32, long long R, int D,
                        hipStream_t s) {
  int grid = acco_norm_bwd_grid(R);
  const int lds = D * sizeof(u16);
#define L(CH) hipLaunchKernelGGL(layernorm_bwd_kernel<CH>, dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)dy, (const u16*)x, (const u16*)w, \
    (const float*)mean, (const float*)rstd, (u16*)dx, (float*)dw_fp32, \
    (float*)db_fp32, R, D)
  switch (chunks_for(D)) {
    case 1: L(1); break; case 2: L(2); break;
    case 4: L(4); break; default: L(8); break;
  }
#undef L
}

}  // extern "C"
