// RMSNorm (Llama) and LayerNorm (GPT-Neo) forward + backward for gfx950.
// One 256-thread block per G rows (G = 4/2/1 picked from D so every wave
// has work even at small D — D=768 fills 96 of 256 lanes in the one-row
// layout, measured 1.4 TB/s; the grouped layout packs 2 rows per block),
// grid-striding rows; bf16 loads vectorized ×8 (guide G13); row statistics
// by wave shuffle + LDS cross-wave reduce; weight/bias staged in LDS once
// per block.
//
// The per-thread value arrays are templated on the compile-time chunk
// count (guide §5.4 rule 20: runtime-indexed ext-vector arrays go to
// scratch — a first version with a runtime chunk loop ran 8-14× off
// roofline; see profiles/r01_bench_llama1b_acco_1gpu_kernels.txt).
// dW/dB land as one [partial_rows, D] fp32 scratch row per (block, group)
// reduced by a tiny torch sum in the wrapper (atomicAdd serialized ~3×
// worse).
// Replaces the HF RMSNorm / nn.LayerNorm ATen chains.

#include "common.h"

namespace {

using u16 = unsigned short;

constexpr int BLOCK = 256;
constexpr int VEC = 8;                  // bf16 per thread per chunk

// Reduce a float over one row-group (TPG = BLOCK/G threads). All groups
// hit the same __syncthreads unconditionally — barriers stay block-uniform
// even when a group's row index runs past R.
template <int G>
ACCO_DEV float group_reduce_sum(float x, float* lds, int g, int lane) {
  constexpr int WPG = (BLOCK / G) / 64;   // waves per group
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down(x, off, 64);
  if ((lane & 63) == 0) lds[g * WPG + lane / 64] = x;
  __syncthreads();
  float r = 0.0f;
#pragma unroll
  for (int wv = 0; wv < WPG; ++wv) r += lds[g * WPG + wv];
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
// res != nullptr fuses the residual add: s = bf16(x + res) is written to
// sum_out and the statistics/normalization run on s — one kernel replaces
// the eager add + norm pair (saves a full read+write pass of the hidden
// state per fusion site; the bf16 rounding of s matches the eager add).
template <int CH, int G>
__global__ __launch_bounds__(BLOCK)
void rmsnorm_fwd_kernel(const u16* __restrict__ x, const u16* __restrict__ w,
                        u16* __restrict__ y, float* __restrict__ rstd,
                        const u16* __restrict__ res, u16* __restrict__ sum_out,
                        long long R, int D, float eps) {
  constexpr int TPG = BLOCK / G;
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  const int g = threadIdx.x / TPG;
  const int lane = threadIdx.x % TPG;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * TPG;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
  }

  for (long long base = (long long)blockIdx.x * G; base < R;
       base += (long long)gridDim.x * G) {
    const long long row = base + g;
    const bool live = row < R;
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[CH][VEC];
    float ssq = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
        if (res != nullptr) {
          float rr[VEC];
          load8(res + row * D + cid[j] * VEC, rr);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk)
            xs[j][kk] = bf16_to_f32(f32_to_bf16(xs[j][kk] + rr[kk]));
          store8(sum_out + row * D + cid[j] * VEC, xs[j]);
        }
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) ssq += xs[j][kk] * xs[j][kk];
      }
    ssq = group_reduce_sum<G>(ssq, lds, g, lane);
    const float r = rsqrtf(ssq / (float)D + eps);
    if (live && lane == 0 && rstd != nullptr) rstd[row] = r;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) o[kk] = xs[j][kk] * r * wf[j][kk];
        store8(yr + cid[j] * VEC, o);
      }
  }
}

// ------------------------------------------------------------ RMSNorm bwd
// dx = r*(dy*w) - x * r^3/D * sum(dy*w*x);  dw_col = sum_rows dy*x*r
template <int CH, int G>
__global__ __launch_bounds__(BLOCK)
void rmsnorm_bwd_kernel(const u16* __restrict__ dy, const u16* __restrict__ x,
                        const u16* __restrict__ w,
                        const float* __restrict__ rstd, u16* __restrict__ dx,
                        float* __restrict__ dw_part,  // [grid*G, D] fp32
                        const u16* __restrict__ dadd,  // fused += residual grad
                        long long R, int D) {
  constexpr int TPG = BLOCK / G;
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  const int g = threadIdx.x / TPG;
  const int lane = threadIdx.x % TPG;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * TPG;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) dwacc[j][kk] = 0.0f;
  }

  for (long long base = (long long)blockIdx.x * G; base < R;
       base += (long long)gridDim.x * G) {
    const long long row = base + g;
    const bool live = row < R;
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float r = live ? rstd[row] : 0.0f;
    float xs[CH][VEC], ds[CH][VEC];
    float dot = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
        load8(dyr + cid[j] * VEC, ds[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk)
          dot += ds[j][kk] * wf[j][kk] * xs[j][kk];
      }
    dot = group_reduce_sum<G>(dot, lds, g, lane);
    const float coef = r * r * r * dot / (float)D;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          o[kk] = r * ds[j][kk] * wf[j][kk] - xs[j][kk] * coef;
          dwacc[j][kk] += ds[j][kk] * xs[j][kk] * r;
        }
        if (dadd != nullptr) {
          float da[VEC];
          load8(dadd + row * D + cid[j] * VEC, da);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk) o[kk] += da[kk];
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }
  // one coalesced partial row per (block, group); groups that never saw a
  // live row still write their zeros (the scratch is allocated w/ empty)
  float* out_row = dw_part + ((long long)blockIdx.x * G + g) * D;
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk)
        out_row[cid[j] * VEC + kk] = dwacc[j][kk];
}

// ------------------------------------------- wave-per-row bwd (nv ≤ 128)
// The grouped bwd kernels above split one row over 2-4 waves, paying 2
// barriers per cross-wave reduction and idling lanes at D=768 (96 of 128):
// measured 1.58 TB/s on the GPT-Neo LayerNorm bwd. These variants give each
// WAVE a whole row (zero barriers in the row loop, reductions are 6
// shfl_xor), run 1024 blocks (4 waves/SIMD), and combine the block's 4
// per-group dW/dB partials through LDS at the end so the scratch stays at
// ≤1024 partial rows.

ACCO_DEV float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

template <int CH>
__global__ __launch_bounds__(BLOCK)
void rmsnorm_bwd_wr_kernel(const u16* __restrict__ dy, const u16* __restrict__ x,
                           const u16* __restrict__ w,
                           const float* __restrict__ rstd, u16* __restrict__ dx,
                           float* __restrict__ dw_part,  // [grid, D] fp32
                           const u16* __restrict__ dadd,
                           long long R, int D) {
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];  // 16·D bytes
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  const int g = threadIdx.x >> 6;        // wave = row group
  const int lane = threadIdx.x & 63;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * 64;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) dwacc[j][kk] = 0.0f;
  }

  for (long long row = (long long)blockIdx.x * 4 + g; row < R;
       row += (long long)gridDim.x * 4) {
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
    dot = wave_reduce_sum(dot);
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
        if (dadd != nullptr) {
          float da[VEC];
          load8(dadd + row * D + cid[j] * VEC, da);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk) o[kk] += da[kk];
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }

  // combine the 4 groups' dW through LDS → ONE partial row per block
  __syncthreads();                       // weights no longer needed
  float* part = reinterpret_cast<float*>(w_lds);
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk)
        part[g * D + cid[j] * VEC + kk] = dwacc[j][kk];
  __syncthreads();
  float* out_row = dw_part + (long long)blockIdx.x * D;
  for (int c = threadIdx.x; c < D; c += BLOCK)
    out_row[c] = part[c] + part[D + c] + part[2 * D + c] + part[3 * D + c];
}

template <int CH>
__global__ __launch_bounds__(BLOCK)
void layernorm_bwd_wr_kernel(const u16* __restrict__ dy,
                             const u16* __restrict__ x,
                             const u16* __restrict__ w,
                             const float* __restrict__ mean_in,
                             const float* __restrict__ rstd_in,
                             u16* __restrict__ dx,
                             float* __restrict__ dw_part,  // [grid, D]
                             float* __restrict__ db_part,  // [grid, D]
                             const u16* __restrict__ dadd,
                             long long R, int D) {
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];  // 16·D bytes
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  const int g = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC], dbacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * 64;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) { dwacc[j][kk] = 0.f; dbacc[j][kk] = 0.f; }
  }

  // 1-deep software pipeline over rows: the next row's x/dy uint4 loads
  // (and its mean/rstd) issue before the current row's reductions, hiding
  // the ~900-cycle HBM latency under the shfl-reduce + epilogue (PMC
  // showed the plain loop 55% parked at waits)
  const long long stride = (long long)gridDim.x * 4;
  long long row = (long long)blockIdx.x * 4 + g;
  uint4 xv[CH], dv[CH];
  float mean = 0.f, r = 0.f;
  auto issue = [&](long long rw) {
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        xv[j] = *reinterpret_cast<const uint4*>(x + rw * D + cid[j] * VEC);
        dv[j] = *reinterpret_cast<const uint4*>(dy + rw * D + cid[j] * VEC);
      }
    mean = mean_in[rw];
    r = rstd_in[rw];
  };
  auto unpack8 = [](const uint4& v, float* f) {
    const u16* u = reinterpret_cast<const u16*>(&v);
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) f[kk] = bf16_to_f32(u[kk]);
  };
  if (row < R) issue(row);
  for (; row < R; row += stride) {
    float xh[CH][VEC], ds[CH][VEC];
    const float mean_c = mean, r_c = r;
    float s1 = 0.0f, s2 = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float xs[VEC];
        unpack8(xv[j], xs);
        unpack8(dv[j], ds[j]);
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          xh[j][kk] = (xs[kk] - mean_c) * r_c;
          const float dyw = ds[j][kk] * wf[j][kk];
          s1 += dyw;
          s2 += dyw * xh[j][kk];
        }
      }
    if (row + stride < R) issue(row + stride);   // in flight under reduce
    s1 = wave_reduce_sum(s1) / (float)D;
    s2 = wave_reduce_sum(s2) / (float)D;
    u16* dxr = dx + row * D;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          o[kk] = r_c * (ds[j][kk] * wf[j][kk] - s1 - xh[j][kk] * s2);
          dwacc[j][kk] += ds[j][kk] * xh[j][kk];
          dbacc[j][kk] += ds[j][kk];
        }
        if (dadd != nullptr) {
          float da[VEC];
          load8(dadd + row * D + cid[j] * VEC, da);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk) o[kk] += da[kk];
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }

  // combine dW, then dB, through the same LDS region
  float* part = reinterpret_cast<float*>(w_lds);
  float* out_w = dw_part + (long long)blockIdx.x * D;
  float* out_b = db_part + (long long)blockIdx.x * D;
  __syncthreads();
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk)
        part[g * D + cid[j] * VEC + kk] = dwacc[j][kk];
  __syncthreads();
  for (int c = threadIdx.x; c < D; c += BLOCK)
    out_w[c] = part[c] + part[D + c] + part[2 * D + c] + part[3 * D + c];
  __syncthreads();
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk)
        part[g * D + cid[j] * VEC + kk] = dbacc[j][kk];
  __syncthreads();
  for (int c = threadIdx.x; c < D; c += BLOCK)
    out_b[c] = part[c] + part[D + c] + part[2 * D + c] + part[3 * D + c];
}

// ---------------------------------------------------------- LayerNorm fwd
template <int CH, int G>
__global__ __launch_bounds__(BLOCK)
void layernorm_fwd_kernel(const u16* __restrict__ x, const u16* __restrict__ w,
                          const u16* __restrict__ b, u16* __restrict__ y,
                          float* __restrict__ mean_out,
                          float* __restrict__ rstd_out,
                          const u16* __restrict__ res,
                          u16* __restrict__ sum_out,
                          long long R, int D, float eps) {
  constexpr int TPG = BLOCK / G;
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 wb_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK) {
    reinterpret_cast<uint4*>(wb_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
    reinterpret_cast<uint4*>(wb_lds + D)[c] = reinterpret_cast<const uint4*>(b)[c];
  }
  __syncthreads();

  const int g = threadIdx.x / TPG;
  const int lane = threadIdx.x % TPG;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], bf[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * TPG;
    act[j] = cid[j] < nv;
    if (act[j]) {
      load8(wb_lds + cid[j] * VEC, wf[j]);
      load8(wb_lds + D + cid[j] * VEC, bf[j]);
    }
  }

  for (long long base = (long long)blockIdx.x * G; base < R;
       base += (long long)gridDim.x * G) {
    const long long row = base + g;
    const bool live = row < R;
    const u16* xr = x + row * D;
    u16* yr = y + row * D;
    float xs[CH][VEC];
    float sum = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        load8(xr + cid[j] * VEC, xs[j]);
        if (res != nullptr) {
          float rr[VEC];
          load8(res + row * D + cid[j] * VEC, rr);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk)
            xs[j][kk] = bf16_to_f32(f32_to_bf16(xs[j][kk] + rr[kk]));
          store8(sum_out + row * D + cid[j] * VEC, xs[j]);
        }
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) sum += xs[j][kk];
      }
    const float mean = group_reduce_sum<G>(sum, lds, g, lane) / (float)D;
    float var = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j])
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          float d = xs[j][kk] - mean;
          var += d * d;
        }
    var = group_reduce_sum<G>(var, lds, g, lane) / (float)D;
    const float r = rsqrtf(var + eps);
    if (live && lane == 0) {
      if (mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = r;
    }
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
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
template <int CH, int G>
__global__ __launch_bounds__(BLOCK)
void layernorm_bwd_kernel(const u16* __restrict__ dy, const u16* __restrict__ x,
                          const u16* __restrict__ w,
                          const float* __restrict__ mean_in,
                          const float* __restrict__ rstd_in,
                          u16* __restrict__ dx,
                          float* __restrict__ dw_part,  // [grid*G, D]
                          float* __restrict__ db_part,  // [grid*G, D]
                          const u16* __restrict__ dadd,
                          long long R, int D) {
  constexpr int TPG = BLOCK / G;
  __shared__ float lds[4];
  extern __shared__ __attribute__((aligned(16))) u16 w_lds[];
  const int nv = D / VEC;
  for (int c = threadIdx.x; c < nv; c += BLOCK)
    reinterpret_cast<uint4*>(w_lds)[c] = reinterpret_cast<const uint4*>(w)[c];
  __syncthreads();

  const int g = threadIdx.x / TPG;
  const int lane = threadIdx.x % TPG;
  int cid[CH];
  bool act[CH];
  float wf[CH][VEC], dwacc[CH][VEC], dbacc[CH][VEC];
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    cid[j] = lane + j * TPG;
    act[j] = cid[j] < nv;
    if (act[j]) load8(w_lds + cid[j] * VEC, wf[j]);
#pragma unroll
    for (int kk = 0; kk < VEC; ++kk) { dwacc[j][kk] = 0.f; dbacc[j][kk] = 0.f; }
  }

  for (long long base = (long long)blockIdx.x * G; base < R;
       base += (long long)gridDim.x * G) {
    const long long row = base + g;
    const bool live = row < R;
    const u16* dyr = dy + row * D;
    const u16* xr = x + row * D;
    u16* dxr = dx + row * D;
    const float mean = live ? mean_in[row] : 0.0f;
    const float r = live ? rstd_in[row] : 0.0f;
    float xh[CH][VEC], ds[CH][VEC];
    float s1 = 0.0f, s2 = 0.0f;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
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
    s1 = group_reduce_sum<G>(s1, lds, g, lane) / (float)D;
    s2 = group_reduce_sum<G>(s2, lds, g, lane) / (float)D;
#pragma unroll
    for (int j = 0; j < CH; ++j)
      if (live && act[j]) {
        float o[VEC];
#pragma unroll
        for (int kk = 0; kk < VEC; ++kk) {
          o[kk] = r * (ds[j][kk] * wf[j][kk] - s1 - xh[j][kk] * s2);
          dwacc[j][kk] += ds[j][kk] * xh[j][kk];
          dbacc[j][kk] += ds[j][kk];
        }
        if (dadd != nullptr) {
          float da[VEC];
          load8(dadd + row * D + cid[j] * VEC, da);
#pragma unroll
          for (int kk = 0; kk < VEC; ++kk) o[kk] += da[kk];
        }
        store8(dxr + cid[j] * VEC, o);
      }
  }
  float* wrow = dw_part + ((long long)blockIdx.x * G + g) * D;
  float* brow = db_part + ((long long)blockIdx.x * G + g) * D;
#pragma unroll
  for (int j = 0; j < CH; ++j)
    if (act[j])
#pragma unroll
      for (int kk = 0; kk < VEC; ++kk) {
        wrow[cid[j] * VEC + kk] = dwacc[j][kk];
        brow[cid[j] * VEC + kk] = dbacc[j][kk];
      }
}

// ------------------------------------------------- column sum (dim-0 sum)
// out[d] = Σ_p in[p·D + d] for a [P, D] row-major matrix — the norm
// dW/dB partial reduce and the bias-wgrad reduce. ATen's dim-0 reduce_kernel
// runs this at ~150 GB/s (20.8 µs on [1024,768] fp32); coalesced row sweeps
// with a P-split + fp32 atomics reach the roofline.
// VECW consecutive columns per thread, 16-byte row loads (guide G13: a
// scalar-load version of this ran at 0.5 TB/s and regressed the GPT-Neo
// bias-wgrad path 4x vs ATen before being caught in the step profile)
template <typename Tin>
__global__ __launch_bounds__(BLOCK)
void colsum_kernel(const Tin* __restrict__ in, float* __restrict__ out,
                   long long P, int D) {
  constexpr int VECW = (sizeof(Tin) == 2) ? 8 : 4;   // 16 B per lane
  const int d0 = (blockIdx.x * BLOCK + threadIdx.x) * VECW;
  if (d0 >= D) return;
  const long long p0 = (long long)blockIdx.y * P / gridDim.y;
  const long long p1 = (long long)(blockIdx.y + 1) * P / gridDim.y;
  float acc[VECW];
#pragma unroll
  for (int k = 0; k < VECW; ++k) acc[k] = 0.0f;
  if (d0 + VECW <= D) {
    for (long long p = p0; p < p1; ++p) {
      if constexpr (sizeof(Tin) == 2) {
        const u16* row = (const u16*)in + p * D + d0;
        ushort4 a = reinterpret_cast<const ushort4*>(row)[0];
        ushort4 b = reinterpret_cast<const ushort4*>(row)[1];
        acc[0] += bf16_to_f32(a.x); acc[1] += bf16_to_f32(a.y);
        acc[2] += bf16_to_f32(a.z); acc[3] += bf16_to_f32(a.w);
        acc[4] += bf16_to_f32(b.x); acc[5] += bf16_to_f32(b.y);
        acc[6] += bf16_to_f32(b.z); acc[7] += bf16_to_f32(b.w);
      } else {
        const float4 v =
            *reinterpret_cast<const float4*>((const float*)in + p * D + d0);
        acc[0] += v.x; acc[1] += v.y; acc[2] += v.z; acc[3] += v.w;
      }
    }
  } else {
    for (long long p = p0; p < p1; ++p)      // ragged tail columns
#pragma unroll
      for (int k = 0; k < VECW; ++k)
        if (d0 + k < D) {
          if constexpr (sizeof(Tin) == 2)
            acc[k] += bf16_to_f32(((const u16*)in)[p * D + d0 + k]);
          else
            acc[k] += ((const float*)in)[p * D + d0 + k];
        }
  }
#pragma unroll
  for (int k = 0; k < VECW; ++k)
    if (d0 + k < D) atomicAdd(out + d0 + k, acc[k]);
}

int groups_for(int D) {
  const int nv = D / VEC;
  if (nv <= 64) return 4;      // one wave per row
  if (nv <= 128) return 2;     // two waves per row (e.g. gptneo D=768)
  return 1;
}

int chunks_for(int D) {
  const int nv = D / VEC;
  if (nv <= BLOCK) return 1;
  if (nv <= 2 * BLOCK) return 2;
  if (nv <= 4 * BLOCK) return 4;
  return 8;
}

int fwd_blocks(long long R, int G) {
  long long b = (R + G - 1) / G;
  const long long cap = 8192 / G;
  return (int)((b < cap) ? (b < 1 ? 1 : b) : cap);
}

int bwd_blocks(long long R, int G) {
  long long b = (R + G - 1) / G;
  // cap partial rows at 1024: the [partials, D] fp32 scratch write + torch
  // reduce is ~25% of bwd traffic at 2048 partials and halves here, while
  // 1024 blocks x 4 waves still fill all 1024 SIMDs
  const long long cap = 1024 / G;
  return (int)((b < cap) ? (b < 1 ? 1 : b) : cap);
}

// wave-per-row bwd (nv ≤ 128): 4 rows per block, 1 partial row per block
bool use_wr(int D) { return D / VEC <= 128; }

int wr_blocks(long long R) {
  long long b = (R + 3) / 4;
  return (int)((b < 1024) ? (b < 1 ? 1 : b) : 1024);
}

}  // namespace

extern "C" {

void acco_colsum(const void* in, float* out, long long P, int D,
                 bool in_is_bf16, hipStream_t s) {
  const int vecw = in_is_bf16 ? 8 : 4;
  const int xblocks = (D + BLOCK * vecw - 1) / (BLOCK * vecw);
  // split P so the grid covers the chip even at small D (out is zeroed by
  // the caller; partial sums combine via fp32 atomics)
  int ysplit = (int)(512 / (xblocks < 1 ? 1 : xblocks));
  if (ysplit < 1) ysplit = 1;
  if ((long long)ysplit > P) ysplit = (int)(P < 1 ? 1 : P);
  dim3 grid(xblocks, ysplit);
  if (in_is_bf16)
    hipLaunchKernelGGL((colsum_kernel<u16>), grid, dim3(BLOCK), 0, s,
                       (const u16*)in, out, P, D);
  else
    hipLaunchKernelGGL((colsum_kernel<float>), grid, dim3(BLOCK), 0, s,
                       (const float*)in, out, P, D);
}

// number of fp32 partial rows the bwd kernels emit (scratch allocation)
int acco_norm_bwd_grid(long long R, int D) {
  if (use_wr(D)) return wr_blocks(R);
  const int G = groups_for(D);
  return bwd_blocks(R, G) * G;
}

void acco_rmsnorm_fwd(const void* x, const void* w, void* y, void* rstd,
                      const void* res, void* sum_out,
                      long long R, int D, float eps, hipStream_t s) {
  const int lds = D * sizeof(u16);
  const int G = groups_for(D);
  const int grid = fwd_blocks(R, G);
#define L(CH, G) hipLaunchKernelGGL((rmsnorm_fwd_kernel<CH, G>), dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)x, (const u16*)w, (u16*)y, \
    (float*)rstd, (const u16*)res, (u16*)sum_out, R, D, eps)
  if (G == 4) { L(1, 4); }
  else if (G == 2) { L(1, 2); }
  else switch (chunks_for(D)) {
    case 1: L(1, 1); break; case 2: L(2, 1); break;
    case 4: L(4, 1); break; default: L(8, 1); break;
  }
#undef L
}

void acco_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                      const void* rstd, void* dx, void* dw_fp32,
                      const void* dadd, long long R, int D, hipStream_t s) {
  if (use_wr(D)) {
    const int wlds = 16 * D;             // 4 fp32 partial rows (≥ weights)
    const int grid = wr_blocks(R);
    if (D / VEC <= 64)
      hipLaunchKernelGGL((rmsnorm_bwd_wr_kernel<1>), dim3(grid), dim3(BLOCK),
                         wlds, s, (const u16*)dy, (const u16*)x,
                         (const u16*)w, (const float*)rstd, (u16*)dx,
                         (float*)dw_fp32, (const u16*)dadd, R, D);
    else
      hipLaunchKernelGGL((rmsnorm_bwd_wr_kernel<2>), dim3(grid), dim3(BLOCK),
                         wlds, s, (const u16*)dy, (const u16*)x,
                         (const u16*)w, (const float*)rstd, (u16*)dx,
                         (float*)dw_fp32, (const u16*)dadd, R, D);
    return;
  }
  const int lds = D * sizeof(u16);
  const int G = groups_for(D);
  const int grid = bwd_blocks(R, G);
#define L(CH, G) hipLaunchKernelGGL((rmsnorm_bwd_kernel<CH, G>), dim3(grid), \
    dim3(BLOCK), lds, s, (const u16*)dy, (const u16*)x, (const u16*)w, \
    (const float*)rstd, (u16*)dx, (float*)dw_fp32, (const u16*)dadd, R, D)
  if (G == 4) { L(1, 4); }
  else if (G == 2) { L(1, 2); }
  else switch (chunks_for(D)) {
    case 1: L(1, 1); break; case 2: L(2, 1); break;
    case 4: L(4, 1); break; default: L(8, 1); break;
  }
#undef L
}

void acco_layernorm_fwd(const void* x, const void* w, const void* b, void* y,
                        void* mean, void* rstd, const void* res,
                        void* sum_out, long long R, int D, float eps,
                        hipStream_t s) {
  const int lds = 2 * D * sizeof(u16);
  const int G = groups_for(D);
  const int grid = fwd_blocks(R, G);
#define L(CH, G) hipLaunchKernelGGL((layernorm_fwd_kernel<CH, G>), \
    dim3(grid), dim3(BLOCK), lds, s, (const u16*)x, (const u16*)w, \
    (const u16*)b, (u16*)y, (float*)mean, (float*)rstd, (const u16*)res, \
    (u16*)sum_out, R, D, eps)
  if (G == 4) { L(1, 4); }
  else if (G == 2) { L(1, 2); }
  else switch (chunks_for(D)) {
    case 1: L(1, 1); break; case 2: L(2, 1); break;
    case 4: L(4, 1); break; default: L(8, 1); break;
  }
#undef L
}

void acco_layernorm_bwd(const void* dy, const void* x, const void* w,
                        const void* mean, const void* rstd, void* dx,
                        void* dw_fp32, void* db_fp32, const void* dadd,
                        long long R, int D, hipStream_t s) {
  if (use_wr(D)) {
    const int wlds = 16 * D;
    const int grid = wr_blocks(R);
    if (D / VEC <= 64)
      hipLaunchKernelGGL((layernorm_bwd_wr_kernel<1>), dim3(grid),
                         dim3(BLOCK), wlds, s, (const u16*)dy, (const u16*)x,
                         (const u16*)w, (const float*)mean,
                         (const float*)rstd, (u16*)dx, (float*)dw_fp32,
                         (float*)db_fp32, (const u16*)dadd, R, D);
    else
      hipLaunchKernelGGL((layernorm_bwd_wr_kernel<2>), dim3(grid),
                         dim3(BLOCK), wlds, s, (const u16*)dy, (const u16*)x,
                         (const u16*)w, (const float*)mean,
                         (const float*)rstd, (u16*)dx, (float*)dw_fp32,
                         (float*)db_fp32, (const u16*)dadd, R, D);
    return;
  }
  const int lds = D * sizeof(u16);
  const int G = groups_for(D);
  const int grid = bwd_blocks(R, G);
#define L(CH, G) hipLaunchKernelGGL((layernorm_bwd_kernel<CH, G>), \
    dim3(grid), dim3(BLOCK), lds, s, (const u16*)dy, (const u16*)x, \
    (const u16*)w, (const float*)mean, (const float*)rstd, (u16*)dx, \
    (float*)dw_fp32, (float*)db_fp32, (const u16*)dadd, R, D)
  if (G == 4) { L(1, 4); }
  else if (G == 2) { L(1, 2); }
  else switch (chunks_for(D)) {
    case 1: L(1, 1); break; case 2: L(2, 1); break;
    case 4: L(4, 1); break; default: L(8, 1); break;
  }
#undef L
}

}  // extern "C"
