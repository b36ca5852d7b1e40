// Experimental NT GEMM for gfx950: C[M,N] = A[M,K] · B[N,K]^T, bf16 in,
// bf16 out, fp32 accumulate — the forward projection shape (y = x · Wᵀ,
// both operands K-contiguous).
//
// Structure = the CDNA4 guide's "step-3" 128² tile: BK=64,
// 4 waves × (64×64) output each (4×4 grid of 16×16×32 MFMA fragments),
// global→LDS staging via 16-byte global_load_lds (lane-linear dest, the
// XOR bank swizzle applied on the per-lane SOURCE address and repeated on
// the ds_read side — guide rule 21), two LDS buffers, the 2-phase pipeline
// (issue next tile's glds, then ds_read+MFMA on the current tile, then one
// __syncthreads whose implicit vmcnt drains the in-flight DMA).
//
// Standalone (bench/refcheck via the gemm_nt binding); wired into the
// model only if it beats hipBLASLt on the live shapes.

#include "common.h"

namespace {

using u16 = unsigned short;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NT = 256;        // threads (4 waves)

#define MFMA(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// XOR swizzle: spread the 16B-slot select bits with the row (guide G4);
// involution on byte offsets within one [128][64]-bf16 (16 KiB) tile.
ACCO_DEV unsigned swz(unsigned byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

__global__ __launch_bounds__(NT)
void gemm_nt_kernel(const u16* __restrict__ A, const u16* __restrict__ B,
                    u16* __restrict__ C, int M, int N, int K) {
  // grid: (N/BN, M/BM); XCD-aware swizzle on the flat id (guide T1)
  const int nbx = N / BN;
  const int nwg = nbx * (M / BM);
  int flat = blockIdx.y * gridDim.x + blockIdx.x;
  {  // bijective %8 XCD remap
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = flat % 8, idx = flat / 8;
    flat = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = flat / nbx, bn = flat % nbx;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4, lc = lane & 15;
  // wave grid 2×2 over the 128×128 tile: each wave 64×64
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;

  // LDS: [2 buffers][A(16KB) + B(16KB)]
  extern __shared__ __attribute__((aligned(16))) u16 smem[];

  const long long lda = K, ldb = K;
  const u16* Ab = A + (long long)(bm * BM) * lda;
  const u16* Bb = B + (long long)(bn * BN) * ldb;

  // glds staging: per wave-instruction 64 lanes deposit 16 B each at
  // [wave-uniform LDS base + lane*16] (guide §5: the dest is lane-linear;
  // the swizzle therefore goes on the per-lane SOURCE address — rule 21).
  // Each 16 KiB image = 16 KiB / 1 KiB = 16 wave-instructions = 4 chunks
  // per wave.
  auto stage = [&](int buf, int kt) {
    const u16* ta = Ab + kt * BK;
    const u16* tb = Bb + kt * BK;
    u16* la = smem + buf * 2 * BM * BK;
    u16* lb = la + BM * BK;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const unsigned base = (wave * 4 + c) * 1024;        // bytes, uniform
      const unsigned lds_off = base + lane * 16;          // this lane's slot
      const unsigned elem = swz(lds_off) / 2;
      const unsigned r = elem / BK, k = elem % BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(ta + (long long)r * lda + k),
          (__attribute__((address_space(3))) unsigned int*)(la + base / 2),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(tb + (long long)r * ldb + k),
          (__attribute__((address_space(3))) unsigned int*)(lb + base / 2),
          16, 0, 0);
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int KT = K / BK;
  stage(0, 0);
  __syncthreads();

  for (int kt = 0; kt < KT; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < KT) stage(cur ^ 1, kt + 1);     // glds in flight under MFMA
    const u16* a_lds = smem + cur * 2 * BM * BK;
    const u16* b_lds = a_lds + BM * BK;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {             // two K=32 steps
      short8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        // A fragment: row = wm + i*16 + lc, k = ks*32 + lg*8
        const unsigned abyte = ((wm + i * 16 + lc) * BK + ks * 32 + lg * 8) * 2;
        af[i] = *reinterpret_cast<const short8*>(
            a_lds + swz(abyte) / 2);
        const unsigned bbyte = ((wn + i * 16 + lc) * BK + ks * 32 + lg * 8) * 2;
        bf[i] = *reinterpret_cast<const short8*>(
            b_lds + swz(bbyte) / 2);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = MFMA(af[i], bf[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();     // implicit vmcnt drains next tile's glds
  }

  // epilogue: C[row][col]: row = wm+i*16+(lg*4+r), col = wn+j*16+lc
  u16* Cb = C + (long long)(bm * BM) * N + bn * BN;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        Cb[(long long)(wm + i * 16 + lg * 4 + r) * N + wn + j * 16 + lc] =
            f32_to_bf16(acc[i][j][r]);
}

}  // namespace

extern "C" void acco_gemm_nt(const void* A, const void* B, void* C, int M,
                             int N, int K, hipStream_t stream) {
  dim3 grid(N / BN, M / BM);
  const int lds = 4 * BM * BK * sizeof(u16);     // 2 bufs × (A+B)
  hipLaunchKernelGGL(gemm_nt_kernel, grid, dim3(NT), lds, stream,
                     (const u16*)A, (const u16*)B, (u16*)C, M, N, K);
}
