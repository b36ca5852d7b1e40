// Shared helpers for the acco_amd gfx950 (CDNA4) HIP kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define ACCO_DEV __device__ __forceinline__

// wave64 is the CDNA scheduling quantum; hard-coded per the CDNA4 guide.
constexpr int kWave = 64;

ACCO_DEV float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}

ACCO_DEV unsigned short f32_to_bf16(float f) {
  // __float2bfloat16 is round-to-nearest-even on ROCm (identical to
  // PyTorch's cast and to the manual +0x7FFF/odd-bit twiddle this
  // replaced); the compiler pairs adjacent casts into v_cvt_pk_bf16_f32 —
  // ~4 VALU ops fewer per element in pack-heavy epilogues
  union { __hip_bfloat16 b; unsigned short u; } c;
  c.b = __float2bfloat16(f);
  return c.u;
}

// Grid sizing for memory-bound elementwise kernels (guide Guideline 11):
// cap blocks and grid-stride the rest.
inline int elementwise_grid(long long n_items, int block, int cap = 2048) {
  long long g = (n_items + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}
