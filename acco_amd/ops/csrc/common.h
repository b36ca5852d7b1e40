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
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even, matching PyTorch's float->bfloat16 cast
  unsigned int x = c.i;
  unsigned int rounding_bias = 0x7FFF + ((x >> 16) & 1);
  x += rounding_bias;
  return (unsigned short)(x >> 16);
}

// Grid sizing for memory-bound elementwise kernels (guide Guideline 11):
// cap blocks and grid-stride the rest.
inline int elementwise_grid(long long n_items, int block, int cap = 2048) {
  long long g = (n_items + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}
