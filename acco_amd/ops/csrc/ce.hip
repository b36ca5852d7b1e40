// Fused shifted causal-LM cross-entropy for gfx950 (SURVEY.md §2.5 K1 loss
// + K9). The eager path materializes fp32 log-softmax over [B,S,V] (vocab
// 50k: ~1.6 GB of HBM traffic per cast); this kernel does one online
// logsumexp pass over bf16 logits (fwd) and one fused softmax-minus-onehot
// pass (bwd), both at HBM-roofline.
//
// Forward: one 256-thread block per token (b, s<S-1), label = labels[b,s+1];
// emits per-token lse into [B,S] fp32 (bwd recompute support), and
// atomicAdds {loss_sum, n_valid} into a 2-elem fp32 accumulator.
// ignore_index = -100 tokens contribute nothing.

#include "common.h"

namespace {

using u16 = unsigned short;
constexpr int BLOCK = 256;

// epsilon != 0 adds HF-LabelSmoother semantics (SURVEY.md §2.5 K9;
// reference utils/trainer_utils.py:862-902): per valid row,
// loss = (1-eps)·(lse - x[label]) + eps·(lse - Σ_v x_v / V).
__global__ void ce_fwd_kernel(const u16* __restrict__ logits,  // [T, V]
                              const long long* __restrict__ labels,  // [T]
                              float* __restrict__ lse,         // [T]
                              float* __restrict__ acc,         // {loss, n}
                              long long T, int S, int V, float epsilon) {
  __shared__ float lds_m[8];
  __shared__ float lds_s[8];
  __shared__ float lds_x[8];
  for (long long t = blockIdx.x; t < T; t += gridDim.x) {
    const int s_pos = (int)(t % S);
    long long label = -100;
    if (s_pos < S - 1) label = labels[t + 1];
    if (label < 0) {
      if (threadIdx.x == 0) lse[t] = 0.0f;
      continue;
    }
    const u16* row = logits + t * (long long)V;
    // online logsumexp over the row, vec8
    float m = -3.4e38f, sum = 0.0f, xsum = 0.0f;
    const int nv = V / 8;
    for (int c = threadIdx.x; c < nv; c += BLOCK) {
      ushort4 a = reinterpret_cast<const ushort4*>(row)[2 * c];
      ushort4 b = reinterpret_cast<const ushort4*>(row)[2 * c + 1];
      float f[8] = {bf16_to_f32(a.x), bf16_to_f32(a.y), bf16_to_f32(a.z),
                    bf16_to_f32(a.w), bf16_to_f32(b.x), bf16_to_f32(b.y),
                    bf16_to_f32(b.z), bf16_to_f32(b.w)};
      if (epsilon != 0.0f)
        xsum += ((f[0] + f[1]) + (f[2] + f[3])) +
                ((f[4] + f[5]) + (f[6] + f[7]));
      // chunk max first, one rescale of the running sum per 8 elements —
      // replaces the per-element branchy online update (8 dependent
      // branch+rescale chains) with independent exps. The 8 exps combine as
      // a TREE: without fast-math the naive `sum += e[k]` loop is a strictly
      // ordered 9-deep dependent fp chain per chunk; the tree cuts the
      // cross-chunk dependency to rescale-mul + one add.
      float cm = fmaxf(fmaxf(fmaxf(f[0], f[1]), fmaxf(f[2], f[3])),
                       fmaxf(fmaxf(f[4], f[5]), fmaxf(f[6], f[7])));
      const float mn = fmaxf(m, cm);
      float e[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) e[k] = __expf(f[k] - mn);
      const float t = ((e[0] + e[1]) + (e[2] + e[3])) +
                      ((e[4] + e[5]) + (e[6] + e[7]));
      sum = sum * __expf(m - mn) + t;
      m = mn;
    }
    for (int tail = nv * 8 + threadIdx.x; tail < V; tail += BLOCK) {
      float x = bf16_to_f32(row[tail]);
      if (x > m) { sum *= __expf(m - x); m = x; }
      sum += __expf(x - m);
      if (epsilon != 0.0f) xsum += x;
    }
    // wave-reduce the (m, sum) pairs (+ xsum for the smoothing term)
    for (int off = 32; off > 0; off >>= 1) {
      float mo = __shfl_down(m, off, 64);
      float so = __shfl_down(sum, off, 64);
      float mn = fmaxf(m, mo);
      sum = sum * __expf(m - mn) + so * __expf(mo - mn);
      m = mn;
      if (epsilon != 0.0f) xsum += __shfl_down(xsum, off, 64);
    }
    const int wave = threadIdx.x / 64;
    if ((threadIdx.x & 63) == 0) {
      lds_m[wave] = m; lds_s[wave] = sum; lds_x[wave] = xsum;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float M = lds_m[0], Ssum = lds_s[0], Xsum = lds_x[0];
      for (int w2 = 1; w2 < BLOCK / 64; ++w2) {
        float mo = lds_m[w2], so = lds_s[w2];
        float mn = fmaxf(M, mo);
        Ssum = Ssum * __expf(M - mn) + so * __expf(mo - mn);
        M = mn;
        Xsum += lds_x[w2];
      }
      const float l = M + __logf(Ssum);
      lse[t] = l;
      float gold = bf16_to_f32(row[label]);
      if (epsilon != 0.0f)
        gold = (1.0f - epsilon) * gold + epsilon * (Xsum / (float)V);
      // SHARDED accumulator: 16k per-row atomicAdds on one L2 cell
      // serialize at ~88/us and were the kernel's real bound (PMC: 87%
      // WAIT_ANY, loads already dwordx4); 128 shards cut the contention
      // 128x and the wrapper reduces [128,2] with one tiny sum
      float* a = acc + 2 * (blockIdx.x & 127);
      atomicAdd(a, l - gold);
      atomicAdd(a + 1, 1.0f);
    }
    __syncthreads();
  }
}

// dlogits[t, v] = scale * (exp(x - lse[t]) - (1-eps)·1[v == label] - eps/V)
//                 (valid tokens; eps = 0 is plain CE)
//              = 0                                           (otherwise)
__global__ void ce_bwd_kernel(const u16* __restrict__ logits,
                              const long long* __restrict__ labels,
                              const float* __restrict__ lse,
                              u16* __restrict__ dlogits,
                              const float* __restrict__ acc,  // n_valid at [1]
                              float dloss,                    // upstream grad
                              const float* __restrict__ dloss_dev,  // device override (avoids a D2H sync in backward)
                              long long T, int S, int V, float epsilon) {
  const float dl = (dloss_dev != nullptr) ? *dloss_dev : dloss;
  const float scale = dl / fmaxf(acc[1], 1.0f);
  const float c_lab = 1.0f - epsilon;       // onehot weight
  const float c_uni = epsilon / (float)V;   // uniform smoothing weight
  for (long long t = blockIdx.x; t < T; t += gridDim.x) {
    const int s_pos = (int)(t % S);
    long long label = -100;
    if (s_pos < S - 1) label = labels[t + 1];
    const u16* row = logits + t * (long long)V;
    u16* drow = dlogits + t * (long long)V;
    const int nv = V / 8;
    if (label < 0) {
      const ushort4 z = make_ushort4(0, 0, 0, 0);
      for (int c = threadIdx.x; c < nv; c += BLOCK) {
        reinterpret_cast<ushort4*>(drow)[2 * c] = z;
        reinterpret_cast<ushort4*>(drow)[2 * c + 1] = z;
      }
      for (int tail = nv * 8 + threadIdx.x; tail < V; tail += BLOCK)
        drow[tail] = 0;
      continue;
    }
    const float l = lse[t];
    for (int c = threadIdx.x; c < nv; c += BLOCK) {
      ushort4 a = reinterpret_cast<const ushort4*>(row)[2 * c];
      ushort4 b = reinterpret_cast<const ushort4*>(row)[2 * c + 1];
      float f[8] = {bf16_to_f32(a.x), bf16_to_f32(a.y), bf16_to_f32(a.z),
                    bf16_to_f32(a.w), bf16_to_f32(b.x), bf16_to_f32(b.y),
                    bf16_to_f32(b.z), bf16_to_f32(b.w)};
      u16 o[8];
      const int base = c * 8;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float p = __expf(f[k] - l) - c_uni;
        if ((long long)(base + k) == label) p -= c_lab;
        o[k] = f32_to_bf16(p * scale);
      }
      reinterpret_cast<ushort4*>(drow)[2 * c] = make_ushort4(o[0], o[1], o[2], o[3]);
      reinterpret_cast<ushort4*>(drow)[2 * c + 1] = make_ushort4(o[4], o[5], o[6], o[7]);
    }
    for (int tail = nv * 8 + threadIdx.x; tail < V; tail += BLOCK) {
      float p = __expf(bf16_to_f32(row[tail]) - l) - c_uni;
      if ((long long)tail == label) p -= c_lab;
      drow[tail] = f32_to_bf16(p * scale);
    }
  }
}

}  // namespace

extern "C" {

void acco_ce_fwd(const void* logits, const long long* labels, float* lse,
                 float* acc, long long T, int S, int V, float epsilon,
                 hipStream_t s) {
  int grid = (int)((T < 8192) ? T : 8192);
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid), dim3(BLOCK), 0, s,
                     (const u16*)logits, labels, lse, acc, T, S, V, epsilon);
}

void acco_ce_bwd(const void* logits, const long long* labels,
                 const float* lse, void* dlogits, const float* acc,
                 float dloss, const float* dloss_dev, long long T, int S,
                 int V, float epsilon, hipStream_t s) {
  int grid = (int)((T < 8192) ? T : 8192);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid), dim3(BLOCK), 0, s,
                     (const u16*)logits, labels, lse, (u16*)dlogits, acc,
                     dloss, dloss_dev, T, S, V, epsilon);
}

}  // extern "C"
