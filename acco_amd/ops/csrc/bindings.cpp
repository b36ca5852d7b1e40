// Python bindings for the acco_amd gfx950 HIP kernels.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime.h>

extern "C" void acco_fused_adamw_launch(
    void* p, const void* g, void* m, void* v, void* out,
    const float* scale_dev,
    long long n, bool buf_is_bf16, bool commit,
    float scale, float lr, float beta1, float beta2, float eps,
    float weight_decay, long long step_plus_1, hipStream_t stream);
extern "C" void acco_swiglu_fwd(const void*, const void*, void*, long long,
                                int, long long, hipStream_t);
extern "C" void acco_swiglu_bwd(const void*, const void*, const void*, void*,
                                void*, long long, int, long long,
                                hipStream_t);
extern "C" void acco_gelu_fwd(const void*, void*, long long, hipStream_t);
extern "C" void acco_gelu_bwd(const void*, const void*, void*, long long,
                              hipStream_t);
extern "C" void acco_rmsnorm_fwd(const void*, const void*, void*, void*,
                                 const void*, void*,
                                 long long, int, float, hipStream_t);
extern "C" void acco_rmsnorm_bwd(const void*, const void*, const void*,
                                 const void*, void*, void*, const void*,
                                 long long, int, hipStream_t);
extern "C" int acco_norm_bwd_grid(long long R, int D);
extern "C" void acco_colsum(const void*, float*, long long, int, bool,
                            hipStream_t);
extern "C" void acco_layernorm_fwd(const void*, const void*, const void*,
                                   void*, void*, void*, const void*, void*,
                                   long long, int, float, hipStream_t);
extern "C" void acco_layernorm_bwd(const void*, const void*, const void*,
                                   const void*, const void*, void*, void*,
                                   void*, const void*, long long, int,
                                   hipStream_t);
extern "C" void acco_rope(const void*, void*, const float*, const float*,
                          long long, int, int, int, bool, long long,
                          hipStream_t);
extern "C" void acco_ce_fwd(const void*, const long long*, float*, float*,
                            long long, int, int, float, hipStream_t);
extern "C" void acco_ce_bwd(const void*, const long long*, const float*,
                            void*, const float*, float, const float*,
                            long long, int, int, float, hipStream_t);
extern "C" void acco_gemm_nt(const void*, const void*, void*, int, int,
                             int, hipStream_t);
extern "C" void acco_attn_fwd(const void*, const void*, const void*, void*,
                              float*, int, int, int, int, int, float, int,
                              hipStream_t);
extern "C" void acco_attn_fwd32(const void*, const void*, const void*, void*,
                                float*, int, int, int, int, int, float, int,
                                long long, long long, long long, hipStream_t);
extern "C" void acco_attn_bwd32_dq(const void*, const void*, const void*,
                                   const void*, const float*, const float*,
                                   void*, int, int, int, int, int, float,
                                   int, long long, long long, long long,
                                   long long, hipStream_t);
extern "C" void acco_attn_bwd32_dkv(const void*, const void*, const void*,
                                    const void*, const float*, const float*,
                                    void*, void*, int, int, int, int, int,
                                    float, int, long long, long long,
                                    long long, hipStream_t);
extern "C" void acco_attn_bwd_dq(const void*, const void*, const void*,
                                 const void*, const float*, const float*,
                                 void*, int, int, int, int, int, float, int,
                                 hipStream_t);
extern "C" void acco_attn_delta(const void*, const void*, float*, long long,
                                int, int, int, hipStream_t);
extern "C" void acco_attn_gqa_reduce(const void*, const void*, void*,
                                     long long, int, int, int, long long,
                                     long long, long long, hipStream_t);
extern "C" void acco_attn_bwd_dkv(const void*, const void*, const void*,
                                  const void*, const float*, const float*,
                                  void*, void*, int, int, int, int, int,
                                  float, int, hipStream_t);

namespace {

using u16 = unsigned short;

void fused_adamw(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                 int64_t step, double lr, double beta1, double beta2,
                 double eps, double weight_decay, double scale,
                 at::Tensor scale_dev, at::Tensor out, bool commit) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat && p.is_contiguous());
  TORCH_CHECK(m.sizes() == p.sizes() && v.sizes() == p.sizes());
  TORCH_CHECK(g.numel() == p.numel() && g.is_contiguous());
  const bool bf16 = g.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 || g.scalar_type() == at::kFloat,
              "grad must be bf16 or fp32");
  void* out_ptr = nullptr;
  if (out.numel() > 0) {
    TORCH_CHECK(out.numel() == p.numel() && out.is_contiguous());
    TORCH_CHECK(out.scalar_type() == g.scalar_type(),
                "out dtype must match grad (com-buffer) dtype");
    out_ptr = out.data_ptr();
  }
  const float* sd = nullptr;
  if (scale_dev.numel() > 0) {
    TORCH_CHECK(scale_dev.scalar_type() == at::kFloat && scale_dev.is_cuda());
    sd = scale_dev.data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  acco_fused_adamw_launch(p.data_ptr(), g.data_ptr(), m.data_ptr(),
                          v.data_ptr(), out_ptr, sd, (long long)p.numel(),
                          bf16, commit, (float)scale, (float)lr, (float)beta1,
                          (float)beta2, (float)eps, (float)weight_decay,
                          step + 1, stream.stream());
}

at::Tensor colsum(at::Tensor in) {
  TORCH_CHECK(in.dim() == 2 && in.is_contiguous() && in.is_cuda());
  TORCH_CHECK(in.scalar_type() == at::kBFloat16 ||
              in.scalar_type() == at::kFloat);
  const long long P = in.size(0);
  const int D = (int)in.size(1);
  auto out = at::zeros({D}, in.options().dtype(at::kFloat));
  acco_colsum(in.data_ptr(), out.data_ptr<float>(), P, D,
              in.scalar_type() == at::kBFloat16,
              at::hip::getCurrentHIPStream().stream());
  return out;
}

#define CHECK_BF16_CONTIG(t) \
  TORCH_CHECK((t).is_cuda() && (t).scalar_type() == at::kBFloat16 && \
              (t).is_contiguous(), #t " must be contiguous CUDA bf16")

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---- SwiGLU / gelu_new
at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u) {
  CHECK_BF16_CONTIG(g); CHECK_BF16_CONTIG(u);
  TORCH_CHECK(g.numel() == u.numel() && g.numel() % 8 == 0);
  const int I = (int)g.size(-1);
  TORCH_CHECK(I % 8 == 0);
  auto out = at::empty_like(g);
  acco_swiglu_fwd(g.data_ptr(), u.data_ptr(), out.data_ptr(), g.numel(),
                  I, I / 8, cur_stream());
  return out;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dout, at::Tensor g, at::Tensor u) {
  CHECK_BF16_CONTIG(dout); CHECK_BF16_CONTIG(g); CHECK_BF16_CONTIG(u);
  const int I = (int)g.size(-1);
  auto dg = at::empty_like(g);
  auto du = at::empty_like(u);
  acco_swiglu_bwd(dout.data_ptr(), g.data_ptr(), u.data_ptr(), dg.data_ptr(),
                  du.data_ptr(), g.numel(), I, I / 8, cur_stream());
  return {dg, du};
}

// SwiGLU over the packed [rows, 2I] fused gate_up projection output:
// out = silu(gu[:, :I]) * gu[:, I:]; backward emits dgu in one pass
// (no torch.split → no contiguous copies forward, no cat backward).
at::Tensor swiglu_packed_fwd(at::Tensor gu) {
  CHECK_BF16_CONTIG(gu);
  const int twoI = (int)gu.size(-1);
  TORCH_CHECK(twoI % 16 == 0);
  const int I = twoI / 2;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = at::empty(sizes, gu.options());
  const u16* base = (const u16*)gu.data_ptr();
  acco_swiglu_fwd(base, base + I, out.data_ptr(), gu.numel() / 2, I,
                  twoI / 8, cur_stream());
  return out;
}

at::Tensor swiglu_packed_bwd(at::Tensor dout, at::Tensor gu) {
  CHECK_BF16_CONTIG(dout); CHECK_BF16_CONTIG(gu);
  const int twoI = (int)gu.size(-1);
  const int I = twoI / 2;
  auto dgu = at::empty_like(gu);
  const u16* base = (const u16*)gu.data_ptr();
  u16* dbase = (u16*)dgu.data_ptr();
  acco_swiglu_bwd(dout.data_ptr(), base, base + I, dbase, dbase + I,
                  gu.numel() / 2, I, twoI / 8, cur_stream());
  return dgu;
}

at::Tensor gelu_fwd(at::Tensor x) {
  CHECK_BF16_CONTIG(x);
  TORCH_CHECK(x.numel() % 8 == 0);
  auto out = at::empty_like(x);
  acco_gelu_fwd(x.data_ptr(), out.data_ptr(), x.numel(), cur_stream());
  return out;
}

at::Tensor gelu_bwd(at::Tensor dout, at::Tensor x) {
  CHECK_BF16_CONTIG(dout); CHECK_BF16_CONTIG(x);
  auto dx = at::empty_like(x);
  acco_gelu_bwd(dout.data_ptr(), x.data_ptr(), dx.data_ptr(), x.numel(),
                cur_stream());
  return dx;
}

// ---- RMSNorm
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(w);
  const int D = (int)w.numel();
  TORCH_CHECK(D % 8 == 0 && D <= 16384 && x.numel() % D == 0);
  const long long R = x.numel() / D;
  auto y = at::empty_like(x);
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  acco_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), rstd.data_ptr(),
                   nullptr, nullptr, R, D, (float)eps, cur_stream());
  return {y, rstd};
}

// fused residual add + RMSNorm: s = bf16(x + res), y = rmsnorm(s)·w
std::vector<at::Tensor> add_rmsnorm_fwd(at::Tensor x, at::Tensor res,
                                        at::Tensor w, double eps) {
  CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(res); CHECK_BF16_CONTIG(w);
  const int D = (int)w.numel();
  TORCH_CHECK(D % 8 == 0 && D <= 16384 && x.numel() % D == 0);
  TORCH_CHECK(res.numel() == x.numel());
  const long long R = x.numel() / D;
  auto y = at::empty_like(x);
  auto sum_out = at::empty_like(x);
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  acco_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), rstd.data_ptr(),
                   res.data_ptr(), sum_out.data_ptr(), R, D, (float)eps,
                   cur_stream());
  return {y, sum_out, rstd};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor rstd,
                                    c10::optional<at::Tensor> dadd_opt) {
  at::Tensor dadd = dadd_opt.value_or(at::Tensor());
  CHECK_BF16_CONTIG(dy); CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(w);
  const int D = (int)w.numel();
  const long long R = x.numel() / D;
  auto dx = at::empty_like(x);
  const int grid = acco_norm_bwd_grid(R, D);
  auto dw_part = at::empty({grid, D}, x.options().dtype(at::kFloat));
  acco_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                   rstd.data_ptr(), dx.data_ptr(), dw_part.data_ptr(),
                   (dadd.defined() && dadd.numel() > 0) ? dadd.data_ptr() : nullptr, R, D,
                   cur_stream());
  return {dx, colsum(dw_part)};
}

// ---- LayerNorm
std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                                      double eps) {
  CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(w); CHECK_BF16_CONTIG(b);
  const int D = (int)w.numel();
  TORCH_CHECK(D % 8 == 0 && D <= 16384 && x.numel() % D == 0);
  const long long R = x.numel() / D;
  auto y = at::empty_like(x);
  auto mean = at::empty({R}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  acco_layernorm_fwd(x.data_ptr(), w.data_ptr(), b.data_ptr(), y.data_ptr(),
                     mean.data_ptr(), rstd.data_ptr(), nullptr, nullptr, R,
                     D, (float)eps, cur_stream());
  return {y, mean, rstd};
}

// fused residual add + LayerNorm: s = bf16(x + res), y = ln(s)·w + b
std::vector<at::Tensor> add_layernorm_fwd(at::Tensor x, at::Tensor res,
                                          at::Tensor w, at::Tensor b,
                                          double eps) {
  CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(res);
  CHECK_BF16_CONTIG(w); CHECK_BF16_CONTIG(b);
  const int D = (int)w.numel();
  TORCH_CHECK(D % 8 == 0 && D <= 16384 && x.numel() % D == 0);
  TORCH_CHECK(res.numel() == x.numel());
  const long long R = x.numel() / D;
  auto y = at::empty_like(x);
  auto sum_out = at::empty_like(x);
  auto mean = at::empty({R}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  acco_layernorm_fwd(x.data_ptr(), w.data_ptr(), b.data_ptr(), y.data_ptr(),
                     mean.data_ptr(), rstd.data_ptr(), res.data_ptr(),
                     sum_out.data_ptr(), R, D, (float)eps, cur_stream());
  return {y, sum_out, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd,
                                      c10::optional<at::Tensor> dadd_opt) {
  at::Tensor dadd = dadd_opt.value_or(at::Tensor());
  CHECK_BF16_CONTIG(dy); CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(w);
  const int D = (int)w.numel();
  const long long R = x.numel() / D;
  auto dx = at::empty_like(x);
  const int grid = acco_norm_bwd_grid(R, D);
  auto dw_part = at::empty({grid, D}, x.options().dtype(at::kFloat));
  auto db_part = at::empty({grid, D}, x.options().dtype(at::kFloat));
  acco_layernorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     mean.data_ptr(), rstd.data_ptr(), dx.data_ptr(),
                     dw_part.data_ptr(), db_part.data_ptr(),
                     (dadd.defined() && dadd.numel() > 0) ? dadd.data_ptr() : nullptr, R, D,
                     cur_stream());
  return {dx, colsum(dw_part), colsum(db_part)};
}

// ---- RoPE ([B, S, H, D] contiguous)
at::Tensor rope_fwd(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                    bool bwd) {
  CHECK_BF16_CONTIG(x);
  TORCH_CHECK(x.dim() == 4);
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && cos_t.is_contiguous());
  TORCH_CHECK(sin_t.scalar_type() == at::kFloat && sin_t.is_contiguous());
  const long long B = x.size(0);
  const int S = (int)x.size(1), H = (int)x.size(2), D = (int)x.size(3);
  TORCH_CHECK(D % 8 == 0 && cos_t.size(0) >= S && cos_t.size(1) == D);
  auto y = at::empty_like(x);
  acco_rope(x.data_ptr(), y.data_ptr(), cos_t.data_ptr<float>(),
            sin_t.data_ptr<float>(), B, S, H, D, bwd,
            (long long)H * D, cur_stream());
  return y;
}

// ---- fused shifted causal-LM CE
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor labels,
                               double epsilon) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(logits.dim() == 3 && labels.dim() == 2);
  TORCH_CHECK(labels.scalar_type() == at::kLong && labels.is_contiguous());
  const long long Bn = logits.size(0);
  const int S = (int)logits.size(1), V = (int)logits.size(2);
  const long long T = Bn * S;
  auto lse = at::empty({T}, logits.options().dtype(at::kFloat));
  // 128 accumulator shards (atomic-contention fix in ce.hip) reduced here
  auto accs = at::zeros({128, 2}, logits.options().dtype(at::kFloat));
  acco_ce_fwd(logits.data_ptr(),
              reinterpret_cast<const long long*>(labels.data_ptr<int64_t>()),
              lse.data_ptr<float>(), accs.data_ptr<float>(), T, S, V,
              (float)epsilon, cur_stream());
  return {accs.sum(0), lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor labels, at::Tensor lse,
                  at::Tensor acc, double dloss, double epsilon) {
  CHECK_BF16_CONTIG(logits);
  const long long Bn = logits.size(0);
  const int S = (int)logits.size(1), V = (int)logits.size(2);
  const long long T = Bn * S;
  auto dlogits = at::empty_like(logits);
  acco_ce_bwd(logits.data_ptr(),
              reinterpret_cast<const long long*>(labels.data_ptr<int64_t>()),
              lse.data_ptr<float>(), dlogits.data_ptr(),
              acc.data_ptr<float>(), (float)dloss, nullptr, T, S, V,
              (float)epsilon, cur_stream());
  return dlogits;
}

// dloss read from a 1-elem fp32 device tensor: no D2H sync in backward
at::Tensor ce_bwd_dev(at::Tensor logits, at::Tensor labels, at::Tensor lse,
                      at::Tensor acc, at::Tensor dloss_dev, double epsilon) {
  CHECK_BF16_CONTIG(logits);
  TORCH_CHECK(dloss_dev.is_cuda() && dloss_dev.scalar_type() == at::kFloat &&
              dloss_dev.numel() == 1);
  const long long Bn = logits.size(0);
  const int S = (int)logits.size(1), V = (int)logits.size(2);
  const long long T = Bn * S;
  auto dlogits = at::empty_like(logits);
  acco_ce_bwd(logits.data_ptr(),
              reinterpret_cast<const long long*>(labels.data_ptr<int64_t>()),
              lse.data_ptr<float>(), dlogits.data_ptr(),
              acc.data_ptr<float>(), 0.0f, dloss_dev.data_ptr<float>(), T, S,
              V, (float)epsilon, cur_stream());
  return dlogits;
}

// ---- flash attention ([B, S, H, D] layout, D in {64, 128}, S % 64 == 0)
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double scale, int64_t window) {
  CHECK_BF16_CONTIG(q); CHECK_BF16_CONTIG(k); CHECK_BF16_CONTIG(v);
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  const int B = (int)q.size(0), S = (int)q.size(1), H = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  TORCH_CHECK(S % 64 == 0 && (D == 64 || D == 128) && H % Hkv == 0);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  acco_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                lse.data_ptr<float>(), B, S, H, Hkv, D, (float)scale,
                (int)window, cur_stream());
  return {o, lse};
}

at::Tensor attn_delta(at::Tensor dO, at::Tensor o) {
  CHECK_BF16_CONTIG(dO); CHECK_BF16_CONTIG(o);
  const long long B = dO.size(0);
  const int S = (int)dO.size(1), H = (int)dO.size(2), D = (int)dO.size(3);
  TORCH_CHECK(D % 8 == 0);
  auto delta = at::empty({B, H, S}, dO.options().dtype(at::kFloat));
  acco_attn_delta(dO.data_ptr(), o.data_ptr(), delta.data_ptr<float>(), B, S,
                  H, D, cur_stream());
  return delta;
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor dO, at::Tensor lse,
                                 at::Tensor delta, double scale,
                                 int64_t window) {
  CHECK_BF16_CONTIG(q); CHECK_BF16_CONTIG(k); CHECK_BF16_CONTIG(v);
  CHECK_BF16_CONTIG(dO);
  const int B = (int)q.size(0), S = (int)q.size(1), H = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  TORCH_CHECK(delta.scalar_type() == at::kFloat && delta.is_contiguous());
  auto dq = at::empty_like(q);
  // dk/dv are emitted per QUERY head; the Python wrapper group-reduces
  auto dk = at::empty_like(q);
  auto dv = at::empty_like(q);
  acco_attn_bwd_dq(q.data_ptr(), k.data_ptr(), v.data_ptr(), dO.data_ptr(),
                   lse.data_ptr<float>(), delta.data_ptr<float>(),
                   dq.data_ptr(), B, S, H, Hkv, D, (float)scale, (int)window,
                   cur_stream());
  acco_attn_bwd_dkv(q.data_ptr(), k.data_ptr(), v.data_ptr(), dO.data_ptr(),
                    lse.data_ptr<float>(), delta.data_ptr<float>(),
                    dk.data_ptr(), dv.data_ptr(), B, S, H, Hkv, D,
                    (float)scale, (int)window, cur_stream());
  return {dq, dk, dv};
}

// GQA group-reduce of per-query-head dK/dV straight into the packed grad
void attn_gqa_reduce(at::Tensor dkq, at::Tensor dvq, at::Tensor dqkv,
                     int64_t Hkv, int64_t rep, int64_t D, int64_t k_off,
                     int64_t v_off) {
  CHECK_BF16_CONTIG(dkq); CHECK_BF16_CONTIG(dvq); CHECK_BF16_CONTIG(dqkv);
  TORCH_CHECK(D % 8 == 0);
  const long long T = dkq.size(0) * dkq.size(1);
  const long long W = dqkv.size(2);
  acco_attn_gqa_reduce(dkq.data_ptr(), dvq.data_ptr(), dqkv.data_ptr(), T,
                       (int)Hkv, (int)rep, (int)D, W, k_off, v_off,
                       cur_stream());
}

// ---- experimental NT GEMM (C = A · B^T, bf16; bench/refcheck only)
at::Tensor gemm_nt(at::Tensor A, at::Tensor B) {
  CHECK_BF16_CONTIG(A); CHECK_BF16_CONTIG(B);
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(0);
  TORCH_CHECK(B.size(1) == K && M % 128 == 0 && N % 128 == 0 && K % 64 == 0);
  auto C = at::empty({M, N}, A.options());
  acco_gemm_nt(A.data_ptr(), B.data_ptr(), C.data_ptr(), M, N, K,
               cur_stream());
  return C;
}

// ---- packed-QKV attention path (fused projection output consumed and
// grad produced with ZERO split/cat copies). qkv: [B, S, W] contiguous with
// W = (H + 2*Hkv)*D, sections in q|k|v order; D=64, S%256==0 (v4 kernels).
std::vector<at::Tensor> rope_packed(at::Tensor src, at::Tensor dst,
                                    at::Tensor cos_t, at::Tensor sin_t,
                                    int64_t off, int64_t Hsec, int64_t D,
                                    bool bwd) {
  CHECK_BF16_CONTIG(src); CHECK_BF16_CONTIG(dst);
  const long long B = src.size(0);
  const int S = (int)src.size(1);
  const long long W = src.size(2);
  acco_rope((const u16*)src.data_ptr() + off,
            (u16*)dst.data_ptr() + off, cos_t.data_ptr<float>(),
            sin_t.data_ptr<float>(), B, S, (int)Hsec, (int)D, bwd, W,
            cur_stream());
  return {dst};
}

std::vector<at::Tensor> attn_fwd_packed(at::Tensor qkv, int64_t H,
                                        int64_t Hkv, int64_t D, double scale,
                                        int64_t window, int64_t q_off,
                                        int64_t k_off, int64_t v_off) {
  CHECK_BF16_CONTIG(qkv);
  const int B = (int)qkv.size(0), S = (int)qkv.size(1);
  const long long W = qkv.size(2);
  TORCH_CHECK(W >= (H + 2 * Hkv) * D && (D == 64 || D == 128) && S % 256 == 0);
  const u16* base = (const u16*)qkv.data_ptr();
  auto o = at::empty({B, S, H * D}, qkv.options());
  auto lse = at::empty({B, H, S}, qkv.options().dtype(at::kFloat));
  acco_attn_fwd32(base + q_off, base + k_off, base + v_off, o.data_ptr(),
                  lse.data_ptr<float>(), B, S, (int)H, (int)Hkv, (int)D,
                  (float)scale, (int)window, W, W, H * D, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd_packed(at::Tensor qkv, at::Tensor dO,
                                        at::Tensor lse, at::Tensor delta,
                                        at::Tensor dqkv, int64_t H,
                                        int64_t Hkv, int64_t D, double scale,
                                        int64_t window, int64_t q_off,
                                        int64_t k_off, int64_t v_off) {
  CHECK_BF16_CONTIG(qkv); CHECK_BF16_CONTIG(dO); CHECK_BF16_CONTIG(dqkv);
  const int B = (int)qkv.size(0), S = (int)qkv.size(1);
  const long long W = qkv.size(2);
  const u16* base = (const u16*)qkv.data_ptr();
  u16* dbase = (u16*)dqkv.data_ptr();
  // dq straight into the packed grad (stride W); dk/dv per-QUERY-head temps
  auto dkq = at::empty({B, S, H, D}, qkv.options());
  auto dvq = at::empty({B, S, H, D}, qkv.options());
  acco_attn_bwd32_dq(base + q_off, base + k_off, base + v_off, dO.data_ptr(),
                     lse.data_ptr<float>(), delta.data_ptr<float>(),
                     dbase + q_off, B, S, (int)H, (int)Hkv, (int)D,
                     (float)scale, (int)window, W, W, H * D, W,
                     cur_stream());
  acco_attn_bwd32_dkv(base + q_off, base + k_off, base + v_off,
                      dO.data_ptr(), lse.data_ptr<float>(),
                      delta.data_ptr<float>(), dkq.data_ptr(),
                      dvq.data_ptr(), B, S, (int)H, (int)Hkv, (int)D,
                      (float)scale, (int)window, W, W, H * D, cur_stream());
  return {dkq, dvq};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_adamw", &fused_adamw,
        "Fused sharded AdamW (gfx950): cast+scale+AdamW+bf16 writeout; "
        "commit=false = ACCO tentative step");
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("swiglu_packed_fwd", &swiglu_packed_fwd);
  m.def("swiglu_packed_bwd", &swiglu_packed_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("colsum", &colsum);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd, py::arg("dy"), py::arg("x"),
        py::arg("w"), py::arg("rstd"), py::arg("dadd") = py::none());
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd, py::arg("dy"), py::arg("x"),
        py::arg("w"), py::arg("mean"), py::arg("rstd"),
        py::arg("dadd") = py::none());
  m.def("add_layernorm_fwd", &add_layernorm_fwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("ce_fwd", &ce_fwd, py::arg("logits"), py::arg("labels"),
        py::arg("epsilon") = 0.0);
  m.def("ce_bwd", &ce_bwd, py::arg("logits"), py::arg("labels"),
        py::arg("lse"), py::arg("acc"), py::arg("dloss"),
        py::arg("epsilon") = 0.0);
  m.def("ce_bwd_dev", &ce_bwd_dev, py::arg("logits"), py::arg("labels"),
        py::arg("lse"), py::arg("acc"), py::arg("dloss_dev"),
        py::arg("epsilon") = 0.0);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_delta", &attn_delta);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_gqa_reduce", &attn_gqa_reduce);
  m.def("gemm_nt", &gemm_nt);
  m.def("rope_packed", &rope_packed);
  m.def("attn_fwd_packed", &attn_fwd_packed);
  m.def("attn_bwd_packed", &attn_bwd_packed);
  m.attr("_gfx950") = true;
}
