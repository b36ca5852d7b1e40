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

namespace {

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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_adamw", &fused_adamw,
        "Fused sharded AdamW (gfx950): cast+scale+AdamW+bf16 writeout; "
        "commit=false = ACCO tentative step");
  m.attr("_gfx950") = true;
}
