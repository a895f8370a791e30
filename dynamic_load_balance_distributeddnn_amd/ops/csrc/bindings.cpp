// Python bindings for the gfx950 kernel set (_dlb_kernels).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

extern "C" void dlb_sgd_momentum(float* p, const float* g, float* m,
                                 float lr, float mu, long n,
                                 hipStream_t stream);

static void sgd_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                         double lr, double mu) {
  TORCH_CHECK(p.is_cuda() && g.is_cuda() && m.is_cuda(), "expects GPU tensors");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous());
  TORCH_CHECK(p.scalar_type() == torch::kFloat32, "fp32 master weights only");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel());
  auto stream = at::hip::getCurrentHIPStream();
  dlb_sgd_momentum(p.data_ptr<float>(), g.data_ptr<float>(),
                   m.data_ptr<float>(), (float)lr, (float)mu, p.numel(),
                   stream.stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sgd_momentum", &sgd_momentum,
        "Fused SGD momentum step over flat arenas (gfx950)");
}
