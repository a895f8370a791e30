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

extern "C" void dlb_gn_fwd(const void* x, void* y, const float* gamma,
                           const float* beta, float* mean, float* rstd,
                           int N, int HW, int C, int G, float eps, int relu,
                           hipStream_t stream);
extern "C" void dlb_gn_bwd(const void* x, const void* dz, void* dx,
                           const float* gamma, const float* beta,
                           const float* mean, const float* rstd, float* dgamma,
                           float* dbeta, int N, int HW, int C, int G, int relu,
                           hipStream_t stream);

// x: [N, HW, C] bf16 contiguous (an NHWC view of a channels_last NCHW
// tensor).  Returns (y, mean, rstd).
static std::vector<torch::Tensor> gn_fwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor beta, int64_t groups,
                                         double eps, bool relu) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "bf16 activations only");
  TORCH_CHECK(gamma.scalar_type() == torch::kFloat32);
  const int N = x.size(0), HW = x.size(1), C = x.size(2);
  TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
  TORCH_CHECK(groups <= 64 && C % groups == 0);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({N, groups}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  auto stream = at::hip::getCurrentHIPStream();
  dlb_gn_fwd(x.data_ptr(), y.data_ptr(), gamma.data_ptr<float>(),
             beta.data_ptr<float>(), mean.data_ptr<float>(),
             rstd.data_ptr<float>(), N, HW, C, (int)groups, (float)eps,
             relu ? 1 : 0, stream.stream());
  return {y, mean, rstd};
}

static std::vector<torch::Tensor> gn_bwd(torch::Tensor x, torch::Tensor dz,
                                         torch::Tensor gamma,
                                         torch::Tensor beta,
                                         torch::Tensor mean,
                                         torch::Tensor rstd, int64_t groups,
                                         bool relu) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
  TORCH_CHECK(dz.is_contiguous() && dz.sizes() == x.sizes());
  const int N = x.size(0), HW = x.size(1), C = x.size(2);
  auto dx = torch::empty_like(x);
  auto dgamma = torch::zeros({C}, x.options().dtype(torch::kFloat32));
  auto dbeta = torch::zeros_like(dgamma);
  auto stream = at::hip::getCurrentHIPStream();
  dlb_gn_bwd(x.data_ptr(), dz.data_ptr(), dx.data_ptr(),
             gamma.data_ptr<float>(), beta.data_ptr<float>(),
             mean.data_ptr<float>(), rstd.data_ptr<float>(),
             dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), N, HW, C,
             (int)groups, relu ? 1 : 0, stream.stream());
  return {dx, dgamma, dbeta};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sgd_momentum", &sgd_momentum,
        "Fused SGD momentum step over flat arenas (gfx950)");
  m.def("gn_fwd", &gn_fwd, "Fused GroupNorm(+ReLU) forward, NHWC bf16");
  m.def("gn_bwd", &gn_bwd, "Fused GroupNorm(+ReLU) backward, NHWC bf16");
}
