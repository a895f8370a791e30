// Python bindings for the gfx950 kernel set (_dlb_kernels).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

extern "C" void dlb_sgd_momentum(float* p, const float* g, float* m,
                                 void* q, float lr, float mu, long n,
                                 hipStream_t stream);

static void sgd_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                         double lr, double mu,
                         c10::optional<torch::Tensor> mirror = c10::nullopt) {
  TORCH_CHECK(p.is_cuda() && g.is_cuda() && m.is_cuda(), "expects GPU tensors");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous());
  TORCH_CHECK(p.scalar_type() == torch::kFloat32, "fp32 master weights only");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel());
  void* q = nullptr;
  if (mirror.has_value()) {
    TORCH_CHECK(mirror->scalar_type() == torch::kBFloat16 &&
                mirror->is_contiguous() && mirror->numel() == p.numel());
    q = mirror->data_ptr();
  }
  auto stream = at::hip::getCurrentHIPStream();
  dlb_sgd_momentum(p.data_ptr<float>(), g.data_ptr<float>(),
                   m.data_ptr<float>(), q, (float)lr, (float)mu, p.numel(),
                   stream.stream());
}

extern "C" void dlb_gn_fwd_segs(const void* const* xs, const int* starts,
                                int nseg, void* y, const float* gamma,
                                const float* beta, float* mean, float* rstd,
                                const void* res,
                                int N, int HW, int C, int G,
                                float eps, int relu, hipStream_t stream);
extern "C" void dlb_gn_bwd_segs(const void* const* xs, const int* starts,
                                int nseg, const void* dz, void* const* dxs,
                                const float* gamma, const float* beta,
                                const float* mean, const float* rstd,
                                float* dgb_part,
                                const void* res, void* dres,
                                int N, int HW, int C, int G, int relu,
                                int accumulate, hipStream_t stream);
extern "C" void dlb_gn_dgb_reduce(const float* part, int N, int C,
                                  float* dgamma, float* dbeta,
                                  hipStream_t stream);
extern "C" void dlb_gn_stats_segs(const void* const* xs, const int* starts,
                                  int nseg, float* mean, float* rstd, int N,
                                  int HW, int C, int G, float eps,
                                  hipStream_t stream);
extern "C" void dlb_gnconv1x1_fwd(const void* const* xs, const int* starts,
                                  int nseg, const float* mean,
                                  const float* rstd, const float* gamma,
                                  const float* beta, const void* w, void* y,
                                  int N, int HW, int C, int G, int Co,
                                  int relu, hipStream_t stream);
extern "C" int dlb_gnconv1x1_wrw(const void* const* xs, const int* starts,
                                 int nseg, const float* mean,
                                 const float* rstd, const float* gamma,
                                 const float* beta, const void* dy, float* dw,
                                 int N, int HW, int C, int G, int Co,
                                 int relu, int splits, hipStream_t stream);

// Segments: [N, HW, Ci] bf16 contiguous views of channels_last tensors
// forming a virtual channel-concat.  Returns (y packed, mean, rstd).
static std::vector<torch::Tensor> gn_fwd(std::vector<torch::Tensor> xs,
                                         torch::Tensor gamma,
                                         torch::Tensor beta, int64_t groups,
                                         double eps, bool relu,
                                         c10::optional<torch::Tensor>
                                             res = c10::nullopt) {
  TORCH_CHECK(!xs.empty() && xs.size() <= 56);
  int C = 0;
  const void* ptrs[56];
  int starts[57];
  for (size_t i = 0; i < xs.size(); ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "bf16 activations only");
    TORCH_CHECK(x.size(2) % 8 == 0);
    ptrs[i] = x.data_ptr();
    starts[i] = C;
    C += (int)x.size(2);
  }
  starts[xs.size()] = C;
  TORCH_CHECK(gamma.scalar_type() == torch::kFloat32);
  const int N = xs[0].size(0), HW = xs[0].size(1);
  TORCH_CHECK(groups <= 64 && C % groups == 0);
  auto y = torch::empty({N, HW, C}, xs[0].options());
  auto mean = torch::empty({N, groups},
                           xs[0].options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  auto stream = at::hip::getCurrentHIPStream();
  const void* resp = nullptr;
  if (res.has_value()) {
    TORCH_CHECK(res->is_contiguous() && res->scalar_type() == torch::kBFloat16
                && res->numel() == (long)N * HW * C,
                "residual must be a packed bf16 [N,HW,C] tensor");
    resp = res->data_ptr();
  }
  dlb_gn_fwd_segs(ptrs, starts, (int)xs.size(), y.data_ptr(),
                  gamma.data_ptr<float>(), beta.data_ptr<float>(),
                  mean.data_ptr<float>(), rstd.data_ptr<float>(), resp, N,
                  HW, C, (int)groups, (float)eps, relu ? 1 : 0,
                  stream.stream());
  return {y, mean, rstd};
}

static std::vector<torch::Tensor> gn_bwd(std::vector<torch::Tensor> xs,
                                         torch::Tensor dz,
                                         torch::Tensor gamma,
                                         torch::Tensor beta,
                                         torch::Tensor mean,
                                         torch::Tensor rstd, int64_t groups,
                                         bool relu,
                                         c10::optional<std::vector<torch::Tensor>>
                                             dx_accum = c10::nullopt,
                                         c10::optional<torch::Tensor>
                                             dgamma_out = c10::nullopt,
                                         c10::optional<torch::Tensor>
                                             dbeta_out = c10::nullopt,
                                         bool dgb_defer = false,
                                         c10::optional<torch::Tensor>
                                             res = c10::nullopt) {
  // dx_accum: preallocated per-segment grad buffers — the kernel ADDS
  // into them (the dense-stream manual backward), instead of allocating
  // fresh outputs for autograd to sum pairwise.
  // dgamma_out/dbeta_out: pre-ZEROED [C] fp32 buffers (flat-arena grad
  // views; the kernel accumulates into them with atomics).
  TORCH_CHECK(!xs.empty() && xs.size() <= 56);
  int C = 0;
  const void* ptrs[56];
  void* dptrs[56];
  int starts[57];
  const bool acc = dx_accum.has_value();
  std::vector<torch::Tensor> out;
  for (size_t i = 0; i < xs.size(); ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
    ptrs[i] = x.data_ptr();
    starts[i] = C;
    C += (int)x.size(2);
    out.push_back(acc ? (*dx_accum)[i] : torch::empty_like(x));
    TORCH_CHECK(out.back().is_contiguous() &&
                out.back().numel() == x.numel());
    dptrs[i] = out.back().data_ptr();
  }
  starts[xs.size()] = C;
  const int N = xs[0].size(0), HW = xs[0].size(1);
  TORCH_CHECK(dz.is_contiguous() && dz.size(2) == C);
  auto stream = at::hip::getCurrentHIPStream();
  // dgamma/dbeta path: per-sample partial rows [N, 2C] (plain stores in
  // the kernel) + one deterministic column-sum kernel — replaces the
  // contended global-atomic publish (see reduce.hip).  With dgb_defer
  // the caller batches many layers' reductions into one launch
  // (gn_dgb_reduce_multi) and gets the raw partial rows back instead.
  auto part = torch::empty({N, 2 * C},
                           xs[0].options().dtype(torch::kFloat32));
  const void* resp = nullptr;
  void* dresp = nullptr;
  torch::Tensor dres;
  if (res.has_value()) {
    resp = res->data_ptr();
    dres = torch::empty({N, HW, C}, xs[0].options());
    dresp = dres.data_ptr();
  }
  dlb_gn_bwd_segs(ptrs, starts, (int)xs.size(), dz.data_ptr(), dptrs,
                  gamma.data_ptr<float>(), beta.data_ptr<float>(),
                  mean.data_ptr<float>(), rstd.data_ptr<float>(),
                  part.data_ptr<float>(), resp, dresp,
                  N, HW, C, (int)groups, relu ? 1 : 0, acc ? 1 : 0,
                  stream.stream());
  if (res.has_value()) out.push_back(dres);
  if (dgb_defer) {
    out.push_back(part);
    return out;  // [dx_0..dx_{k-1}, (dres,) part]
  }
  torch::Tensor dgamma, dbeta;
  if (dgamma_out.has_value()) {
    TORCH_CHECK(dgamma_out->is_contiguous() && dgamma_out->numel() == C &&
                dbeta_out.has_value() && dbeta_out->is_contiguous() &&
                dbeta_out->numel() == C);
    dgamma = *dgamma_out;
    dbeta = *dbeta_out;
  } else {
    // one zero-fill kernel for both reductions (they were 2 of the ~288
    // fill launches per DenseNet step)
    auto gbuf = torch::zeros({2, C}, xs[0].options().dtype(torch::kFloat32));
    dgamma = gbuf[0];
    dbeta = gbuf[1];
  }
  dlb_gn_dgb_reduce(part.data_ptr<float>(), N, C,
                    dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                    stream.stream());
  out.push_back(dgamma);
  out.push_back(dbeta);
  return out;  // [dx_0..dx_{k-1}, dgamma, dbeta]
}

extern "C" void dlb_gn_dgb_reduce_multi(const void* const* parts,
                                        void* const* dgs, void* const* dbs,
                                        const int* Cs, int nl, int N,
                                        hipStream_t stream);
extern "C" void dlb_chansum(const void* x, float* sum, float* ssq, int N,
                            int HW, int Cs, hipStream_t stream);
extern "C" void dlb_gn_stats_sums(const void* const* sums,
                                  const void* const* ssqs,
                                  const int* starts, int nseg, float* mean,
                                  float* rstd, int N, int HW, int C, int G,
                                  float eps, hipStream_t stream);

// Per-channel (sum, ssq) of one [N, HW, Cs] bf16 segment.
static std::vector<torch::Tensor> chan_sums(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  const int N = (int)x.size(0), HW = (int)x.size(1), Cs = (int)x.size(2);
  auto sum = torch::empty({N, Cs}, x.options().dtype(torch::kFloat32));
  auto ssq = torch::empty_like(sum);
  dlb_chansum(x.data_ptr(), sum.data_ptr<float>(), ssq.data_ptr<float>(),
              N, HW, Cs, at::hip::getCurrentHIPStream().stream());
  return {sum, ssq};
}

// mean/rstd [N, G] from accumulated per-channel sums (newest-first
// segment lists, mirroring the activation segment order).
static std::vector<torch::Tensor> gn_stats_from_sums(
    std::vector<torch::Tensor> sums, std::vector<torch::Tensor> ssqs,
    int64_t groups, int64_t HW, double eps) {
  const int nseg = (int)sums.size();
  TORCH_CHECK(nseg >= 1 && nseg <= 56 && ssqs.size() == sums.size());
  const void* ps[56];
  const void* qs[56];
  int starts[57];
  int C = 0;
  const int N = (int)sums[0].size(0);
  for (int i = 0; i < nseg; ++i) {
    TORCH_CHECK(sums[i].is_contiguous() && ssqs[i].is_contiguous() &&
                sums[i].size(0) == N);
    ps[i] = sums[i].data_ptr();
    qs[i] = ssqs[i].data_ptr();
    starts[i] = C;
    C += (int)sums[i].size(1);
  }
  starts[nseg] = C;
  TORCH_CHECK(C % groups == 0);
  auto mean = torch::empty({N, groups},
                           sums[0].options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  dlb_gn_stats_sums(ps, qs, starts, nseg, mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), N, (int)HW, C, (int)groups,
                    (float)eps, at::hip::getCurrentHIPStream().stream());
  return {mean, rstd};
}

// Batched dgamma/dbeta reduction over many layers' deferred partials
// (one launch; += into the pre-zeroed arena grad views).
static void gn_dgb_reduce_multi(std::vector<torch::Tensor> parts,
                                std::vector<torch::Tensor> dgs,
                                std::vector<torch::Tensor> dbs) {
  const int nl = (int)parts.size();
  TORCH_CHECK(nl >= 1 && nl <= 52 && dgs.size() == parts.size() &&
              dbs.size() == parts.size());
  const void* pp[52];
  void* pg[52];
  void* pb[52];
  int cs[52];
  const int N = (int)parts[0].size(0);
  for (int i = 0; i < nl; ++i) {
    TORCH_CHECK(parts[i].is_contiguous() && parts[i].size(0) == N);
    const int C = (int)(parts[i].size(1) / 2);
    TORCH_CHECK(dgs[i].is_contiguous() && dgs[i].numel() == C &&
                dbs[i].is_contiguous() && dbs[i].numel() == C);
    pp[i] = parts[i].data_ptr();
    pg[i] = dgs[i].data_ptr();
    pb[i] = dbs[i].data_ptr();
    cs[i] = C;
  }
  dlb_gn_dgb_reduce_multi(pp, pg, pb, cs, nl, N,
                          at::hip::getCurrentHIPStream().stream());
}

// Stats-only GroupNorm over the virtual concat (mean/rstd for the fused
// GN->1x1-conv kernels; the normalized activation is never materialized).
static std::vector<torch::Tensor> gn_stats(std::vector<torch::Tensor> xs,
                                           int64_t groups, double eps) {
  TORCH_CHECK(!xs.empty() && xs.size() <= 56);
  int C = 0;
  const void* ptrs[56];
  int starts[57];
  for (size_t i = 0; i < xs.size(); ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(x.size(2) % 8 == 0);
    ptrs[i] = x.data_ptr();
    starts[i] = C;
    C += (int)x.size(2);
  }
  starts[xs.size()] = C;
  const int N = xs[0].size(0), HW = xs[0].size(1);
  TORCH_CHECK(groups <= 64 && C % groups == 0);
  auto mean = torch::empty({N, groups},
                           xs[0].options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  auto stream = at::hip::getCurrentHIPStream();
  dlb_gn_stats_segs(ptrs, starts, (int)xs.size(),
                    mean.data_ptr<float>(), rstd.data_ptr<float>(), N, HW, C,
                    (int)groups, (float)eps, stream.stream());
  return {mean, rstd};
}

extern "C" void dlb_conv_fwd(const void* x, const void* w, void* y,
                             const float* bias, int N, int IH, int IW, int Ci,
                             int OH, int OW, int Co, int R, int S, int stride,
                             int pad, hipStream_t stream);
extern "C" bool dlb_conv3x3_bwd_halo(const void* dy, const void* w, void* dx,
                                     int N, int H, int W, int Ci, int Co,
                                     hipStream_t stream);
extern "C" void dlb_conv_bwd_data(const void* dy, const void* wt, void* dx,
                                  int N, int IH, int IW, int Ci, int OH,
                                  int OW, int Co, int R, int S, int stride,
                                  int pad, hipStream_t stream);
extern "C" void dlb_conv_bwd_data_acc(const void* dy, const void* wt,
                                      void* dx, int N, int IH, int IW,
                                      int Ci, int OH, int OW, int Co, int R,
                                      int S, int stride, int pad,
                                      hipStream_t stream);
extern "C" void dlb_conv_wrw(const void* x, const void* dy, float* dw, int N,
                             int IH, int IW, int Ci, int OH, int OW, int Co,
                             int R, int S, int stride, int pad, int splits,
                             hipStream_t stream);
extern "C" int dlb_conv_wrw_nsplits(int N, int OH, int OW, int Ci, int Co,
                                    int R, int S);
extern "C" void dlb_slab_sum(const float* part, float* out, int splits,
                             long len, hipStream_t stream);

// Reduce [splits, len] fp32 slabs into `out` (len) with the
// deterministic column-sum kernel (reduce.hip) — replaces the generic
// torch reducer that measured 11 µs/call at these shapes.
static void slab_reduce_into(torch::Tensor part, torch::Tensor out,
                             int splits, long len) {
  dlb_slab_sum(part.data_ptr<float>(), out.data_ptr<float>(), splits, len,
               at::hip::getCurrentHIPStream().stream());
}

static inline bool is_cl(const torch::Tensor& t) {
  return t.is_contiguous(torch::MemoryFormat::ChannelsLast);
}

// x [N,Ci,H,W] channels_last bf16; w [Co,Ci,R,S] channels_last bf16;
// bias fp32 [Co] or empty.  Returns y [N,Co,OH,OW] channels_last bf16.
// y[N,HW,Co] = (GN(concat xs) with given stats) @ w^T   (1x1 conv)
static torch::Tensor gn_conv1x1_fwd(std::vector<torch::Tensor> xs,
                                    torch::Tensor mean, torch::Tensor rstd,
                                    torch::Tensor gamma, torch::Tensor beta,
                                    bool relu, torch::Tensor w) {
  TORCH_CHECK(!xs.empty() && xs.size() <= 56);
  int C = 0;
  const void* ptrs[56];
  int starts[57];
  for (size_t i = 0; i < xs.size(); ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
    TORCH_CHECK(x.size(2) % 8 == 0);
    ptrs[i] = x.data_ptr();
    starts[i] = C;
    C += (int)x.size(2);
  }
  starts[xs.size()] = C;
  const int N = xs[0].size(0), HW = xs[0].size(1);
  TORCH_CHECK(is_cl(w) && w.size(1) == C && w.size(2) == 1 && w.size(3) == 1);
  TORCH_CHECK(w.scalar_type() == torch::kBFloat16);
  TORCH_CHECK((reinterpret_cast<uintptr_t>(gamma.data_ptr()) & 15) == 0 &&
              (reinterpret_cast<uintptr_t>(beta.data_ptr()) & 15) == 0,
              "gamma/beta must be 16B-aligned (vector loads)");
  TORCH_CHECK(C / mean.size(1) >= 2, "fused path needs >=2 channels/group");
  TORCH_CHECK(HW >= 32 || (HW >= 16 && 128 % HW == 0),
              "row-chunk sample span exceeds the staged table");
  const int Co = w.size(0);
  const int G = mean.size(1);
  auto y = torch::empty({N, HW, Co}, xs[0].options());
  auto stream = at::hip::getCurrentHIPStream();
  dlb_gnconv1x1_fwd(ptrs, starts, (int)xs.size(), mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                    beta.data_ptr<float>(), w.data_ptr(), y.data_ptr(), N,
                    HW, C, G, Co, relu ? 1 : 0, stream.stream());
  return y;
}

// dW[Co, C] fp32 = dy^T @ GN(concat xs)  (1x1 conv weight grad)
static torch::Tensor gn_conv1x1_wrw(std::vector<torch::Tensor> xs,
                                    torch::Tensor mean, torch::Tensor rstd,
                                    torch::Tensor gamma, torch::Tensor beta,
                                    bool relu, torch::Tensor dy,
                                    c10::optional<torch::Tensor> out = c10::nullopt) {
  TORCH_CHECK(!xs.empty() && xs.size() <= 56);
  int C = 0;
  const void* ptrs[56];
  int starts[57];
  for (size_t i = 0; i < xs.size(); ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
    ptrs[i] = x.data_ptr();
    starts[i] = C;
    C += (int)x.size(2);
  }
  starts[xs.size()] = C;
  const int N = xs[0].size(0), HW = xs[0].size(1);
  TORCH_CHECK(is_cl(dy) && dy.size(0) == N && dy.size(2) * dy.size(3) == HW);
  const int Co = dy.size(1);
  const int G = mean.size(1);
  TORCH_CHECK((reinterpret_cast<uintptr_t>(gamma.data_ptr()) & 15) == 0 &&
              (reinterpret_cast<uintptr_t>(beta.data_ptr()) & 15) == 0,
              "gamma/beta must be 16B-aligned (vector loads)");
  TORCH_CHECK(C / G >= 2, "fused path needs >=2 channels/group");
  TORCH_CHECK(HW >= 32 || (HW >= 16 && 64 % HW == 0),
              "row-chunk sample span exceeds the staged table");
  int splits = dlb_conv_wrw_nsplits(N, 1, HW, C, Co, 1, 1);
  // Sweepable split cap.  MEASURED (r2c22): capping LOSES — 128 -> -5%,
  // 64 -> -15% flagship: the fp32 slab traffic is cheaper than the
  // occupancy the extra splits buy on the few-tile deep-stream GEMMs.
  // Default 0 = uncapped; kept as a knob for other batch shapes.
  static const int gnc_cap = [] {
    const char* e = getenv("DLB_GNC_SPLITS_MAX");
    return e ? atoi(e) : 0;
  }();
  if (gnc_cap > 0 && splits > gnc_cap) splits = gnc_cap;
  const int mps = ((N * HW + splits - 1) / splits + 63) / 64 * 64;
  splits = (N * HW + mps - 1) / mps;
  auto part = torch::empty({splits, Co, C},
                           xs[0].options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  dlb_gnconv1x1_wrw(ptrs, starts, (int)xs.size(), mean.data_ptr<float>(),
                    rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                    beta.data_ptr<float>(), dy.data_ptr(),
                    part.data_ptr<float>(), N, HW, C, G, Co, relu ? 1 : 0,
                    splits, stream.stream());
  if (out.has_value()) {
    TORCH_CHECK(out->numel() == (long)Co * C &&
                out->scalar_type() == torch::kFloat32);
    slab_reduce_into(part, *out, splits, (long)Co * C);
    return *out;
  }
  if (splits == 1) return part.squeeze(0);
  auto o = torch::empty({Co, C}, part.options());
  slab_reduce_into(part, o, splits, (long)Co * C);
  return o;
}

static torch::Tensor conv_fwd(torch::Tensor x, torch::Tensor w,
                              c10::optional<torch::Tensor> bias,
                              int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(is_cl(x) && is_cl(w), "conv expects channels_last tensors");
  const int N = x.size(0), Ci = x.size(1), IH = x.size(2), IW = x.size(3);
  const int Co = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == Ci);
  const int OH = (IH + 2 * pad - R) / stride + 1;
  const int OW = (IW + 2 * pad - S) / stride + 1;
  auto y = torch::empty({N, Co, OH, OW},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  const float* bp = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32);
    bp = bias->data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  dlb_conv_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), bp, N, IH, IW, Ci,
               OH, OW, Co, R, S, (int)stride, (int)pad, stream.stream());
  return y;
}

// dy [N,Co,OH,OW] channels_last bf16; w [Co,Ci,R,S] channels_last bf16
// (the natural layout — the kernel gathers it directly, no host reshape).
static torch::Tensor conv_bwd_data(torch::Tensor dy, torch::Tensor w,
                                   int64_t IH, int64_t IW, int64_t stride,
                                   int64_t pad,
                                   c10::optional<torch::Tensor> accum_into
                                       = c10::nullopt) {
  // accum_into: an existing [N, Ci, IH, IW] channels_last bf16 grad —
  // the generic kernel ADDS the data-grad into it (residual junctions),
  // replacing a separate elementwise add pass over the tensor.
  TORCH_CHECK(dy.is_cuda() && is_cl(dy) && is_cl(w));
  const int N = dy.size(0), Co = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  const int Ci = w.size(1), R = w.size(2), S = w.size(3);
  auto stream = at::hip::getCurrentHIPStream();
  if (accum_into.has_value()) {
    auto& dx0 = *accum_into;
    TORCH_CHECK(is_cl(dx0) && dx0.size(0) == N && dx0.size(1) == Ci &&
                dx0.size(2) == IH && dx0.size(3) == IW &&
                dx0.scalar_type() == dy.scalar_type());
    dlb_conv_bwd_data_acc(dy.data_ptr(), w.data_ptr(), dx0.data_ptr(), N,
                          (int)IH, (int)IW, Ci, OH, OW, Co, R, S,
                          (int)stride, (int)pad, stream.stream());
    return dx0;
  }
  auto dx = torch::empty({N, Ci, IH, IW},
                         dy.options().memory_format(torch::MemoryFormat::ChannelsLast));
  // 3x3/s1/p1: the LDS-halo kernel in transpose-read mode reads the
  // weight unmodified (no flip+copy transform, ~2x the generic kernel)
  // measured: the halo-WTR route wins for shallow reductions (DenseNet
  // growth convs, Co=32) and loses to the generic tr-read kernel at
  // ResNet depths — gate on the reduction size
  static const bool no_bwd_halo = getenv("DLB_NO_BWD_HALO") != nullptr;
  if (!no_bwd_halo && Co <= 64 &&
      R == 3 && S == 3 && stride == 1 && pad == 1 && IH == OH && IW == OW &&
      dlb_conv3x3_bwd_halo(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), N,
                           (int)IH, (int)IW, Ci, Co, stream.stream()))
    return dx;
  dlb_conv_bwd_data(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), N, (int)IH,
                    (int)IW, Ci, OH, OW, Co, R, S, (int)stride,
                    (int)pad, stream.stream());
  return dx;
}

// Returns dw fp32 [Co, R*S*Ci] (the [Co][R][S][Ci] channels_last image).
// Split-K partials go to per-split slabs (no atomic contention) and are
// reduced with one sum kernel here.
static torch::Tensor conv_wrw(torch::Tensor x, torch::Tensor dy, int64_t R,
                              int64_t S, int64_t stride, int64_t pad,
                              c10::optional<torch::Tensor> out = c10::nullopt) {
  // out: a [Co, R*S*Ci] fp32 buffer the slab sum is written into (the
  // flat-arena grad view of the weight) — skips the separate grad add.
  TORCH_CHECK(x.is_cuda() && is_cl(x) && is_cl(dy));
  const int N = x.size(0), Ci = x.size(1), IH = x.size(2), IW = x.size(3);
  const int Co = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  int splits = dlb_conv_wrw_nsplits(N, OH, OW, Ci, Co, (int)R, (int)S);
  bool env_override = false;
  if (const char* env = getenv("DLB_WRW_SPLITS")) {
    int v = atoi(env);
    if (v > 0 && v < splits) { splits = v; env_override = true; }
  }
  const long K = R * S * Ci;
  // every slab element is written by exactly one block when the tiles
  // cover Co and K exactly — skip the memset then (it measured 4% of
  // the whole step as FillFunctor calls)
  const bool halo3 = (R == 3 && S == 3 && Ci % 32 == 0 && Co % 8 == 0 &&
                      OW <= 32 && (OW & (OW - 1)) == 0 &&
                      (OH * OW) % 128 == 0 && stride == 1 && pad == 1);
  // generic path: the store loop guards every (co, k) exactly, edge
  // tiles included, and a block with an empty m-range stores its
  // zero-initialized accumulators — so the slab never needs a memset
  // (the Co/K divisibility test here was over-conservative and cost
  // ~113 fill launches per ResNet-101 step)
  const bool full = !env_override && (!halo3 || Co % 32 == 0);
  auto part = full
      ? torch::empty({splits, Co, K}, x.options().dtype(torch::kFloat32))
      : torch::zeros({splits, Co, K}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  dlb_conv_wrw(x.data_ptr(), dy.data_ptr(), part.data_ptr<float>(), N, IH,
               IW, Ci, OH, OW, Co, (int)R, (int)S, (int)stride, (int)pad,
               splits, stream.stream());
  if (out.has_value()) {
    TORCH_CHECK(out->numel() == (long)Co * K &&
                out->scalar_type() == torch::kFloat32);
    slab_reduce_into(part, *out, splits, (long)Co * K);
    return *out;
  }
  if (splits == 1) return part.squeeze(0);
  auto o = torch::empty({Co, K}, part.options());
  slab_reduce_into(part, o, splits, (long)Co * K);
  return o;
}

extern "C" void dlb_avgpool_fwd(const void* x, void* y, int N, int H, int W,
                                int C, int k, hipStream_t stream);
extern "C" void dlb_avgpool_bwd(const void* dy, void* dx, int N, int H, int W,
                                int C, int k, hipStream_t stream);
extern "C" void dlb_gavg_fwd(const void* x, void* y, int N, int HW, int C,
                             hipStream_t stream);
extern "C" void dlb_gavg_bwd(const void* dy, void* dx, int N, int HW, int C,
                             hipStream_t stream);
extern "C" void dlb_ln_fwd(const void* x, void* y, const float* gamma,
                           const float* beta, float* mean, float* rstd, int R,
                           int D, float eps, hipStream_t stream);
extern "C" void dlb_ln_bwd(const void* x, const void* dz, void* dx,
                           const float* gamma, const float* mean,
                           const float* rstd, float* dgamma, float* dbeta,
                           int R, int D, hipStream_t stream);
extern "C" void dlb_attn_fwd(const void* q, const void* k, const void* v,
                             void* o, float* p_save, int S, int B, int H,
                             int DH, int ld_qkv, int ld_o, float pd,
                             unsigned long long seed, hipStream_t stream);
extern "C" void dlb_attn_bwd(const void* q, const void* k, const void* v,
                             const void* dout, const float* p_save, void* dq,
                             void* dk, void* dv, int S, int B, int H, int DH,
                             int ld_qkv, int ld_o, int ld_g, float pd,
                             unsigned long long seed, hipStream_t stream);
extern "C" void dlb_logsoftmax_fwd(const void* x, void* y, long R, int D,
                                   hipStream_t stream);
extern "C" void dlb_logsoftmax_bwd(const void* y, const void* dy, void* dx,
                                   long R, int D, hipStream_t stream);

// ---- pooling (channels_last bf16) -----------------------------------
static torch::Tensor avgpool_fwd(torch::Tensor x, int64_t k) {
  TORCH_CHECK(x.is_cuda() && is_cl(x) && x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0 && H % k == 0 && W % k == 0);
  auto y = torch::empty({N, C, H / k, W / k},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_avgpool_fwd(x.data_ptr(), y.data_ptr(), N, H, W, C, (int)k,
                  at::hip::getCurrentHIPStream().stream());
  return y;
}
static torch::Tensor avgpool_bwd(torch::Tensor dy, int64_t k, int64_t H,
                                 int64_t W) {
  TORCH_CHECK(dy.is_cuda() && is_cl(dy));
  const int N = dy.size(0), C = dy.size(1);
  auto dx = torch::empty({N, C, H, W},
                         dy.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_avgpool_bwd(dy.data_ptr(), dx.data_ptr(), N, (int)H, (int)W, C, (int)k,
                  at::hip::getCurrentHIPStream().stream());
  return dx;
}
static torch::Tensor gavg_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && is_cl(x) && x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(C % 8 == 0);
  auto y = torch::empty({N, C, 1, 1},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_gavg_fwd(x.data_ptr(), y.data_ptr(), N, HW, C,
               at::hip::getCurrentHIPStream().stream());
  return y;
}
static torch::Tensor gavg_bwd(torch::Tensor dy, int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda());
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  const int N = dyc.size(0), C = dyc.size(1);
  auto dx = torch::empty({N, C, H, W},
                         dyc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_gavg_bwd(dyc.data_ptr(), dx.data_ptr(), N, (int)(H * W), C,
               at::hip::getCurrentHIPStream().stream());
  return dx;
}

// ---- layernorm (last-dim, bf16 rows) --------------------------------
static std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  const int D = x.size(-1);
  const long R = x.numel() / D;
  TORCH_CHECK(D <= 1024);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  dlb_ln_fwd(x.data_ptr(), y.data_ptr(), gamma.data_ptr<float>(),
             beta.data_ptr<float>(), mean.data_ptr<float>(),
             rstd.data_ptr<float>(), (int)R, D, (float)eps,
             at::hip::getCurrentHIPStream().stream());
  return {y, mean, rstd};
}
static std::vector<torch::Tensor> ln_bwd(torch::Tensor x, torch::Tensor dz,
                                         torch::Tensor gamma,
                                         torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int D = x.size(-1);
  const long R = x.numel() / D;
  auto dzc = dz.contiguous();
  auto dx = torch::empty_like(x);
  auto gbuf = torch::zeros({2, D}, x.options().dtype(torch::kFloat32));
  auto dgamma = gbuf[0];
  auto dbeta = gbuf[1];
  dlb_ln_bwd(x.data_ptr(), dzc.data_ptr(), dx.data_ptr(),
             gamma.data_ptr<float>(), mean.data_ptr<float>(),
             rstd.data_ptr<float>(), dgamma.data_ptr<float>(),
             dbeta.data_ptr<float>(), (int)R, D,
             at::hip::getCurrentHIPStream().stream());
  return {dx, dgamma, dbeta};
}

// ---- causal attention ------------------------------------------------
static std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, int64_t nhead,
                                           double pd, int64_t seed) {
  // q/k/v: [S, B, E] bf16 slices sharing a storage row stride.
  // pd > 0 applies philox dropout to the attention PROBABILITIES
  // (train mode; reference MHA p=0.2) keyed by (seed, element).
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  const int S = q.size(0), B = q.size(1), E = q.size(2);
  const int DH = E / (int)nhead;
  TORCH_CHECK(S <= 40 && DH <= 104, "attention kernel caps: S<=40, DH<=104");
  TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1 && v.stride(2) == 1);
  const int ld = q.stride(0) / B;  // elements per (s,b) row
  TORCH_CHECK(q.stride(1) == ld && (long)B * ld == q.stride(0));
  auto o = torch::empty({S, B, E}, q.options());
  auto p_save = torch::empty({(long)B * nhead, S, S},
                             q.options().dtype(torch::kFloat32));
  dlb_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
               p_save.data_ptr<float>(), S, B, (int)nhead, DH, ld, E,
               (float)pd, (unsigned long long)seed,
               at::hip::getCurrentHIPStream().stream());
  return {o, p_save};
}
static std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, torch::Tensor do_,
                                           torch::Tensor p_save,
                                           int64_t nhead, double pd,
                                           int64_t seed) {
  const int S = q.size(0), B = q.size(1), E = q.size(2);
  const int DH = E / (int)nhead;
  const int ld = q.stride(0) / B;
  auto dq = torch::empty({S, B, E}, q.options());
  auto dk = torch::empty_like(dq);
  auto dv = torch::empty_like(dq);
  auto doc = do_.contiguous();
  dlb_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), doc.data_ptr(),
               p_save.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
               dv.data_ptr(), S, B, (int)nhead, DH, ld, E, E, (float)pd,
               (unsigned long long)seed,
               at::hip::getCurrentHIPStream().stream());
  return {dq, dk, dv};
}

// ---- log_softmax -----------------------------------------------------
static torch::Tensor logsoftmax_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  const int D = x.size(-1);
  const long R = x.numel() / D;
  auto y = torch::empty_like(x);
  dlb_logsoftmax_fwd(x.data_ptr(), y.data_ptr(), R, D,
                     at::hip::getCurrentHIPStream().stream());
  return y;
}
static torch::Tensor logsoftmax_bwd(torch::Tensor y, torch::Tensor dy) {
  const int D = y.size(-1);
  const long R = y.numel() / D;
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(y);
  dlb_logsoftmax_bwd(y.data_ptr(), dyc.data_ptr(), dx.data_ptr(), R, D,
                     at::hip::getCurrentHIPStream().stream());
  return dx;
}

extern "C" void dlb_lmloss_fwd(const void* h, const void* w,
                               const float* bias, const int* tgt,
                               float* part, float* ztgt, float* lse,
                               float* loss, long T, int d, int V, int P,
                               hipStream_t stream);
extern "C" void dlb_lmloss_bwd(const void* h, const void* w,
                               const float* bias, const int* tgt,
                               const float* lse, const float* go, float* dh,
                               float* dw, float* db, long T, int d, int V,
                               int P, hipStream_t stream);

static int lmloss_partitions(long T) {
  // enough (row-block x partition) blocks to fill 256 CUs
  const long rb = (T + 63) / 64;
  long p = (512 + rb - 1) / rb;
  if (p < 1) p = 1;
  if (p > 16) p = 16;
  return (int)p;
}

// h [T,d] bf16, w [V,d] bf16, bias [V] fp32, tgt [T] int32.
// Returns {loss (0-dim fp32), lse [T] fp32}.
static std::vector<torch::Tensor> lmloss_fwd(torch::Tensor h, torch::Tensor w,
                                             torch::Tensor bias,
                                             torch::Tensor tgt) {
  TORCH_CHECK(h.is_cuda() && h.dim() == 2 && h.is_contiguous() &&
              h.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.is_contiguous() && w.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(bias.is_contiguous() && bias.scalar_type() == torch::kFloat32);
  TORCH_CHECK(tgt.is_contiguous() && tgt.scalar_type() == torch::kInt32);
  const long T = h.size(0);
  const int d = h.size(1), V = w.size(0);
  TORCH_CHECK(w.size(1) == d && d <= 224 && d % 8 == 0);
  TORCH_CHECK(tgt.numel() == T && bias.numel() == V);
  const int P = lmloss_partitions(T);
  auto opt = h.options().dtype(torch::kFloat32);
  auto part = torch::empty({T, P, 2}, opt);
  auto ztgt = torch::empty({T}, opt);
  auto lse = torch::empty({T}, opt);
  auto loss = torch::zeros({}, opt);
  dlb_lmloss_fwd(h.data_ptr(), w.data_ptr(), bias.data_ptr<float>(),
                 tgt.data_ptr<int>(), part.data_ptr<float>(),
                 ztgt.data_ptr<float>(), lse.data_ptr<float>(),
                 loss.data_ptr<float>(), T, d, V, P,
                 at::hip::getCurrentHIPStream().stream());
  return {loss, lse};
}

// Returns {dh [T,d] fp32, dw [V,d] fp32, db [V] fp32}.
static std::vector<torch::Tensor> lmloss_bwd(torch::Tensor h, torch::Tensor w,
                                             torch::Tensor bias,
                                             torch::Tensor tgt,
                                             torch::Tensor lse,
                                             torch::Tensor go) {
  const long T = h.size(0);
  const int d = h.size(1), V = w.size(0);
  TORCH_CHECK(go.is_cuda() && go.scalar_type() == torch::kFloat32 &&
              go.numel() == 1);
  const int P = lmloss_partitions(T);
  auto opt = h.options().dtype(torch::kFloat32);
  auto dh = torch::zeros({T, d}, opt);   // fp32 atomics accumulate
  auto dw = torch::empty({V, d}, opt);   // direct stores
  auto db = torch::empty({V}, opt);
  dlb_lmloss_bwd(h.data_ptr(), w.data_ptr(), bias.data_ptr<float>(),
                 tgt.data_ptr<int>(), lse.data_ptr<float>(),
                 go.data_ptr<float>(), dh.data_ptr<float>(),
                 dw.data_ptr<float>(), db.data_ptr<float>(), T, d, V, P,
                 at::hip::getCurrentHIPStream().stream());
  return {dh, dw, db};
}

extern "C" void dlb_gconv_fwd(const void* x, const void* w, void* y, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream);
extern "C" void dlb_gconv_bwd(const void* dy, const void* w, void* dx, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream);
extern "C" void dlb_gconv_wrw(const void* x, const void* dy, float* dw, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream);

// grouped 3x3 (RegNet): x [N,C,H,W] cl bf16, w [C,GW,3,3] cl bf16
static torch::Tensor gconv_fwd(torch::Tensor x, torch::Tensor w,
                               int64_t stride) {
  TORCH_CHECK(x.is_cuda() && is_cl(x) && is_cl(w));
  const int N = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int GW = w.size(1);
  TORCH_CHECK(w.size(0) == C && w.size(2) == 3 && w.size(3) == 3);
  const int OH = (IH + 2 - 3) / stride + 1, OW = (IW + 2 - 3) / stride + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_gconv_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), N, IH, IW, C, GW,
                (int)stride, at::hip::getCurrentHIPStream().stream());
  return y;
}
static torch::Tensor gconv_bwd(torch::Tensor dy, torch::Tensor w, int64_t IH,
                               int64_t IW, int64_t stride) {
  const int N = dy.size(0), C = dy.size(1);
  const int GW = w.size(1);
  auto dx = torch::empty({N, C, IH, IW},
                         dy.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_gconv_bwd(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), N, (int)IH,
                (int)IW, C, GW, (int)stride,
                at::hip::getCurrentHIPStream().stream());
  return dx;
}
static torch::Tensor gconv_wrw(torch::Tensor x, torch::Tensor dy, int64_t GW,
                               int64_t stride,
                               c10::optional<torch::Tensor> out
                                   = c10::nullopt) {
  // out: a pre-zeroed [C, 9*GW] fp32 buffer (flat-arena grad view) the
  // kernel's atomics accumulate into directly.
  const int N = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  torch::Tensor dw;
  if (out.has_value()) {
    dw = *out;
    TORCH_CHECK(dw.is_contiguous() && dw.numel() == (long)C * 9 * GW &&
                dw.scalar_type() == torch::kFloat32);
  } else {
    dw = torch::zeros({C, 9 * GW}, x.options().dtype(torch::kFloat32));
  }
  dlb_gconv_wrw(x.data_ptr(), dy.data_ptr(), dw.data_ptr<float>(), N, IH, IW,
                C, (int)GW, (int)stride,
                at::hip::getCurrentHIPStream().stream());
  return dw;
}

extern "C" void dlb_maxpool_fwd(const void* x, void* y, unsigned char* idx,
                                int N, int H, int W, int C, int k, int stride,
                                int pad, hipStream_t stream);
extern "C" void dlb_maxpool_bwd(const void* dy, const unsigned char* idx,
                                void* dx, int N, int H, int W, int C, int k,
                                int stride, int pad, hipStream_t stream);

static std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k,
                                              int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && is_cl(x) && x.scalar_type() == torch::kBFloat16);
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int OH = (H + 2 * pad - k) / stride + 1;
  const int OW = (W + 2 * pad - k) / stride + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  auto idx = torch::empty({N, OH, OW, C}, x.options().dtype(torch::kByte));
  dlb_maxpool_fwd(x.data_ptr(), y.data_ptr(), idx.data_ptr<unsigned char>(),
                  N, H, W, C, (int)k, (int)stride, (int)pad,
                  at::hip::getCurrentHIPStream().stream());
  return {y, idx};
}
static torch::Tensor maxpool_bwd(torch::Tensor dy, torch::Tensor idx,
                                 int64_t H, int64_t W, int64_t k,
                                 int64_t stride, int64_t pad) {
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  const int N = dyc.size(0), C = dyc.size(1);
  auto dx = torch::empty({N, C, H, W},
                         dyc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  dlb_maxpool_bwd(dyc.data_ptr(), idx.data_ptr<unsigned char>(),
                  dx.data_ptr(), N, (int)H, (int)W, C, (int)k, (int)stride,
                  (int)pad, at::hip::getCurrentHIPStream().stream());
  return dx;
}

extern "C" void dlb_dropout(const void* x, void* y, long n, float pd,
                            unsigned long long seed, hipStream_t stream);
extern "C" void dlb_embed_fwd(const void* table, const int* idx, void* out,
                              long T, int d, float scale,
                              hipStream_t stream);
extern "C" void dlb_embed_bwd(const void* dy, const int* idx, float* dtable,
                              long T, int d, float scale,
                              hipStream_t stream);
extern "C" void dlb_se_fwd(const void* x, const void* g, void* y, long NHW,
                           int HW, int C, hipStream_t stream);
extern "C" void dlb_se_bwd(const void* x, const void* g, const void* dy,
                           void* dx, float* dg, int N, int HW, int C,
                           hipStream_t stream);

// philox dropout; calling twice with the same seed applies the same
// mask (the backward IS a forward on dy)
static torch::Tensor dropout_op(torch::Tensor x, double pd, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  auto xc = x.contiguous();
  auto y = torch::empty_like(xc);
  dlb_dropout(xc.data_ptr(), y.data_ptr(), xc.numel(), (float)pd,
              (unsigned long long)seed,
              at::hip::getCurrentHIPStream().stream());
  return y.view_as(x);
}

static torch::Tensor embed_fwd(torch::Tensor table, torch::Tensor idx,
                               double scale) {
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(idx.is_contiguous() && idx.scalar_type() == torch::kInt32);
  const int d = (int)table.size(1);
  TORCH_CHECK(d % 8 == 0);
  const long T = idx.numel();
  auto out = torch::empty({T, d}, table.options());
  dlb_embed_fwd(table.data_ptr(), idx.data_ptr<int>(), out.data_ptr(), T, d,
                (float)scale, at::hip::getCurrentHIPStream().stream());
  return out;
}

static torch::Tensor embed_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t V, double scale) {
  const int d = (int)dy.size(-1);
  const long T = idx.numel();
  auto dyc = dy.contiguous();
  auto dt = torch::zeros({V, d}, dy.options().dtype(torch::kFloat32));
  dlb_embed_bwd(dyc.data_ptr(), idx.data_ptr<int>(), dt.data_ptr<float>(),
                T, d, (float)scale,
                at::hip::getCurrentHIPStream().stream());
  return dt;
}

// x [N,HW,C] bf16; gate [N,C] bf16 (pre-sigmoid)
static torch::Tensor se_fwd(torch::Tensor x, torch::Tensor g) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && g.is_contiguous());
  const int N = (int)x.size(0), HW = (int)x.size(1), C = (int)x.size(2);
  TORCH_CHECK(C % 8 == 0 && g.numel() == (long)N * C);
  auto y = torch::empty_like(x);
  dlb_se_fwd(x.data_ptr(), g.data_ptr(), y.data_ptr(), (long)N * HW, HW, C,
             at::hip::getCurrentHIPStream().stream());
  return y;
}

static std::vector<torch::Tensor> se_bwd(torch::Tensor x, torch::Tensor g,
                                         torch::Tensor dy) {
  const int N = (int)x.size(0), HW = (int)x.size(1), C = (int)x.size(2);
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(x);
  auto dg = torch::empty({N, C}, x.options().dtype(torch::kFloat32));
  dlb_se_bwd(x.data_ptr(), g.data_ptr(), dyc.data_ptr(), dx.data_ptr(),
             dg.data_ptr<float>(), N, HW, C,
             at::hip::getCurrentHIPStream().stream());
  return {dx, dg};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv_fwd", &conv_fwd, "Implicit-GEMM NHWC bf16 conv forward");
  m.def("conv_bwd_data", &conv_bwd_data, "NHWC bf16 conv input-grad",
        py::arg("dy"), py::arg("w"), py::arg("IH"), py::arg("IW"),
        py::arg("stride"), py::arg("pad"),
        py::arg("accum_into") = py::none());
  m.def("conv_wrw", &conv_wrw,
        py::arg("x"), py::arg("dy"), py::arg("R"), py::arg("S"),
        py::arg("stride"), py::arg("pad"), py::arg("out") = py::none(),
        "NHWC bf16 conv weight-grad (fp32 out)");
  m.def("sgd_momentum", &sgd_momentum,
        py::arg("p"), py::arg("g"), py::arg("m"), py::arg("lr"),
        py::arg("mu"), py::arg("mirror") = py::none(),
        "Fused SGD momentum step over flat arenas (gfx950)");
  m.def("gn_fwd", &gn_fwd, "Fused GroupNorm(+residual-add)(+ReLU) forward",
        py::arg("xs"), py::arg("gamma"), py::arg("beta"), py::arg("groups"),
        py::arg("eps"), py::arg("relu"), py::arg("res") = py::none());
  m.def("gn_bwd", &gn_bwd, "Fused GroupNorm(+ReLU) backward, NHWC bf16",
        py::arg("xs"), py::arg("dz"), py::arg("gamma"), py::arg("beta"),
        py::arg("mean"), py::arg("rstd"), py::arg("groups"), py::arg("relu"),
        py::arg("dx_accum") = py::none(),
        py::arg("dgamma_out") = py::none(), py::arg("dbeta_out") = py::none(),
        py::arg("dgb_defer") = false, py::arg("res") = py::none());
  m.def("gn_dgb_reduce_multi", &gn_dgb_reduce_multi,
        "Batched deterministic dgamma/dbeta reduction (one launch)");
  m.def("chan_sums", &chan_sums, "per-channel (sum, ssq) of a segment");
  m.def("slab_sum", [](torch::Tensor part) {
          TORCH_CHECK(part.is_cuda() && part.dim() == 2 &&
                      part.is_contiguous() &&
                      part.scalar_type() == torch::kFloat32);
          auto out = torch::empty({part.size(1)}, part.options());
          slab_reduce_into(part, out, (int)part.size(0), part.size(1));
          return out;
        },
        "deterministic column sum of a [S, len] fp32 matrix");
  m.def("gn_stats_from_sums", &gn_stats_from_sums,
        "GroupNorm mean/rstd from accumulated per-channel sums");
  m.def("gn_stats", &gn_stats, "Stats-only GroupNorm over virtual concat");
  m.def("gn_conv1x1_fwd", &gn_conv1x1_fwd,
        "Fused GroupNorm(+ReLU) -> 1x1 conv forward (stream never packed)");
  m.def("gn_conv1x1_wrw", &gn_conv1x1_wrw,
        py::arg("xs"), py::arg("mean"), py::arg("rstd"), py::arg("gamma"),
        py::arg("beta"), py::arg("relu"), py::arg("dy"),
        py::arg("out") = py::none(),
        "Fused GroupNorm(+ReLU) -> 1x1 conv weight grad");
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("gavg_fwd", &gavg_fwd);
  m.def("gavg_bwd", &gavg_bwd);
  m.def("ln_fwd", &ln_fwd);
  m.def("ln_bwd", &ln_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("gconv_fwd", &gconv_fwd);
  m.def("gconv_bwd", &gconv_bwd);
  m.def("gconv_wrw", &gconv_wrw, "grouped 3x3 weight-grad",
        py::arg("x"), py::arg("dy"), py::arg("GW"),
        py::arg("stride"), py::arg("out") = py::none());
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("logsoftmax_fwd", &logsoftmax_fwd);
  m.def("logsoftmax_bwd", &logsoftmax_bwd);
  m.def("lmloss_fwd", &lmloss_fwd,
        "Fused decoder GEMM -> log_softmax -> NLL forward (no logits)");
  m.def("lmloss_bwd", &lmloss_bwd,
        "Fused LM loss backward: dh, dW, db with recomputed logits tiles");
  m.def("dropout", &dropout_op, "philox dropout (same seed = same mask)");
  m.def("embed_fwd", &embed_fwd, "embedding lookup x scale (bf16)");
  m.def("embed_bwd", &embed_bwd, "embedding grad scatter (fp32)");
  m.def("se_fwd", &se_fwd, "SE sigmoid-gate broadcast multiply");
  m.def("se_bwd", &se_bwd, "SE backward (dx, dgate)");
}
