// Split-K slab reduction for the conv weight-grad kernels (gfx950).
//
// The wrw kernels write per-split fp32 partial slabs [splits, Co*K]
// (contention-free, deterministic).  Round 1 reduced them with
// torch::sum_out — at::native::reduce_kernel measured 4.4% of the
// DenseNet step (profiles/SUMMARY.md: 24.3 ms / 2160 calls ≈ 11 µs per
// call for reductions whose traffic is < 1 µs at HBM speed; the generic
// reducer's config is launch/occupancy-bound at these shapes).  This
// kernel is a flat float4 streaming sum: out[i] = Σ_s part[s*len + i],
// summed in split order (bitwise deterministic across runs).
#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
slab_sum_kernel(const float* __restrict__ part, float* __restrict__ out,
                const int splits, const long len) {
  const long q = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (q >= len) return;
  float4 acc = *reinterpret_cast<const float4*>(part + q);
  for (int s = 1; s < splits; ++s) {
    float4 v = *reinterpret_cast<const float4*>(part + (long)s * len + q);
    acc.x += v.x;
    acc.y += v.y;
    acc.z += v.z;
    acc.w += v.w;
  }
  *reinterpret_cast<float4*>(out + q) = acc;
}

// len must be a multiple of 4 (every wrw slab is: K and C are multiples
// of 8); the binding falls back to torch otherwise.
extern "C" void dlb_slab_sum(const float* part, float* out, int splits,
                             long len, hipStream_t stream) {
  const long quads = len / 4;
  const int block = 256;
  const long grid = (quads + block - 1) / block;
  hipLaunchKernelGGL(slab_sum_kernel, dim3((unsigned)grid), dim3(block), 0,
                     stream, part, out, splits, len);
}

// --------- GroupNorm dgamma/dbeta deterministic column reduction --------
// gn_bwd publishes per-sample partials part[n][2C] with plain stores
// (global atomicAdd on the [2C] words measured a +10..30 us per-dispatch
// tail under 512-way contention — tools/gn_probe — and made dgamma
// nondeterministic).  Level 1 sums sample groups; level 2 adds the group
// sums into the dgamma/dbeta buffers.  Fixed split order -> bitwise
// deterministic across runs.
extern "C" __global__ void __launch_bounds__(256)
gn_dgb_l1_kernel(const float* __restrict__ part, float* __restrict__ mid,
                 const int N, const int gs, const long len2) {
  const long c = (long)blockIdx.x * 256 + threadIdx.x;
  if (c >= len2) return;
  const int n0 = blockIdx.y * gs;
  const int n1 = min(n0 + gs, N);
  float s = 0.f;
  for (int n = n0; n < n1; ++n) s += part[(long)n * len2 + c];
  mid[(long)blockIdx.y * len2 + c] = s;
}

extern "C" __global__ void __launch_bounds__(256)
gn_dgb_l2_kernel(const float* __restrict__ mid, const int groups,
                 const long len2, const int C, float* __restrict__ dgamma,
                 float* __restrict__ dbeta) {
  const long c = (long)blockIdx.x * 256 + threadIdx.x;
  if (c >= len2) return;
  float s = 0.f;
  for (int g = 0; g < groups; ++g) s += mid[(long)g * len2 + c];
  if (c < C)
    dgamma[c] += s;
  else
    dbeta[c - C] += s;
}

extern "C" void dlb_gn_dgb_reduce(const float* part, float* mid, int N,
                                  int groups, int C, float* dgamma,
                                  float* dbeta, hipStream_t stream) {
  const long len2 = 2L * C;
  const int gs = (N + groups - 1) / groups;
  const unsigned cb = (unsigned)((len2 + 255) / 256);
  hipLaunchKernelGGL(gn_dgb_l1_kernel, dim3(cb, groups), dim3(256), 0,
                     stream, part, mid, N, gs, len2);
  hipLaunchKernelGGL(gn_dgb_l2_kernel, dim3(cb), dim3(256), 0, stream, mid,
                     groups, len2, C, dgamma, dbeta);
}
