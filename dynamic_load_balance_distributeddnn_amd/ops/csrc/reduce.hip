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
