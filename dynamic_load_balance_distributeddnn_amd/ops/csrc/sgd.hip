// Fused SGD-with-momentum over the flat parameter/gradient/momentum arenas.
//
// The reference steps torch.optim.SGD(lr, momentum=0.9) over up to 362
// separate tensors per iteration (dbs.py:369).  Here parameters, gradients
// and momentum each live in ONE flat fp32 buffer (parallel/grad_sync.py
// + parallel/optim.py), so the whole update is a single HBM-bound
// elementwise pass: v = mu*v + g ; p -= lr*v.  float4 vectorized, grid
// sized to fill 256 CUs.

#include "common.h"

typedef __hip_bfloat16 bf16;

extern "C" __global__ void __launch_bounds__(256)
sgd_momentum_kernel(float* __restrict__ p, const float* __restrict__ g,
                    float* __restrict__ m, bf16* __restrict__ q,
                    const float lr, const float mu, const long n) {
  // q (optional): bf16 mirror of the updated parameters, written in the
  // same pass — the conv forwards read weights from this mirror, which
  // removes the ~120 per-tensor fp32->bf16 cast kernels per step.
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n;
       i += stride) {
    if (i + 3 < n) {
      float4 gv = *reinterpret_cast<const float4*>(g + i);
      float4 mv = *reinterpret_cast<float4*>(m + i);
      float4 pv = *reinterpret_cast<float4*>(p + i);
      mv.x = mu * mv.x + gv.x;
      mv.y = mu * mv.y + gv.y;
      mv.z = mu * mv.z + gv.z;
      mv.w = mu * mv.w + gv.w;
      pv.x -= lr * mv.x;
      pv.y -= lr * mv.y;
      pv.z -= lr * mv.z;
      pv.w -= lr * mv.w;
      *reinterpret_cast<float4*>(m + i) = mv;
      *reinterpret_cast<float4*>(p + i) = pv;
      if (q) {
        union { bf16 h[4]; unsigned long long u; } qv;
        qv.h[0] = __float2bfloat16(pv.x);
        qv.h[1] = __float2bfloat16(pv.y);
        qv.h[2] = __float2bfloat16(pv.z);
        qv.h[3] = __float2bfloat16(pv.w);
        *reinterpret_cast<unsigned long long*>(q + i) = qv.u;
      }
    } else {
      for (long j = i; j < n; ++j) {
        float mv = mu * m[j] + g[j];
        m[j] = mv;
        p[j] -= lr * mv;
        if (q) q[j] = __float2bfloat16(p[j]);
      }
    }
  }
}

extern "C" void dlb_sgd_momentum(float* p, const float* g, float* m,
                                 void* q, float lr, float mu, long n,
                                 hipStream_t stream) {
  const int block = 256;
  // >> 256 workgroups to fill 8 XCDs; cap so the grid-stride loop amortizes
  int grid = (int)std::min<long>(cdiv((long)n, block * 4), 8192);
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(sgd_momentum_kernel, dim3(grid), dim3(block), 0, stream,
                     p, g, m, (bf16*)q, lr, mu, n);
}
