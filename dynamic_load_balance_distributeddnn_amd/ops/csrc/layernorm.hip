// LayerNorm fwd/bwd over the last dim (gfx950), bf16 in/out, fp32 stats.
// Shape regime: the LM encoder layers normalize [S, B, 200] rows
// (reference Net/Transformer.py via TransformerEncoderLayer).  One wave
// per row; the feature dim (200) is lane-strided with shfl reductions.

#include "common.h"

typedef __hip_bfloat16 bf16;

#define LN_BLOCK 256
#define LN_ROWS_PER_BLOCK 4  // 4 waves -> 4 rows

__global__ void __launch_bounds__(LN_BLOCK)
ln_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
              const float* __restrict__ gamma, const float* __restrict__ beta,
              float* __restrict__ mean_out, float* __restrict__ rstd_out,
              const int R, const int D, const float eps) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * LN_ROWS_PER_BLOCK + wave;
  if (row >= R) return;
  const bf16* xr = x + (long)row * D;

  float s = 0.f, ss = 0.f;
  for (int i = lane; i < D; i += 64) {
    float v = __bfloat162float(xr[i]);
    s += v;
    ss += v * v;
  }
  s = wave_reduce_sum(s);
  ss = wave_reduce_sum(ss);
  const float mu = __shfl(s, 0, 64) / D;
  const float var = __shfl(ss, 0, 64) / D - mu * mu;
  const float r = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mu;
    rstd_out[row] = r;
  }
  bf16* yr = y + (long)row * D;
  for (int i = lane; i < D; i += 64) {
    float v = (__bfloat162float(xr[i]) - mu) * r;
    yr[i] = __float2bfloat16(v * gamma[i] + beta[i]);
  }
}

// dx = r*(g*dy - mean(g*dy) - xhat*mean(g*dy*xhat))
__global__ void __launch_bounds__(LN_BLOCK)
ln_bwd_kernel(const bf16* __restrict__ x, const bf16* __restrict__ dz,
              bf16* __restrict__ dx, const float* __restrict__ gamma,
              const float* __restrict__ mean_in,
              const float* __restrict__ rstd_in, float* __restrict__ dgamma,
              float* __restrict__ dbeta, const int R, const int D) {
  __shared__ float s_dg[1024];  // per-block channel partials (D <= 1024)
  __shared__ float s_db[1024];
  for (int i = threadIdx.x; i < D; i += LN_BLOCK) {
    s_dg[i] = 0.f;
    s_db[i] = 0.f;
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * LN_ROWS_PER_BLOCK + wave;
  if (row < R) {
    const bf16* xr = x + (long)row * D;
    const bf16* dr = dz + (long)row * D;
    const float mu = mean_in[row], r = rstd_in[row];

    float s1 = 0.f, s2 = 0.f;
    for (int i = lane; i < D; i += 64) {
      const float xhat = (__bfloat162float(xr[i]) - mu) * r;
      const float dy = __bfloat162float(dr[i]);
      const float g = gamma[i];
      s1 += g * dy;
      s2 += g * dy * xhat;
      atomicAdd(&s_dg[i], dy * xhat);
      atomicAdd(&s_db[i], dy);
    }
    s1 = wave_reduce_sum(s1);
    s2 = wave_reduce_sum(s2);
    const float m1 = __shfl(s1, 0, 64) / D;
    const float m2 = __shfl(s2, 0, 64) / D;

    bf16* dxr = dx + (long)row * D;
    for (int i = lane; i < D; i += 64) {
      const float xhat = (__bfloat162float(xr[i]) - mu) * r;
      const float dy = __bfloat162float(dr[i]);
      dxr[i] = __float2bfloat16(r * (gamma[i] * dy - m1 - xhat * m2));
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += LN_BLOCK) {
    atomicAdd(&dgamma[i], s_dg[i]);
    atomicAdd(&dbeta[i], s_db[i]);
  }
}

extern "C" void dlb_ln_fwd(const void* x, void* y, const float* gamma,
                           const float* beta, float* mean, float* rstd, int R,
                           int D, float eps, hipStream_t stream) {
  hipLaunchKernelGGL(ln_fwd_kernel, dim3(cdiv(R, LN_ROWS_PER_BLOCK)),
                     dim3(LN_BLOCK), 0, stream, (const bf16*)x, (bf16*)y,
                     gamma, beta, mean, rstd, R, D, eps);
}
extern "C" void dlb_ln_bwd(const void* x, const void* dz, void* dx,
                           const float* gamma, const float* mean,
                           const float* rstd, float* dgamma, float* dbeta,
                           int R, int D, hipStream_t stream) {
  hipLaunchKernelGGL(ln_bwd_kernel, dim3(cdiv(R, LN_ROWS_PER_BLOCK)),
                     dim3(LN_BLOCK), 0, stream, (const bf16*)x,
                     (const bf16*)dz, (bf16*)dx, gamma, mean, rstd, dgamma,
                     dbeta, R, D);
}
