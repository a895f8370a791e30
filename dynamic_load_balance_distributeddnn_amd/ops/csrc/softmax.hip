// Row-wise log_softmax fwd/bwd for wide rows (gfx950).
//
// The LM decoder emits [S*B, 33278] bf16 logits and the reference model
// applies log_softmax before NLL (Net/Transformer.py:95, dbs.py:372);
// MnistNet does the same over 10 classes.  One workgroup per row, fp32
// reductions in LDS, bf16 stream in/out.
//   fwd: y = x - max(x) - log(sum(exp(x - max)))
//   bwd: dx = dy - exp(y) * sum(dy)

#include "common.h"

typedef __hip_bfloat16 bf16;

#define SM_BLOCK 256

__device__ inline float block_reduce(float v, float* sbuf, int op_max) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float o = __shfl_down(v, off, 64);
    v = op_max ? fmaxf(v, o) : v + o;
  }
  if (lane == 0) sbuf[wave] = v;
  __syncthreads();
  float r = sbuf[0];
  for (int w = 1; w < SM_BLOCK / 64; ++w)
    r = op_max ? fmaxf(r, sbuf[w]) : r + sbuf[w];
  __syncthreads();
  return r;
}

__global__ void __launch_bounds__(SM_BLOCK)
logsoftmax_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                      const int D) {
  __shared__ float sbuf[SM_BLOCK / 64];
  const long row = blockIdx.x;
  const bf16* xr = x + row * D;
  bf16* yr = y + row * D;

  float mx = -1e30f;
  for (int i = threadIdx.x; i < D; i += SM_BLOCK)
    mx = fmaxf(mx, __bfloat162float(xr[i]));
  mx = block_reduce(mx, sbuf, 1);

  float sum = 0.f;
  for (int i = threadIdx.x; i < D; i += SM_BLOCK)
    sum += __expf(__bfloat162float(xr[i]) - mx);
  sum = block_reduce(sum, sbuf, 0);
  const float lse = mx + __logf(sum);

  for (int i = threadIdx.x; i < D; i += SM_BLOCK)
    yr[i] = __float2bfloat16(__bfloat162float(xr[i]) - lse);
}

__global__ void __launch_bounds__(SM_BLOCK)
logsoftmax_bwd_kernel(const bf16* __restrict__ y, const bf16* __restrict__ dy,
                      bf16* __restrict__ dx, const int D) {
  __shared__ float sbuf[SM_BLOCK / 64];
  const long row = blockIdx.x;
  const bf16* yr = y + row * D;
  const bf16* dr = dy + row * D;
  bf16* xr = dx + row * D;

  float s = 0.f;
  for (int i = threadIdx.x; i < D; i += SM_BLOCK)
    s += __bfloat162float(dr[i]);
  s = block_reduce(s, sbuf, 0);

  for (int i = threadIdx.x; i < D; i += SM_BLOCK) {
    const float g = __bfloat162float(dr[i]) -
                    __expf(__bfloat162float(yr[i])) * s;
    xr[i] = __float2bfloat16(g);
  }
}

extern "C" void dlb_logsoftmax_fwd(const void* x, void* y, long R, int D,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(logsoftmax_fwd_kernel, dim3((unsigned)R),
                     dim3(SM_BLOCK), 0, stream, (const bf16*)x, (bf16*)y, D);
}
extern "C" void dlb_logsoftmax_bwd(const void* y, const void* dy, void* dx,
                                   long R, int D, hipStream_t stream) {
  hipLaunchKernelGGL(logsoftmax_bwd_kernel, dim3((unsigned)R),
                     dim3(SM_BLOCK), 0, stream, (const bf16*)y,
                     (const bf16*)dy, (bf16*)dx, D);
}
