// NHWC bf16 pooling kernels (gfx950): non-overlapping average pool
// (kernel k, stride k — the zoo's only avg-pool shape: DenseNet 2/4,
// ResNet 4, GoogLeNet 8-on-8), global average pool (RegNet adaptive-1 /
// SE), and their backwards.  torch's channels_last avg_pool2d backward
// measured 185us/call on the DenseNet step; these are trivial streaming
// kernels.

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

#define PBLOCK 256

__global__ void __launch_bounds__(PBLOCK)
avgpool_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                   const int N, const int H, const int W, const int C,
                   const int k, const int OH, const int OW) {
  const long total = (long)N * OH * OW * (C / 8);
  const float inv = 1.0f / (k * k);
  for (long i = (long)blockIdx.x * PBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * PBLOCK) {
    const int c8 = (int)(i % (C / 8)) * 8;
    long rest = i / (C / 8);
    const int ow = (int)(rest % OW); rest /= OW;
    const int oh = (int)(rest % OH);
    const int n = (int)(rest / OH);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < k; ++r)
      for (int s = 0; s < k; ++s) {
        const long off =
            (((long)n * H + oh * k + r) * W + ow * k + s) * C + c8;
        bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(x + off);
        const bf16* vv = reinterpret_cast<const bf16*>(&v);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += __bfloat162float(vv[j]);
      }
    bf16x8_t out;
    bf16* ov = reinterpret_cast<bf16*>(&out);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = __float2bfloat16(acc[j] * inv);
    *reinterpret_cast<bf16x8_t*>(
        y + (((long)n * OH + oh) * OW + ow) * C + c8) = out;
  }
}

__global__ void __launch_bounds__(PBLOCK)
avgpool_bwd_kernel(const bf16* __restrict__ dy, bf16* __restrict__ dx,
                   const int N, const int H, const int W, const int C,
                   const int k, const int OH, const int OW) {
  const long total = (long)N * H * W * (C / 8);
  const float inv = 1.0f / (k * k);
  for (long i = (long)blockIdx.x * PBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * PBLOCK) {
    const int c8 = (int)(i % (C / 8)) * 8;
    long rest = i / (C / 8);
    const int iw = (int)(rest % W); rest /= W;
    const int ih = (int)(rest % H);
    const int n = (int)(rest / H);
    const int oh = ih / k, ow = iw / k;
    bf16x8_t v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (oh < OH && ow < OW)
      v = *reinterpret_cast<const bf16x8_t*>(
          dy + (((long)n * OH + oh) * OW + ow) * C + c8);
    bf16x8_t out;
    const bf16* vv = reinterpret_cast<const bf16*>(&v);
    bf16* ov = reinterpret_cast<bf16*>(&out);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov[j] = __float2bfloat16(__bfloat162float(vv[j]) * inv);
    *reinterpret_cast<bf16x8_t*>(
        dx + (((long)n * H + ih) * W + iw) * C + c8) = out;
  }
}

// global average: one wave-group per (n, c-octet strip)
__global__ void __launch_bounds__(PBLOCK)
gavg_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                const int N, const int HW, const int C) {
  const long total = (long)N * (C / 8);
  const float inv = 1.0f / HW;
  for (long i = (long)blockIdx.x * PBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * PBLOCK) {
    const int c8 = (int)(i % (C / 8)) * 8;
    const int n = (int)(i / (C / 8));
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    const bf16* base = x + (long)n * HW * C + c8;
    for (int p = 0; p < HW; ++p) {
      bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(base + (long)p * C);
      const bf16* vv = reinterpret_cast<const bf16*>(&v);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += __bfloat162float(vv[j]);
    }
    bf16x8_t out;
    bf16* ov = reinterpret_cast<bf16*>(&out);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = __float2bfloat16(acc[j] * inv);
    *reinterpret_cast<bf16x8_t*>(y + (long)n * C + c8) = out;
  }
}

__global__ void __launch_bounds__(PBLOCK)
gavg_bwd_kernel(const bf16* __restrict__ dy, bf16* __restrict__ dx,
                const int N, const int HW, const int C) {
  const long total = (long)N * HW * (C / 8);
  const float inv = 1.0f / HW;
  for (long i = (long)blockIdx.x * PBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * PBLOCK) {
    const int c8 = (int)(i % (C / 8)) * 8;
    long rest = i / (C / 8);
    const int n = (int)(rest / HW);
    bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(dy + (long)n * C + c8);
    bf16x8_t out;
    const bf16* vv = reinterpret_cast<const bf16*>(&v);
    bf16* ov = reinterpret_cast<bf16*>(&out);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov[j] = __float2bfloat16(__bfloat162float(vv[j]) * inv);
    *reinterpret_cast<bf16x8_t*>(dx + i * 8) = out;
  }
}

static inline int pgrid(long total) {
  long g = (total + PBLOCK - 1) / PBLOCK;
  return (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
}

extern "C" void dlb_avgpool_fwd(const void* x, void* y, int N, int H, int W,
                                int C, int k, hipStream_t stream) {
  const int OH = H / k, OW = W / k;
  hipLaunchKernelGGL(avgpool_fwd_kernel,
                     dim3(pgrid((long)N * OH * OW * (C / 8))), dim3(PBLOCK),
                     0, stream, (const bf16*)x, (bf16*)y, N, H, W, C, k, OH,
                     OW);
}
extern "C" void dlb_avgpool_bwd(const void* dy, void* dx, int N, int H, int W,
                                int C, int k, hipStream_t stream) {
  const int OH = H / k, OW = W / k;
  hipLaunchKernelGGL(avgpool_bwd_kernel,
                     dim3(pgrid((long)N * H * W * (C / 8))), dim3(PBLOCK), 0,
                     stream, (const bf16*)dy, (bf16*)dx, N, H, W, C, k, OH,
                     OW);
}
extern "C" void dlb_gavg_fwd(const void* x, void* y, int N, int HW, int C,
                             hipStream_t stream) {
  hipLaunchKernelGGL(gavg_fwd_kernel, dim3(pgrid((long)N * (C / 8))),
                     dim3(PBLOCK), 0, stream, (const bf16*)x, (bf16*)y, N, HW,
                     C);
}
extern "C" void dlb_gavg_bwd(const void* dy, void* dx, int N, int HW, int C,
                             hipStream_t stream) {
  hipLaunchKernelGGL(gavg_bwd_kernel, dim3(pgrid((long)N * HW * (C / 8))),
                     dim3(PBLOCK), 0, stream, (const bf16*)dy, (bf16*)dx, N,
                     HW, C);
}
