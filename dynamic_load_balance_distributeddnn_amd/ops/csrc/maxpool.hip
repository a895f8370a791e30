// NHWC bf16 max pooling fwd/bwd (gfx950) — the zoo's shapes: MnistNet
// 2x2/s2/p0, GoogLeNet 3x3/s1/p1 and 3x3/s2/p1 (overlapping windows).
// Forward stores the argmax window index (u8); backward routes each
// input pixel's grad by scanning the <=9 windows that cover it — no
// atomics, works for overlapped and non-overlapped pools alike.
// Channel-scalar threads (consecutive c -> coalesced), so MnistNet's
// C=10/20 needs no special casing.

#include "common.h"

typedef __hip_bfloat16 bf16;

#define MPBLOCK 256

__global__ void __launch_bounds__(MPBLOCK)
maxpool_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                   unsigned char* __restrict__ idx, const int N, const int H,
                   const int W, const int C, const int k, const int stride,
                   const int pad, const int OH, const int OW) {
  const long total = (long)N * OH * OW * C;
  for (long i = (long)blockIdx.x * MPBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * MPBLOCK) {
    const int c = (int)(i % C);
    long rest = i / C;
    const int ow = (int)(rest % OW); rest /= OW;
    const int oh = (int)(rest % OH);
    const int n = (int)(rest / OH);
    float best = -1e30f;
    int bi = 0;
    for (int r = 0; r < k; ++r) {
      const int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < k; ++s) {
        const int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        const float v = __bfloat162float(
            x[(((long)n * H + ih) * W + iw) * C + c]);
        if (v > best) { best = v; bi = r * k + s; }
      }
    }
    y[i] = __float2bfloat16(best);
    idx[i] = (unsigned char)bi;
  }
}

__global__ void __launch_bounds__(MPBLOCK)
maxpool_bwd_kernel(const bf16* __restrict__ dy,
                   const unsigned char* __restrict__ idx,
                   bf16* __restrict__ dx, const int N, const int H,
                   const int W, const int C, const int k, const int stride,
                   const int pad, const int OH, const int OW) {
  const long total = (long)N * H * W * C;
  for (long i = (long)blockIdx.x * MPBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * MPBLOCK) {
    const int c = (int)(i % C);
    long rest = i / C;
    const int iw = (int)(rest % W); rest /= W;
    const int ih = (int)(rest % H);
    const int n = (int)(rest / H);
    float acc = 0.f;
    for (int r = 0; r < k; ++r) {
      const int ohn = ih + pad - r;
      if (ohn < 0 || ohn % stride) continue;
      const int oh = ohn / stride;
      if (oh >= OH) continue;
      for (int s = 0; s < k; ++s) {
        const int own = iw + pad - s;
        if (own < 0 || own % stride) continue;
        const int ow = own / stride;
        if (ow >= OW) continue;
        const long o = (((long)n * OH + oh) * OW + ow) * C + c;
        if (idx[o] == r * k + s) acc += __bfloat162float(dy[o]);
      }
    }
    dx[i] = __float2bfloat16(acc);
  }
}

static inline int mgrid(long total) {
  long g = (total + MPBLOCK - 1) / MPBLOCK;
  return (int)(g > 8192 ? 8192 : (g < 1 ? 1 : g));
}

extern "C" void dlb_maxpool_fwd(const void* x, void* y, unsigned char* idx,
                                int N, int H, int W, int C, int k, int stride,
                                int pad, hipStream_t stream) {
  const int OH = (H + 2 * pad - k) / stride + 1;
  const int OW = (W + 2 * pad - k) / stride + 1;
  hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(mgrid((long)N * OH * OW * C)),
                     dim3(MPBLOCK), 0, stream, (const bf16*)x, (bf16*)y, idx,
                     N, H, W, C, k, stride, pad, OH, OW);
}
extern "C" void dlb_maxpool_bwd(const void* dy, const unsigned char* idx,
                                void* dx, int N, int H, int W, int C, int k,
                                int stride, int pad, hipStream_t stream) {
  const int OH = (H + 2 * pad - k) / stride + 1;
  const int OW = (W + 2 * pad - k) / stride + 1;
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(mgrid((long)N * H * W * C)),
                     dim3(MPBLOCK), 0, stream, (const bf16*)dy, idx,
                     (bf16*)dx, N, H, W, C, k, stride, pad, OH, OW);
}
