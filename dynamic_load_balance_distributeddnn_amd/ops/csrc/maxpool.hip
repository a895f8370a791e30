// NHWC bf16 max pooling fwd/bwd (gfx950) — the zoo's shapes: MnistNet
// 2x2/s2/p0, GoogLeNet 3x3/s1/p1 and 3x3/s2/p1 (overlapping windows).
// Forward stores the argmax window index (u8); backward routes each
// input pixel's grad by scanning the <=9 windows that cover it — no
// atomics, works for overlapped and non-overlapped pools alike.
//
// VEC=8 path: one thread owns 8 consecutive channels, so every window
// tap is a single 16-byte load (the original channel-scalar threads
// issued 2-byte loads and the GoogLeNet pools measured 33% of its step
// — profiles r2c30).  MnistNet's C=10/20 takes the VEC=1 fallback.

#include "common.h"

typedef __hip_bfloat16 bf16;
struct MpV8 { bf16 v[8]; };
struct MpU8 { unsigned char b[8]; };

#define MPBLOCK 256

template <int VEC>
__global__ void __launch_bounds__(MPBLOCK)
maxpool_fwd_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                   unsigned char* __restrict__ idx, const int N, const int H,
                   const int W, const int C, const int k, const int stride,
                   const int pad, const int OH, const int OW) {
  const int CV = C / VEC;
  const long total = (long)N * OH * OW * CV;
  for (long i = (long)blockIdx.x * MPBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * MPBLOCK) {
    const int cv = (int)(i % CV);
    long rest = i / CV;
    const int ow = (int)(rest % OW); rest /= OW;
    const int oh = (int)(rest % OH);
    const int n = (int)(rest / OH);
    const int c = cv * VEC;
    float best[VEC];
    int bi[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) { best[j] = -1e30f; bi[j] = 0; }
    for (int r = 0; r < k; ++r) {
      const int ih = oh * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < k; ++s) {
        const int iw = ow * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        const bf16* px = x + (((long)n * H + ih) * W + iw) * C + c;
        if (VEC == 8) {
          MpV8 vv = *reinterpret_cast<const MpV8*>(px);
#pragma unroll
          for (int j = 0; j < VEC; ++j) {
            const float v = __bfloat162float(vv.v[j]);
            if (v > best[j]) { best[j] = v; bi[j] = r * k + s; }
          }
        } else {
          const float v = __bfloat162float(*px);
          if (v > best[0]) { best[0] = v; bi[0] = r * k + s; }
        }
      }
    }
    const long o = (((long)n * OH + oh) * OW + ow) * C + c;
    if (VEC == 8) {
      MpV8 ov;
      MpU8 oi;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        ov.v[j] = __float2bfloat16(best[j]);
        oi.b[j] = (unsigned char)bi[j];
      }
      *reinterpret_cast<MpV8*>(y + o) = ov;
      *reinterpret_cast<MpU8*>(idx + o) = oi;
    } else {
      y[o] = __float2bfloat16(best[0]);
      idx[o] = (unsigned char)bi[0];
    }
  }
}

template <int VEC>
__global__ void __launch_bounds__(MPBLOCK)
maxpool_bwd_kernel(const bf16* __restrict__ dy,
                   const unsigned char* __restrict__ idx,
                   bf16* __restrict__ dx, const int N, const int H,
                   const int W, const int C, const int k, const int stride,
                   const int pad, const int OH, const int OW) {
  const int CV = C / VEC;
  const long total = (long)N * H * W * CV;
  for (long i = (long)blockIdx.x * MPBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * MPBLOCK) {
    const int cv = (int)(i % CV);
    long rest = i / CV;
    const int iw = (int)(rest % W); rest /= W;
    const int ih = (int)(rest % H);
    const int n = (int)(rest / H);
    const int c = cv * VEC;
    float acc[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
    const bool s1 = (stride == 1);  // uniform; skips runtime idivs
    for (int r = 0; r < k; ++r) {
      const int ohn = ih + pad - r;
      if (ohn < 0 || (!s1 && ohn % stride)) continue;
      const int oh = s1 ? ohn : ohn / stride;
      if (oh >= OH) continue;
      for (int s = 0; s < k; ++s) {
        const int own = iw + pad - s;
        if (own < 0 || (!s1 && own % stride)) continue;
        const int ow = s1 ? own : own / stride;
        if (ow >= OW) continue;
        const long o = (((long)n * OH + oh) * OW + ow) * C + c;
        const unsigned char want = (unsigned char)(r * k + s);
        if (VEC == 8) {
          MpU8 ii = *reinterpret_cast<const MpU8*>(idx + o);
          MpV8 dv = *reinterpret_cast<const MpV8*>(dy + o);
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            if (ii.b[j] == want) acc[j] += __bfloat162float(dv.v[j]);
        } else {
          if (idx[o] == want) acc[0] += __bfloat162float(dy[o]);
        }
      }
    }
    const long xo = (((long)n * H + ih) * W + iw) * C + c;
    if (VEC == 8) {
      MpV8 ov;
#pragma unroll
      for (int j = 0; j < VEC; ++j) ov.v[j] = __float2bfloat16(acc[j]);
      *reinterpret_cast<MpV8*>(dx + xo) = ov;
    } else {
      dx[xo] = __float2bfloat16(acc[0]);
    }
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
  if (C % 8 == 0) {
    hipLaunchKernelGGL((maxpool_fwd_kernel<8>),
                       dim3(mgrid((long)N * OH * OW * (C / 8))),
                       dim3(MPBLOCK), 0, stream, (const bf16*)x, (bf16*)y,
                       idx, N, H, W, C, k, stride, pad, OH, OW);
  } else {
    hipLaunchKernelGGL((maxpool_fwd_kernel<1>),
                       dim3(mgrid((long)N * OH * OW * C)), dim3(MPBLOCK), 0,
                       stream, (const bf16*)x, (bf16*)y, idx, N, H, W, C, k,
                       stride, pad, OH, OW);
  }
}
extern "C" void dlb_maxpool_bwd(const void* dy, const unsigned char* idx,
                                void* dx, int N, int H, int W, int C, int k,
                                int stride, int pad, hipStream_t stream) {
  const int OH = (H + 2 * pad - k) / stride + 1;
  const int OW = (W + 2 * pad - k) / stride + 1;
  if (C % 8 == 0) {
    hipLaunchKernelGGL((maxpool_bwd_kernel<8>),
                       dim3(mgrid((long)N * H * W * (C / 8))), dim3(MPBLOCK),
                       0, stream, (const bf16*)dy, idx, (bf16*)dx, N, H, W,
                       C, k, stride, pad, OH, OW);
  } else {
    hipLaunchKernelGGL((maxpool_bwd_kernel<1>),
                       dim3(mgrid((long)N * H * W * C)), dim3(MPBLOCK), 0,
                       stream, (const bf16*)dy, idx, (bf16*)dx, N, H, W, C,
                       k, stride, pad, OH, OW);
  }
}
