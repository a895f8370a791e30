// Grouped 3x3 convolution (gfx950) — RegNet's only grouped shape
// (SURVEY.md K3): channels-per-group == group_width (8 or 16), stride
// 1/2, pad 1, square maps.  Near-depthwise: K per output is 9*GW
// (72-144 MACs), far below the MFMA regime — direct VALU kernels, NHWC.
//
// fwd: thread per (output pixel, co-octet); weights via L2 (<=110 KB).
// bwd-data: thread per (input pixel, ci-octet), gathering matching taps.
// wrw: block per (group, tap, m-split); dy/x chunks staged in LDS,
//      thread per (co_local, ci_local) accumulator, atomics at the end
//      (contention = split count <= 16 — cheap).

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

#define GBLOCK 256

struct GConvParams {
  const bf16* x;   // [N, H, W, C]
  const bf16* w;   // [C, 3, 3, GW]  (channels_last [Co,GW,3,3])
  const bf16* dy;  // bwd/wrw
  bf16* out;       // y or dx
  float* dw;       // wrw
  int N, IH, IW, C, OH, OW, GW, stride;
  int m_per_split;
};

__global__ void __launch_bounds__(GBLOCK)
gconv_fwd_kernel(const GConvParams p) {
  const long total = (long)p.N * p.OH * p.OW * (p.C / 8);
  for (long i = (long)blockIdx.x * GBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * GBLOCK) {
    const int co8 = (int)(i % (p.C / 8)) * 8;
    long rest = i / (p.C / 8);
    const int ow = (int)(rest % p.OW); rest /= p.OW;
    const int oh = (int)(rest % p.OH);
    const int n = (int)(rest / p.OH);
    const int g = co8 / p.GW;           // octet stays in one group (GW>=8)
    const int ci0 = g * p.GW;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < 3; ++r) {
      const int ih = oh * p.stride - 1 + r;
      if (ih < 0 || ih >= p.IH) continue;
      for (int s = 0; s < 3; ++s) {
        const int iw = ow * p.stride - 1 + s;
        if (iw < 0 || iw >= p.IW) continue;
        const bf16* xp = p.x + (((long)n * p.IH + ih) * p.IW + iw) * p.C + ci0;
        for (int c = 0; c < p.GW; ++c) {
          const float xv = __bfloat162float(xp[c]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[j] += xv * __bfloat162float(
                p.w[(((long)(co8 + j) * 3 + r) * 3 + s) * p.GW + c]);
        }
      }
    }
    bf16x8_t o;
    bf16* ov = reinterpret_cast<bf16*>(&o);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = __float2bfloat16(acc[j]);
    *reinterpret_cast<bf16x8_t*>(
        p.out + (((long)n * p.OH + oh) * p.OW + ow) * p.C + co8) = o;
  }
}

__global__ void __launch_bounds__(GBLOCK)
gconv_bwd_kernel(const GConvParams p) {
  const long total = (long)p.N * p.IH * p.IW * (p.C / 8);
  for (long i = (long)blockIdx.x * GBLOCK + threadIdx.x; i < total;
       i += (long)gridDim.x * GBLOCK) {
    const int ci8 = (int)(i % (p.C / 8)) * 8;
    long rest = i / (p.C / 8);
    const int iw = (int)(rest % p.IW); rest /= p.IW;
    const int ih = (int)(rest % p.IH);
    const int n = (int)(rest / p.IH);
    const int g = ci8 / p.GW;
    const int co0 = g * p.GW;
    const int cl0 = ci8 - g * p.GW;     // ci position inside group
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < 3; ++r) {
      const int ohn = ih + 1 - r;
      if (ohn < 0 || ohn % p.stride) continue;
      const int oh = ohn / p.stride;
      if (oh >= p.OH) continue;
      for (int s = 0; s < 3; ++s) {
        const int own = iw + 1 - s;
        if (own < 0 || own % p.stride) continue;
        const int ow = own / p.stride;
        if (ow >= p.OW) continue;
        const bf16* dp =
            p.dy + (((long)n * p.OH + oh) * p.OW + ow) * p.C + co0;
        for (int c = 0; c < p.GW; ++c) {  // c = co within group
          const float dv = __bfloat162float(dp[c]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[j] += dv * __bfloat162float(
                p.w[(((long)(co0 + c) * 3 + r) * 3 + s) * p.GW + cl0 + j]);
        }
      }
    }
    bf16x8_t o;
    bf16* ov = reinterpret_cast<bf16*>(&o);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = __float2bfloat16(acc[j]);
    *reinterpret_cast<bf16x8_t*>(
        p.out + (((long)n * p.IH + ih) * p.IW + iw) * p.C + ci8) = o;
  }
}

// wrw: grid (G, 9 taps, splits); thread (co_l, ci_l) pairs (GW*GW <= 256)
__global__ void __launch_bounds__(GBLOCK)
gconv_wrw_kernel(const GConvParams p) {
  __shared__ bf16 dy_s[32 * 16];  // [mm][co_l]
  __shared__ bf16 x_s[32 * 16];   // [mm][ci_l]

  const int g = blockIdx.x;
  const int tap = blockIdx.y;
  const int r = tap / 3, s = tap % 3;
  const int co0 = g * p.GW, ci0 = g * p.GW;
  const int t = threadIdx.x;
  const int col = t % p.GW;            // ci_l
  const int row = t / p.GW;            // co_l
  const bool active = row < p.GW;
  const int M = p.N * p.OH * p.OW;
  const int mstart = blockIdx.z * p.m_per_split;
  const int mend = min(M, mstart + p.m_per_split);

  float acc = 0.f;
  for (int mt = mstart; mt < mend; mt += 32) {
    // stage dy[mm][co_l] and x(tap)[mm][ci_l]
    for (int c = t; c < 32 * p.GW; c += GBLOCK) {
      const int mm = c / p.GW, cl = c % p.GW;
      const int m = mt + mm;
      bf16 dv = __float2bfloat16(0.f), xv = __float2bfloat16(0.f);
      if (m < mend) {
        const int n = m / (p.OH * p.OW);
        const int rem = m % (p.OH * p.OW);
        const int oh = rem / p.OW, ow = rem % p.OW;
        dv = p.dy[(((long)n * p.OH + oh) * p.OW + ow) * p.C + co0 + cl];
        const int ih = oh * p.stride - 1 + r;
        const int iw = ow * p.stride - 1 + s;
        if (ih >= 0 && ih < p.IH && iw >= 0 && iw < p.IW)
          xv = p.x[(((long)n * p.IH + ih) * p.IW + iw) * p.C + ci0 + cl];
      }
      dy_s[mm * 16 + cl] = dv;
      x_s[mm * 16 + cl] = xv;
    }
    __syncthreads();
    if (active) {
      for (int mm = 0; mm < 32 && mt + mm < mend; ++mm)
        acc += __bfloat162float(dy_s[mm * 16 + row]) *
               __bfloat162float(x_s[mm * 16 + col]);
    }
    __syncthreads();
  }
  if (active)
    atomicAdd(&p.dw[(((long)(co0 + row) * 3 + r) * 3 + s) * p.GW + col], acc);
}

extern "C" void dlb_gconv_fwd(const void* x, const void* w, void* y, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{(const bf16*)x, (const bf16*)w, nullptr, (bf16*)y, nullptr,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  long total = (long)N * OH * OW * (C / 8);
  int grid = (int)std::min<long>(cdiv(total, GBLOCK), 8192);
  hipLaunchKernelGGL(gconv_fwd_kernel, dim3(grid), dim3(GBLOCK), 0, stream, p);
}

extern "C" void dlb_gconv_bwd(const void* dy, const void* w, void* dx, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{nullptr, (const bf16*)w, (const bf16*)dy, (bf16*)dx, nullptr,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  long total = (long)N * IH * IW * (C / 8);
  int grid = (int)std::min<long>(cdiv(total, GBLOCK), 8192);
  hipLaunchKernelGGL(gconv_bwd_kernel, dim3(grid), dim3(GBLOCK), 0, stream, p);
}

extern "C" void dlb_gconv_wrw(const void* x, const void* dy, float* dw, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{(const bf16*)x, nullptr, (const bf16*)dy, nullptr, dw,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  const int M = N * OH * OW;
  int splits = std::min(16, std::max(1, M / (32 * 32)));
  p.m_per_split = cdiv(cdiv(M, splits), 32) * 32;
  splits = cdiv(M, p.m_per_split);
  dim3 grid(C / GW, 9, splits);
  hipLaunchKernelGGL(gconv_wrw_kernel, grid, dim3(GBLOCK), 0, stream, p);
}
