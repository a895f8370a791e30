// Grouped 3x3 convolution (gfx950) — RegNet's only grouped shape
// (SURVEY.md K3): channels-per-group == group_width (8 or 16), stride
// 1/2, pad 1, square maps.  Near-depthwise (K per output = 9*GW), far
// below the MFMA regime — direct VALU kernels, NHWC.
//
// v2: v1 re-loaded every weight from global per output pixel (1152
// scalar loads/thread — measured 3.7 ms/call, 53% of the RegNet step).
// Now each block owns one channel-octet and stages its weight slice in
// LDS once; weight reads are wave-uniform LDS broadcasts and x/dy reads
// are bf16x8 vectors.

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

// fwd: block = (co-octet, 256-pixel tile); weights for the octet in LDS.
__global__ void __launch_bounds__(GBLOCK)
gconv_fwd_kernel(const GConvParams p) {
  __shared__ bf16 wlds[8 * 9 * 16];  // [j][tap][ci], GW<=16

  const int co8 = blockIdx.x * 8;
  const int g = co8 / p.GW;
  const int ci0 = g * p.GW;
  const int t = threadIdx.x;

  // stage the octet's weights: 8 rows of [9*GW] contiguous
  for (int c = t; c < 8 * 9 * p.GW / 8; c += GBLOCK) {
    const int j = c / (9 * p.GW / 8);
    const int e8 = (c % (9 * p.GW / 8)) * 8;
    *reinterpret_cast<bf16x8_t*>(&wlds[(j * 9 * p.GW) + e8]) =
        *reinterpret_cast<const bf16x8_t*>(p.w + (long)(co8 + j) * 9 * p.GW +
                                           e8);
  }
  __syncthreads();

  const long M = (long)p.N * p.OH * p.OW;
  const long m = (long)blockIdx.y * GBLOCK + t;
  if (m >= M) return;
  const int ow = (int)(m % p.OW);
  const int oh = (int)((m / p.OW) % p.OH);
  const int n = (int)(m / ((long)p.OH * p.OW));

  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int r = 0; r < 3; ++r) {
    const int ih = oh * p.stride - 1 + r;
    if (ih < 0 || ih >= p.IH) continue;
    for (int s = 0; s < 3; ++s) {
      const int iw = ow * p.stride - 1 + s;
      if (iw < 0 || iw >= p.IW) continue;
      const bf16* xp = p.x + (((long)n * p.IH + ih) * p.IW + iw) * p.C + ci0;
      const bf16* wp = &wlds[(r * 3 + s) * p.GW];
      for (int c8 = 0; c8 < p.GW; c8 += 8) {
        bf16x8_t xv = *reinterpret_cast<const bf16x8_t*>(xp + c8);
        const bf16* xe = reinterpret_cast<const bf16*>(&xv);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float x1 = __bfloat162float(xe[e]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[j] += x1 * __bfloat162float(wp[j * 9 * p.GW + c8 + e]);
        }
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

// bwd-data: block = (ci-octet, 256-pixel tile); the group's weight
// columns for this ci-octet staged in LDS: [co_l][tap][8].
__global__ void __launch_bounds__(GBLOCK)
gconv_bwd_kernel(const GConvParams p) {
  __shared__ bf16 wlds[16 * 9 * 8];  // [co_l][tap][jj]

  const int ci8 = blockIdx.x * 8;
  const int g = ci8 / p.GW;
  const int co0 = g * p.GW;
  const int cl0 = ci8 - g * p.GW;
  const int t = threadIdx.x;

  for (int c = t; c < p.GW * 9 * 8; c += GBLOCK) {
    const int jj = c % 8;
    const int tap = (c / 8) % 9;
    const int col = c / 72;
    wlds[c] = p.w[((long)(co0 + col) * 9 + tap) * p.GW + cl0 + jj];
  }
  __syncthreads();

  const long M = (long)p.N * p.IH * p.IW;
  const long m = (long)blockIdx.y * GBLOCK + t;
  if (m >= M) return;
  const int iw = (int)(m % p.IW);
  const int ih = (int)((m / p.IW) % p.IH);
  const int n = (int)(m / ((long)p.IH * p.IW));

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
      const bf16* dp = p.dy + (((long)n * p.OH + oh) * p.OW + ow) * p.C + co0;
      const int tap = r * 3 + s;
      for (int c8 = 0; c8 < p.GW; c8 += 8) {
        bf16x8_t dv = *reinterpret_cast<const bf16x8_t*>(dp + c8);
        const bf16* de = reinterpret_cast<const bf16*>(&dv);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float d1 = __bfloat162float(de[e]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[j] += d1 * __bfloat162float(wlds[((c8 + e) * 9 + tap) * 8 + j]);
        }
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

// wrw: grid (G, 9 taps, splits); 128-m stages with vector loads; thread
// = (co_l, ci_l, m-replica); LDS tree-reduce over replicas at the end.
__global__ void __launch_bounds__(GBLOCK)
gconv_wrw_kernel(const GConvParams p) {
  __shared__ bf16 dy_s[128 * 16];  // [mm][co_l]
  __shared__ bf16 x_s[128 * 16];   // [mm][ci_l]
  __shared__ float red[GBLOCK];

  const int g = blockIdx.x;
  const int tap = blockIdx.y;
  const int r = tap / 3, s = tap % 3;
  const int co0 = g * p.GW, ci0 = g * p.GW;
  const int t = threadIdx.x;
  const int pairs = p.GW * p.GW;          // 64 or 256
  const int rep_n = GBLOCK / pairs;       // 4 or 1
  const int pair = t % pairs;
  const int rep = t / pairs;
  const int col = pair % p.GW;            // ci_l
  const int row = pair / p.GW;            // co_l
  const int M = p.N * p.OH * p.OW;
  const int mstart = blockIdx.z * p.m_per_split;
  const int mend = min(M, mstart + p.m_per_split);

  float acc = 0.f;
  for (int mt = mstart; mt < mend; mt += 128) {
    for (int c = t; c < 128 * (p.GW / 8); c += GBLOCK) {
      const int mm = c / (p.GW / 8);
      const int c8 = (c % (p.GW / 8)) * 8;
      const int m = mt + mm;
      bf16x8_t dv = {0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8_t xv = {0, 0, 0, 0, 0, 0, 0, 0};
      if (m < mend) {
        const int n = m / (p.OH * p.OW);
        const int rem = m % (p.OH * p.OW);
        const int oh = rem / p.OW, ow = rem % p.OW;
        dv = *reinterpret_cast<const bf16x8_t*>(
            p.dy + (((long)n * p.OH + oh) * p.OW + ow) * p.C + co0 + c8);
        const int ih = oh * p.stride - 1 + r;
        const int iw = ow * p.stride - 1 + s;
        if (ih >= 0 && ih < p.IH && iw >= 0 && iw < p.IW)
          xv = *reinterpret_cast<const bf16x8_t*>(
              p.x + (((long)n * p.IH + ih) * p.IW + iw) * p.C + ci0 + c8);
      }
      *reinterpret_cast<bf16x8_t*>(&dy_s[mm * 16 + c8]) = dv;
      *reinterpret_cast<bf16x8_t*>(&x_s[mm * 16 + c8]) = xv;
    }
    __syncthreads();
    const int mlim = min(128, mend - mt);
    for (int mm = rep; mm < mlim; mm += rep_n)
      acc += __bfloat162float(dy_s[mm * 16 + row]) *
             __bfloat162float(x_s[mm * 16 + col]);
    __syncthreads();
  }
  red[t] = acc;
  __syncthreads();
  if (rep == 0) {
    for (int rr = 1; rr < rep_n; ++rr) acc += red[pair + rr * pairs];
    atomicAdd(&p.dw[(((long)(co0 + row) * 3 + r) * 3 + s) * p.GW + col], acc);
  }
}

extern "C" void dlb_gconv_fwd(const void* x, const void* w, void* y, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{(const bf16*)x, (const bf16*)w, nullptr, (bf16*)y, nullptr,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  dim3 grid(C / 8, cdiv((long)N * OH * OW, GBLOCK));
  hipLaunchKernelGGL(gconv_fwd_kernel, grid, dim3(GBLOCK), 0, stream, p);
}

extern "C" void dlb_gconv_bwd(const void* dy, const void* w, void* dx, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{nullptr, (const bf16*)w, (const bf16*)dy, (bf16*)dx, nullptr,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  dim3 grid(C / 8, cdiv((long)N * IH * IW, GBLOCK));
  hipLaunchKernelGGL(gconv_bwd_kernel, grid, dim3(GBLOCK), 0, stream, p);
}

extern "C" void dlb_gconv_wrw(const void* x, const void* dy, float* dw, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{(const bf16*)x, nullptr, (const bf16*)dy, nullptr, dw,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  const int M = N * OH * OW;
  const long tiles = (long)(C / GW) * 9;
  int splits = (int)std::min<long>(std::max<long>(1, 2048 / tiles),
                                   std::max<long>(1, M / (4 * 128)));
  p.m_per_split = cdiv(cdiv(M, splits), 128) * 128;
  splits = cdiv(M, p.m_per_split);
  dim3 grid(C / GW, 9, splits);
  hipLaunchKernelGGL(gconv_wrw_kernel, grid, dim3(GBLOCK), 0, stream, p);
}
