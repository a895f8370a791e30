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

// wrw v3: the per-(group, tap) weight grad is a [GW x GW] output GEMM
// with K = M (dW[co,ci] = sum_m dy[m,co] * x_tap[m,ci]) — exactly one
// v_mfma_f32_16x16x32_bf16 tile per 32 pixels.  v2 ran it as scalar
// LDS FMAs with dy re-staged per tap across 9 blocks and measured 28%
// of the RegNetY step (183 ms, profiles r2c30).  Here: grid (G,
// splits); one block stages dy ONCE and the 9 shifted x images per
// 128-m chunk, and the 4 waves split the 9 taps (wave w owns taps
// {w, w+4, w+8}), each accumulating its [16,16] tiles via
// transpose-read fragments.  GW=8 groups zero-pad to the 16-wide tile.
#define GWP 16  // padded tile width (GW is 8 or 16 in the zoo)

typedef __attribute__((__vector_size__(4 * sizeof(float)))) float gf32x4;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 gtrvec;
#define GLDS3 __attribute__((address_space(3)))

// ds_read_b64_tr_b16 fragment from a row-major [m][GWP] LDS image:
// element j = img[mbase + (lane>>4)*8 + j][lane & 15] — the [n=col]
// [k=m] operand both A (cols=co) and B (cols=ci) need here.
__device__ inline bf16x8_t gtr_frag(const bf16* img, int mbase, int lane) {
  const int j15 = lane & 15, q = lane >> 4;
  const int row = mbase + q * 8 + (j15 >> 2);
  const int col = 4 * (j15 & 3);
  auto p0 = (GLDS3 gtrvec*)((GLDS3 bf16*)img + (long)row * GWP + col);
  auto p1 = (GLDS3 gtrvec*)((GLDS3 bf16*)img + (long)(row + 4) * GWP + col);
  gtrvec lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
  gtrvec hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  union { struct { gtrvec a, b; } t; bf16x8_t v; } u;
  u.t.a = lo;
  u.t.b = hi;
  return u.v;
}

__global__ void __launch_bounds__(GBLOCK)
gconv_wrw_kernel(const GConvParams p) {
  __shared__ bf16 dy_s[128 * GWP];      // [mm][co_l]
  __shared__ bf16 x_s[9][128 * GWP];    // per-tap shifted [mm][ci_l]

  const int g = blockIdx.x;
  const int co0 = g * p.GW, ci0 = g * p.GW;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int M = p.N * p.OH * p.OW;
  const int mstart = blockIdx.y * p.m_per_split;
  const int mend = min(M, mstart + p.m_per_split);
  const int ntaps = (wave < 1) ? 3 : 2;  // taps {w, w+4, w+8<9}

  gf32x4 acc[3];
#pragma unroll
  for (int i = 0; i < 3; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int mt = mstart; mt < mend; mt += 128) {
    // stage dy once and each tap's shifted x (zero-padded cols >= GW)
    for (int c = t; c < 128 * (GWP / 8); c += GBLOCK) {
      const int mm = c / (GWP / 8);
      const int c8 = (c % (GWP / 8)) * 8;
      const int m = mt + mm;
      bf16x8_t dv = {0, 0, 0, 0, 0, 0, 0, 0};
      int n = 0, oh = 0, ow = 0;
      const bool mok = m < mend && c8 < p.GW;
      if (mok) {
        n = m / (p.OH * p.OW);
        const int rem = m % (p.OH * p.OW);
        oh = rem / p.OW;
        ow = rem % p.OW;
        dv = *reinterpret_cast<const bf16x8_t*>(
            p.dy + (((long)n * p.OH + oh) * p.OW + ow) * p.C + co0 + c8);
      }
      *reinterpret_cast<bf16x8_t*>(&dy_s[mm * GWP + c8]) = dv;
#pragma unroll
      for (int tap = 0; tap < 9; ++tap) {
        bf16x8_t xv = {0, 0, 0, 0, 0, 0, 0, 0};
        if (mok) {
          const int ih = oh * p.stride - 1 + tap / 3;
          const int iw = ow * p.stride - 1 + tap % 3;
          if (ih >= 0 && ih < p.IH && iw >= 0 && iw < p.IW)
            xv = *reinterpret_cast<const bf16x8_t*>(
                p.x + (((long)n * p.IH + ih) * p.IW + iw) * p.C + ci0 + c8);
        }
        *reinterpret_cast<bf16x8_t*>(&x_s[tap][mm * GWP + c8]) = xv;
      }
    }
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8_t afrag = gtr_frag(dy_s, ks * 32, lane);
      for (int ti = 0; ti < ntaps; ++ti) {
        bf16x8_t bfrag = gtr_frag(x_s[wave + 4 * ti], ks * 32, lane);
        acc[ti] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                          acc[ti], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // C/D layout: col = lane&15 (ci_l), row = (lane>>4)*4 + rr (co_l)
  const int col = lane & 15;
  for (int ti = 0; ti < ntaps; ++ti) {
    const int tap = wave + 4 * ti;
    if (col >= p.GW) continue;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = (lane >> 4) * 4 + rr;
      if (row >= p.GW) continue;
      atomicAdd(&p.dw[(((long)(co0 + row) * 3 + tap / 3) * 3 + tap % 3) *
                          p.GW + col],
                acc[ti][rr]);
    }
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
  const long tiles = C / GW;  // one block covers all 9 taps of a group
  int splits = (int)std::min<long>(std::max<long>(1, 2048 / tiles),
                                   std::max<long>(1, M / (4 * 128)));
  p.m_per_split = cdiv(cdiv(M, splits), 128) * 128;
  splits = cdiv(M, p.m_per_split);
  dim3 grid(C / GW, splits);
  hipLaunchKernelGGL(gconv_wrw_kernel, grid, dim3(GBLOCK), 0, stream, p);
}
