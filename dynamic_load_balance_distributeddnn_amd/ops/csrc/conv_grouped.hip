// Grouped 3x3 convolution (gfx950) — RegNet's only grouped shape
// (SURVEY.md K3): channels-per-group == group_width (8 or 16), stride
// 1/2, pad 1, square maps, NHWC.
//
// History: v1 re-loaded weights per pixel (3.7 ms/call); v2 ran direct
// VALU with LDS weight broadcasts and still measured 48% of the
// RegNetY step (profiles r2c30).  v3 (this file) is MFMA end to end:
// per 16-channel tile the forward/data-grad are [M, 144] x [144, 16]
// GEMMs over tap-gathered operands, and the weight grad is one
// [16 x 16] tile per (tile, tap) with K = pixels — all on
// v_mfma_f32_16x16x32_bf16 with ds_read_b64_tr_b16 fragments.  GW=8
// packs TWO groups per tile (the weight is block-diagonal across
// them), so no MFMA lanes are wasted.  RegNetY-400MF +46%,
// RegNetX-200MF +16% whole-model vs the v2 kernels.

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

#define GBLOCK 256

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

struct GConvParams {
  const bf16* x;   // [N, H, W, C]
  const bf16* w;   // [C, 3, 3, GW]  (channels_last [Co,GW,3,3])
  const bf16* dy;  // bwd/wrw
  bf16* out;       // y or dx
  float* dw;       // wrw
  int N, IH, IW, C, OH, OW, GW, stride;
  int m_per_split;
};

// ---- MFMA forward/backward ------------------------------------------
// Per group, forward is y[M, GW] = A[M, 9*GWP] @ B[9*GWP, GW] with A the
// tap-gathered input (zero-padded to GWP per tap) and B the group's
// weight — small-N GEMM on v_mfma_f32_16x16x32_bf16 (the direct VALU
// versions below measured ~20% of the RegNetY step).  K = 9*GWP = 144,
// padded to 160 (5 k-steps).  Backward-data is the same shape with the
// tap-shift on the dy gather and B[k=tap*GWP+co][ci] = w[co][tap][ci].
#define GK 160
#define GLDA (GK + 8)

template <bool BWD>
__global__ void __launch_bounds__(GBLOCK)
gconv_mfma_kernel(const GConvParams p) {
  __shared__ bf16 a_lds[128 * GLDA];
  __shared__ bf16 b_lds[16 * GLDA];
  __shared__ bf16 o_lds[128 * GWP];

  // a tile covers GWP consecutive channels = GWP/GW whole groups
  // (2 groups for GW=8 — the weight is block-diagonal across them)
  const int c0 = blockIdx.x * GWP;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  // output pixel space: fwd = (OH, OW), bwd = (IH, IW)
  const int PH = BWD ? p.IH : p.OH;
  const int PW = BWD ? p.IW : p.OW;
  const long M = (long)p.N * PH * PW;
  const long m0 = (long)blockIdx.y * 128;

  // stage B [16 rows = out-channel][k]: fwd B[n][tap*GWP+ci] =
  // w[c0+n][tap][ci]; bwd B[n=ci][tap*GWP+co] = w[c0+co][tap][ci]
  for (int c = t; c < 16 * (GK / 8); c += GBLOCK) {
    const int nrow = c / (GK / 8);
    const int k8 = (c % (GK / 8)) * 8;
    union { bf16x8_t v; bf16 h[8]; } u;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k8 + j;
      const int tap = k / GWP;
      const int cl = k - tap * GWP;      // in-tile channel (k side)
      bf16 w = (bf16)__float2bfloat16(0.f);
      const int ga = c0 + nrow, gb = c0 + cl;
      if (tap < 9 && ga < p.C && gb < p.C &&
          ga / p.GW == gb / p.GW) {      // block-diagonal across groups
        if (BWD)  // ga = ci, gb = co
          w = p.w[((long)gb * 9 + tap) * p.GW + (ga - (ga / p.GW) * p.GW)];
        else      // ga = co, gb = ci
          w = p.w[((long)ga * 9 + tap) * p.GW + (gb - (gb / p.GW) * p.GW)];
      }
      u.h[j] = w;
    }
    *reinterpret_cast<bf16x8_t*>(&b_lds[nrow * GLDA + k8]) = u.v;
  }

  // stage A: per output pixel row, the 9 tap-gathered GW-slices
  for (int c = t; c < 128 * (GK / 8); c += GBLOCK) {
    const int mm = c / (GK / 8);
    const int k8 = (c % (GK / 8)) * 8;
    const int tap = k8 / GWP;           // GWP=16, k8 multiple of 8
    const int cl8 = k8 - tap * GWP;     // 0 or 8
    bf16x8_t v = {0, 0, 0, 0, 0, 0, 0, 0};
    const long m = m0 + mm;
    if (m < M && tap < 9 && c0 + cl8 < p.C) {
      const int pw = (int)(m % PW);
      const int ph = (int)((m / PW) % PH);
      const int n = (int)(m / ((long)PH * PW));
      const int r = tap / 3, s = tap % 3;
      bool ok;
      int sh, sw;
      if (BWD) {
        const int ohn = ph + 1 - r, own = pw + 1 - s;
        ok = ohn >= 0 && own >= 0 && ohn % p.stride == 0 &&
             own % p.stride == 0;
        sh = ohn / p.stride;
        sw = own / p.stride;
        ok = ok && sh < p.OH && sw < p.OW;
      } else {
        sh = ph * p.stride - 1 + r;
        sw = pw * p.stride - 1 + s;
        ok = sh >= 0 && sh < p.IH && sw >= 0 && sw < p.IW;
      }
      if (ok) {
        const bf16* src = BWD ? p.dy : p.x;
        const int SH = BWD ? p.OH : p.IH;
        const int SW = BWD ? p.OW : p.IW;
        v = *reinterpret_cast<const bf16x8_t*>(
            src + (((long)n * SH + sh) * SW + sw) * p.C + c0 + cl8);
      }
    }
    *reinterpret_cast<bf16x8_t*>(&a_lds[mm * GLDA + k8]) = v;
  }
  __syncthreads();

  gf32x4 acc[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < GK / 32; ++ks) {
    const int k8 = ks * 32 + (lane >> 4) * 8;
    bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
        &b_lds[(lane & 15) * GLDA + k8]);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
          &a_lds[(wave * 32 + i * 16 + (lane & 15)) * GLDA + k8]);
      acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[i],
                                                       0, 0, 0);
    }
  }

  // bounce [128][GWP] through LDS for coalesced 16B row writes
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int col = lane & 15;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = wave * 32 + i * 16 + (lane >> 4) * 4 + rr;
      o_lds[row * GWP + col] = __float2bfloat16(acc[i][rr]);
    }
  }
  __syncthreads();
  for (int c = t; c < 128 * (GWP / 8); c += GBLOCK) {
    const int mm = c / (GWP / 8);
    const int c8 = (c % (GWP / 8)) * 8;
    const long m = m0 + mm;
    if (m >= M || c0 + c8 >= p.C) continue;
    *reinterpret_cast<bf16x8_t*>(p.out + m * p.C + c0 + c8) =
        *reinterpret_cast<bf16x8_t*>(&o_lds[mm * GWP + c8]);
  }
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

__global__ void __launch_bounds__(GBLOCK)
gconv_wrw_kernel(const GConvParams p) {
  __shared__ bf16 dy_s[128 * GWP];      // [mm][co_l]
  __shared__ bf16 x_s[9][128 * GWP];    // per-tap shifted [mm][ci_l]

  const int c0 = blockIdx.x * GWP;  // GWP channels = 1-2 whole groups
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
      const bool mok = m < mend && c0 + c8 < p.C;
      if (mok) {
        n = m / (p.OH * p.OW);
        const int rem = m % (p.OH * p.OW);
        oh = rem / p.OW;
        ow = rem % p.OW;
        dv = *reinterpret_cast<const bf16x8_t*>(
            p.dy + (((long)n * p.OH + oh) * p.OW + ow) * p.C + c0 + c8);
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
                p.x + (((long)n * p.IH + ih) * p.IW + iw) * p.C + c0 + c8);
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

  // C/D layout: col = lane&15 (in-tile ci), row = (lane>>4)*4 + rr (co);
  // keep only same-group (block-diagonal) in-bounds entries
  const int col = lane & 15;
  for (int ti = 0; ti < ntaps; ++ti) {
    const int tap = wave + 4 * ti;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = (lane >> 4) * 4 + rr;
      const int co = c0 + row, ci = c0 + col;
      if (co >= p.C || ci >= p.C || co / p.GW != ci / p.GW) continue;
      atomicAdd(&p.dw[(((long)co * 3 + tap / 3) * 3 + tap % 3) * p.GW +
                      (ci - (ci / p.GW) * p.GW)],
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
  dim3 grid(cdiv(C, GWP), cdiv((long)N * OH * OW, 128));
  hipLaunchKernelGGL((gconv_mfma_kernel<false>), grid, dim3(GBLOCK), 0,
                     stream, p);
}

extern "C" void dlb_gconv_bwd(const void* dy, const void* w, void* dx, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{nullptr, (const bf16*)w, (const bf16*)dy, (bf16*)dx, nullptr,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  dim3 grid(cdiv(C, GWP), cdiv((long)N * IH * IW, 128));
  hipLaunchKernelGGL((gconv_mfma_kernel<true>), grid, dim3(GBLOCK), 0,
                     stream, p);
}

extern "C" void dlb_gconv_wrw(const void* x, const void* dy, float* dw, int N,
                              int IH, int IW, int C, int GW, int stride,
                              hipStream_t stream) {
  const int OH = (IH + 2 - 3) / stride + 1;
  const int OW = (IW + 2 - 3) / stride + 1;
  GConvParams p{(const bf16*)x, nullptr, (const bf16*)dy, nullptr, dw,
                N, IH, IW, C, OH, OW, GW, stride, 0};
  const int M = N * OH * OW;
  const long tiles = cdiv(C, GWP);  // one block: all 9 taps, GWP chans
  int splits = (int)std::min<long>(std::max<long>(1, 2048 / tiles),
                                   std::max<long>(1, M / (4 * 128)));
  p.m_per_split = cdiv(cdiv(M, splits), 128) * 128;
  splits = cdiv(M, p.m_per_split);
  dim3 grid(cdiv(C, GWP), splits);
  hipLaunchKernelGGL(gconv_wrw_kernel, grid, dim3(GBLOCK), 0, stream, p);
}
