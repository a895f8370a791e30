// Fused GroupNorm -> 1x1 conv (forward and weight-grad) for the
// DenseNet residual stream (gfx950).
//
// DenseNet's 1x1 convs consume a GroupNorm over the virtual channel
// concat (Net/Densenet.py:15-20 in the reference).  Unfused, the stream
// is read for stats, read again and WRITTEN as the packed normalized
// activation, then read a third time by the conv.  Here the stats come
// from the one-pass gn_stats kernel and the conv's A/B operand loaders
// normalize segment data on the fly — the packed activation is never
// materialized (and never saved for backward: the weight-grad kernel
// re-normalizes at load time from the same saved stats).
//
// Numerics: gn(x) is evaluated as x*a + b with per-(sample, channel)
// precomputed a = rstd*gamma and b = beta - mean*a — algebraically equal
// to the unfused kernel's ((x-mu)*rstd)*gamma + beta, fp32-rounded one
// ulp apart; the result feeds the same MFMA tiling (validated against
// the unfused pair in tests/test_gpu_gnconv.py).

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((__vector_size__(8 * sizeof(__bf16)))) __bf16 bf16x8_t;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 trvec_g;
#define LDS3G __attribute__((address_space(3)))

#define CONV_BLOCK 256
#define BK 32
#define GNC_MAXSEG 56

struct GnCSegs {
  const bf16* p[GNC_MAXSEG];
  int start[GNC_MAXSEG + 1];
  int nseg;
};

struct GnCParams {
  GnCSegs segs;
  const float* mean;   // [N, G]
  const float* rstd;   // [N, G]
  const float* gamma;  // [C]
  const float* beta;   // [C]
  const bf16* w;       // fwd: [Co][C] (1x1 channels_last)
  const bf16* dy;      // wrw: [M, Co]
  bf16* y;             // fwd out: [M, Co]
  float* dw;           // wrw out slabs: [splits][Co][C]
  int M, HW, C, G, Co, relu;
  int cg;              // channels per group (C/G), >= 2
  int m_per_split;     // wrw m chunk per split
  FastDiv fd_hw;       // / HW
  FastDiv fd_cg;       // / (C/G)
};

#define GNC_MAXOCT 280  // C <= 2240 (covers DenseNet-161's 2208)

// per-block octet -> segment-index table (replaces the serial segment
// walk in the hot loader with one LDS read)
__device__ inline void gnc_stage_oct(const GnCParams& p, short* soct, int t,
                                     int nthreads) {
  for (int o = t; o < (p.C >> 3); o += nthreads) {
    const int k = o << 3;
    int si = 0;
    while (si + 1 < p.segs.nseg && k >= p.segs.start[si + 1]) ++si;
    soct[o] = (short)si;
  }
}

__device__ inline bf16x8_t gnc_zero8() {
  union { bf16x8_t v; int4 q; } u;
  u.q = {0, 0, 0, 0};
  return u.v;
}

// Per-(sample, channel) affine tables: gn(x) = x*a + b with
//   a[n][c] = rstd[n][g]*gamma[c],  b[n][c] = beta[c] - mean[n][g]*a
// (algebraically equal to the unfused form; fp32-rounds one ulp apart).
// A chunk of rows spans a bounded run of consecutive samples (chunk
// starts are BM-aligned, so at HW=16 the span is exactly BM/HW), and the
// tables are tiny — staged in LDS one pipeline chunk ahead, making the
// hot loader one FMA per element.
#define GNC_NSPAN_FWD 8   // BM=128: HW>=32 spans <=4; HW=16 exactly 8
#define GNC_NSPAN_WRW 4   // BM=64

// stage a/b for channels [c0, c0+W) and samples [n_base, n_base+nspan)
__device__ inline void gnc_stage_ab(const GnCParams& p, float* sa, float* sb,
                                    int c0, int W, int n_base, int nspan,
                                    int t, int nthreads) {
  const int NT = nspan * W;
  for (int idx = t; idx < NT; idx += nthreads) {
    const int nl = idx / W;
    const int c = c0 + idx % W;
    const long n = n_base + nl;
    float a = 0.f, b = 0.f;
    if (c < p.C && n * p.HW < p.M) {
      const int g = (int)p.fd_cg.div((unsigned)c);
      const float rs = p.rstd[n * p.G + g];
      a = rs * p.gamma[c];
      b = p.beta[c] - p.mean[n * p.G + g] * a;
    }
    sa[nl * W + idx % W] = a;
    sb[nl * W + idx % W] = b;
  }
}

// RAW 8-channel segment load at (row m, channel k) — no normalization:
// the affine tables are applied at LDS-WRITE time (gnc_apply8), which
// keeps the global-load loop free of dependent FMA chains (the fused
// wrw measured ~15% behind the plain wrw with the math on the load
// path — round-1 ROADMAP #5).
__device__ inline bf16x8_t gn_raw8(const GnCParams& p, const short* soct,
                                   int m, int k, int mbound, bool* okp) {
  const bool ok = (m < mbound) & (k < p.C);
  *okp = ok;
  if (!ok) return gnc_zero8();
  const unsigned n = p.fd_hw.div((unsigned)m);
  const unsigned pix = (unsigned)m - n * (unsigned)p.HW;
  const int si = soct[k >> 3];
  const int cs = p.segs.start[si + 1] - p.segs.start[si];
  const bf16* ptr = p.segs.p[si] +
                    ((long)n * p.HW + pix) * cs + (k - p.segs.start[si]);
  return *reinterpret_cast<const bf16x8_t*>(ptr);
}

// gn(x) = x*a + b (+ReLU) against the staged tables; zero for padding
// lanes (the GEMM's K/M padding must stay zero, not b).
__device__ inline bf16x8_t gnc_apply8(const GnCParams& p, bf16x8_t x8,
                                      const float* sa, const float* sb,
                                      int base, bool ok) {
  if (!ok) return gnc_zero8();
  union { bf16x8_t v; bf16 h[8]; } in, out;
  in.v = x8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = __bfloat162float(in.h[j]) * sa[base + j] + sb[base + j];
    if (p.relu) v = fmaxf(v, 0.f);
    out.h[j] = __float2bfloat16(v);
  }
  return out.v;
}

__device__ inline bf16x8_t gnc_mask8(bf16x8_t v, bool ok) {
  union { bf16x8_t h; int4 q; } u;
  u.h = v;
  u.q.x = ok ? u.q.x : 0;
  u.q.y = ok ? u.q.y : 0;
  u.q.z = ok ? u.q.z : 0;
  u.q.w = ok ? u.q.w : 0;
  return u.h;
}

template <int LROW>
__device__ inline bf16x8_t tr_frag_g(const bf16* img, int mbase, int colbase,
                                     int lane) {
  const int j15 = lane & 15, q = lane >> 4;
  const int row = mbase + q * 8 + (j15 >> 2);
  const int col = colbase + 4 * (j15 & 3);
  auto p0 = (LDS3G trvec_g*)((LDS3G bf16*)img + (long)row * LROW + col);
  auto p1 = (LDS3G trvec_g*)((LDS3G bf16*)img + (long)(row + 4) * LROW + col);
  trvec_g lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
  trvec_g hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  union { struct { trvec_g a, b; } t; bf16x8_t v; } u;
  u.t.a = lo;
  u.t.b = hi;
  return u.v;
}

// ------------------------------------------------------------- forward
// y[m][co] = sum_c gn(x[m][c]) * w[co][c]; same double-buffered
// register-staged pipeline + LDS-bounced epilogue as conv_fwd_kernel.
template <int BM, int BN, int WM, int WN>
__global__ void __launch_bounds__(CONV_BLOCK)
gnconv1x1_fwd_kernel(const GnCParams p) {
  constexpr int WTM = BM / WM;
  constexpr int WTN = BN / WN;
  constexpr int FA = WTM / 16;
  constexpr int FB = WTN / 16;
  constexpr int LDA = BK + 8;
  constexpr int LDB = BK + 8;
  constexpr int ACH = BM * (BK / 8);
  constexpr int BCH = BN * (BK / 8);
  constexpr int APT = (ACH + CONV_BLOCK - 1) / CONV_BLOCK;
  constexpr int BPT = (BCH + CONV_BLOCK - 1) / CONV_BLOCK;

  __shared__ bf16 a_lds[2][BM * LDA];
  __shared__ bf16 b_lds[2][BN * LDB];
  __shared__ short s_oct[GNC_MAXOCT];
  __shared__ float s_ga[2][GNC_NSPAN_FWD * BK];  // per k-chunk parity
  __shared__ float s_gb[2][GNC_NSPAN_FWD * BK];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave / WN, wc = wave % WN;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int nb0 = (int)p.fd_hw.div((unsigned)m0);  // block rows' first n
  gnc_stage_oct(p, s_oct, t, CONV_BLOCK);

  f32x4 acc[FA][FB];
#pragma unroll
  for (int i = 0; i < FA; ++i)
#pragma unroll
    for (int j = 0; j < FB; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  bf16x8_t areg[APT], breg[BPT];
  bool aok[APT];

  auto load_tile = [&](int kt) {
#pragma unroll
    for (int u = 0; u < APT; ++u) {
      const int c = t + u * CONV_BLOCK;
      const int row = c / (BK / 8);
      const int k8 = (c % (BK / 8)) * 8;
      areg[u] = gn_raw8(p, s_oct, m0 + row, kt + k8, p.M, &aok[u]);
      if (c >= ACH) { areg[u] = gnc_zero8(); aok[u] = false; }
    }
#pragma unroll
    for (int u = 0; u < BPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      const int nrow = c / (BK / 8);
      const int k8 = (c % (BK / 8)) * 8;
      const int n = n0 + nrow;
      const int k = kt + k8;
      const bool ok = (c < BCH) & (n < p.Co) & (k < p.C);
      const long off = ok ? (long)n * p.C + k : 0;
      breg[u] = gnc_mask8(*reinterpret_cast<const bf16x8_t*>(p.w + off), ok);
    }
  };

  auto write_tile = [&](int buf, int kt) {
    const int par = (kt / BK) & 1;
#pragma unroll
    for (int u = 0; u < APT; ++u) {
      const int c = t + u * CONV_BLOCK;
      if (c < ACH) {
        const int row = c / (BK / 8);
        const int k8 = (c % (BK / 8)) * 8;
        const int n = (int)p.fd_hw.div((unsigned)(m0 + row));
        const int base = (n - nb0) * BK + k8;
        *reinterpret_cast<bf16x8_t*>(&a_lds[buf][row * LDA + k8]) =
            gnc_apply8(p, areg[u], s_ga[par], s_gb[par], base, aok[u]);
      }
    }
#pragma unroll
    for (int u = 0; u < BPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      if (c < BCH) {
        const int nrow = c / (BK / 8);
        const int k8 = (c % (BK / 8)) * 8;
        *reinterpret_cast<bf16x8_t*>(&b_lds[buf][nrow * LDB + k8]) = breg[u];
      }
    }
  };

  auto stage_ab = [&](int kt) {
    gnc_stage_ab(p, s_ga[(kt / BK) & 1], s_gb[(kt / BK) & 1], kt, BK, nb0,
                 GNC_NSPAN_FWD, t, CONV_BLOCK);
  };

  stage_ab(0);
  __syncthreads();
  load_tile(0);
  write_tile(0, 0);
  stage_ab(BK);
  __syncthreads();

  int buf = 0;
  for (int kt = 0; kt < p.C; kt += BK) {
    const bool more = kt + BK < p.C;
    if (more) load_tile(kt + BK);

    bf16x8_t afrag[FA], bfrag[FB];
#pragma unroll
    for (int i = 0; i < FA; ++i) {
      const int row = wr * WTM + i * 16 + (lane & 15);
      afrag[i] = *reinterpret_cast<const bf16x8_t*>(
          &a_lds[buf][row * LDA + (lane >> 4) * 8]);
    }
#pragma unroll
    for (int j = 0; j < FB; ++j) {
      const int col = wc * WTN + j * 16 + (lane & 15);
      bfrag[j] = *reinterpret_cast<const bf16x8_t*>(
          &b_lds[buf][col * LDB + (lane >> 4) * 8]);
    }
#pragma unroll
    for (int i = 0; i < FA; ++i)
#pragma unroll
      for (int j = 0; j < FB; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    if (more) write_tile(buf ^ 1, kt + BK);
    if (kt + 2 * BK < p.C) stage_ab(kt + 2 * BK);
    __syncthreads();
    buf ^= 1;
  }

  constexpr int LDO = BN + 8;
  // bounce buffer: a_lds when it fits 32 output rows, else b_lds
  static_assert(BM * LDA >= 32 * LDO || BN * LDB >= 32 * LDO,
                "epilogue chunk must fit an LDS buffer");
  bf16* o_lds = (BM * LDA >= 32 * LDO) ? a_lds[0] : b_lds[0];
#pragma unroll
  for (int ch = 0; ch < BM / 32; ++ch) {
    __syncthreads();
#pragma unroll
    for (int i = 0; i < FA; ++i) {
      if ((wr * WTM + i * 16) / 32 != ch) continue;
#pragma unroll
      for (int j = 0; j < FB; ++j) {
        const int col = wc * WTN + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = (wr * WTM + i * 16 + (lane >> 4) * 4 + r) & 31;
          o_lds[row * LDO + col] = __float2bfloat16(acc[i][j][r]);
        }
      }
    }
    __syncthreads();
    for (int c = t; c < 32 * (BN / 8); c += CONV_BLOCK) {
      const int row = c / (BN / 8);
      const int c8 = (c % (BN / 8)) * 8;
      const int m = m0 + ch * 32 + row;
      if (m >= p.M) continue;
      if (n0 + c8 + 7 < p.Co) {
        *reinterpret_cast<bf16x8_t*>(p.y + (long)m * p.Co + n0 + c8) =
            *reinterpret_cast<const bf16x8_t*>(&o_lds[row * LDO + c8]);
      } else {
        for (int j = 0; j < 8 && n0 + c8 + j < p.Co; ++j)
          p.y[(long)m * p.Co + n0 + c8 + j] = o_lds[row * LDO + c8 + j];
      }
    }
  }
}

// ----------------------------------------------------------- weight grad
// dW[co][c] = sum_m dy[m][co] * gn(x[m][c]) — conv_wrw's natural-stage +
// transpose-read structure with the x loader normalizing on the fly.
template <int BCO, int BKN>
__global__ void __launch_bounds__(CONV_BLOCK)
gnconv1x1_wrw_kernel(const GnCParams p) {
  constexpr int BM = 64;
  constexpr int WTN = BKN / 4;
  constexpr int FA = BCO / 16;
  constexpr int FB = WTN / 16;
  constexpr int LD = BCO + 8;
  constexpr int LX = BKN + 8;
  constexpr int DCH = BM * (BCO / 8);
  constexpr int XCH = BM * (BKN / 8);
  constexpr int DPT = (DCH + CONV_BLOCK - 1) / CONV_BLOCK;
  constexpr int XPT = (XCH + CONV_BLOCK - 1) / CONV_BLOCK;

  __shared__ bf16 dy_t[2][BM * LD];
  __shared__ bf16 x_t[2][BM * LX];
  __shared__ short s_oct[GNC_MAXOCT];
  __shared__ float s_ga[2][GNC_NSPAN_WRW * BKN];
  __shared__ float s_gb[2][GNC_NSPAN_WRW * BKN];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int co0 = blockIdx.x * BCO;
  const int k0 = blockIdx.y * BKN;
  const int mstart = blockIdx.z * p.m_per_split;
  const int mend = min(p.M, mstart + p.m_per_split);

  f32x4 acc[FA][FB];
#pragma unroll
  for (int i = 0; i < FA; ++i)
#pragma unroll
    for (int j = 0; j < FB; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  bf16x8_t dreg[DPT], xreg[XPT];
  bool xok[XPT];

  auto load_chunk = [&](int mt) {
#pragma unroll
    for (int u = 0; u < DPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      const int mm = c / (BCO / 8);
      const int c8 = (c % (BCO / 8)) * 8;
      const int m = mt + mm;
      const bool ok = (c < DCH) & (m < mend) & (co0 + c8 + 7 < p.Co);
      const long off = ok ? (long)m * p.Co + co0 + c8 : 0;
      bf16x8_t v =
          gnc_mask8(*reinterpret_cast<const bf16x8_t*>(p.dy + off), ok);
      if (!ok && c < DCH && m < mend && co0 + c8 < p.Co) {
        bf16* vv = reinterpret_cast<bf16*>(&v);
        for (int j = 0; j < 8 && co0 + c8 + j < p.Co; ++j)
          vv[j] = p.dy[(long)m * p.Co + co0 + c8 + j];
      }
      dreg[u] = v;
    }
#pragma unroll
    for (int u = 0; u < XPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      const int mm = c / (BKN / 8);
      const int k8 = (c % (BKN / 8)) * 8;
      xreg[u] = gn_raw8(p, s_oct, mt + mm, k0 + k8, mend, &xok[u]);
      if (c >= XCH) { xreg[u] = gnc_zero8(); xok[u] = false; }
    }
  };

  auto write_chunk = [&](int buf, int mt) {
#pragma unroll
    for (int u = 0; u < DPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      if (c < DCH) {
        const int mm = c / (BCO / 8);
        const int c8 = (c % (BCO / 8)) * 8;
        *reinterpret_cast<bf16x8_t*>(&dy_t[buf][mm * LD + c8]) = dreg[u];
      }
    }
    const int par = (mt / BM) & 1;
    const int nb = (int)p.fd_hw.div((unsigned)mt);
#pragma unroll
    for (int u = 0; u < XPT; ++u) {
      const int c = t + u * CONV_BLOCK;
      if (c < XCH) {
        const int mm = c / (BKN / 8);
        const int k8 = (c % (BKN / 8)) * 8;
        const int n = (int)p.fd_hw.div((unsigned)(mt + mm));
        const int base = (n - nb) * BKN + k8;
        *reinterpret_cast<bf16x8_t*>(&x_t[buf][mm * LX + k8]) =
            gnc_apply8(p, xreg[u], s_ga[par], s_gb[par], base, xok[u]);
      }
    }
  };

  auto stage_ab = [&](int mt) {
    gnc_stage_ab(p, s_ga[(mt / BM) & 1], s_gb[(mt / BM) & 1], k0, BKN,
                 (int)p.fd_hw.div((unsigned)mt), GNC_NSPAN_WRW, t,
                 CONV_BLOCK);
  };

  gnc_stage_oct(p, s_oct, t, CONV_BLOCK);
  stage_ab(mstart);
  __syncthreads();
  load_chunk(mstart);
  write_chunk(0, mstart);
  stage_ab(mstart + BM);
  __syncthreads();

  int buf = 0;
  for (int mt = mstart; mt < mend; mt += BM) {
    const bool more = mt + BM < mend;
    if (more) load_chunk(mt + BM);

#pragma unroll
    for (int sub = 0; sub < BM / 32; ++sub) {
      bf16x8_t afrag[FA], bfrag[FB];
#pragma unroll
      for (int i = 0; i < FA; ++i)
        afrag[i] = tr_frag_g<LD>(dy_t[buf], sub * 32, i * 16, lane);
#pragma unroll
      for (int j = 0; j < FB; ++j)
        bfrag[j] =
            tr_frag_g<LX>(x_t[buf], sub * 32, wave * WTN + j * 16, lane);
#pragma unroll
      for (int i = 0; i < FA; ++i)
#pragma unroll
        for (int j = 0; j < FB; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    if (more) write_chunk(buf ^ 1, mt + BM);
    if (mt + 2 * BM < mend) stage_ab(mt + 2 * BM);
    __syncthreads();
    buf ^= 1;
  }

  float* slab = p.dw + (long)blockIdx.z * p.Co * p.C;
#pragma unroll
  for (int i = 0; i < FA; ++i) {
#pragma unroll
    for (int j = 0; j < FB; ++j) {
      const int kk = k0 + wave * WTN + j * 16 + (lane & 15);
      if (kk >= p.C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + i * 16 + (lane >> 4) * 4 + r;
        if (co < p.Co) slab[(long)co * p.C + kk] = acc[i][j][r];
      }
    }
  }
}

// --------------------------------------------------------------- launch
static void gnc_fill(GnCParams& p, const void* const* xs, const int* starts,
                     int nseg, const float* mean, const float* rstd,
                     const float* gamma, const float* beta, int N, int HW,
                     int C, int G, int Co, int relu) {
  p.segs.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    p.segs.p[i] = (const bf16*)xs[i];
    p.segs.start[i] = starts[i];
  }
  p.segs.start[nseg] = starts[nseg];
  p.mean = mean;
  p.rstd = rstd;
  p.gamma = gamma;
  p.beta = beta;
  p.M = N * HW;
  p.HW = HW;
  p.C = C;
  p.G = G;
  p.Co = Co;
  p.relu = relu;
  p.cg = C / G;
  p.fd_hw.init(HW);
  p.fd_cg.init(C / G);
}

extern "C" void dlb_gnconv1x1_fwd(const void* const* xs, const int* starts,
                                  int nseg, const float* mean,
                                  const float* rstd, const float* gamma,
                                  const float* beta, const void* w, void* y,
                                  int N, int HW, int C, int G, int Co,
                                  int relu, hipStream_t stream) {
  GnCParams p{};
  gnc_fill(p, xs, starts, nseg, mean, rstd, gamma, beta, N, HW, C, G, Co,
           relu);
  p.w = (const bf16*)w;
  p.y = (bf16*)y;
  // Co=128 layers give grid.y=1; when the 128-row tiling can't fill the
  // 256 CUs (small-HW DenseNet stages), halve BM to double the blocks.
  if ((long)cdiv(p.M, 128) * cdiv(Co, 128) < 400) {
    dim3 grid(cdiv(p.M, 64), cdiv(Co, 128));
    hipLaunchKernelGGL((gnconv1x1_fwd_kernel<64, 128, 1, 4>), grid,
                       dim3(CONV_BLOCK), 0, stream, p);
    return;
  }
  dim3 grid(cdiv(p.M, 128), cdiv(Co, 128));
  hipLaunchKernelGGL((gnconv1x1_fwd_kernel<128, 128, 2, 2>), grid,
                     dim3(CONV_BLOCK), 0, stream, p);
}

extern "C" int dlb_conv_wrw_nsplits(int N, int OH, int OW, int Ci, int Co,
                                    int R, int S);

extern "C" int dlb_gnconv1x1_wrw(const void* const* xs, const int* starts,
                                 int nseg, const float* mean,
                                 const float* rstd, const float* gamma,
                                 const float* beta, const void* dy, float* dw,
                                 int N, int HW, int C, int G, int Co,
                                 int relu, int splits, hipStream_t stream) {
  GnCParams p{};
  gnc_fill(p, xs, starts, nseg, mean, rstd, gamma, beta, N, HW, C, G, Co,
           relu);
  p.dy = (const bf16*)dy;
  p.dw = dw;
  p.m_per_split = cdiv(cdiv(p.M, splits), 64) * 64;
  splits = cdiv(p.M, p.m_per_split);
  if (Co >= 128) {
    dim3 grid(cdiv(Co, 128), cdiv(C, 128), splits);
    hipLaunchKernelGGL((gnconv1x1_wrw_kernel<128, 128>), grid,
                       dim3(CONV_BLOCK), 0, stream, p);
  } else if (Co >= 64) {
    dim3 grid(cdiv(Co, 64), cdiv(C, 128), splits);
    hipLaunchKernelGGL((gnconv1x1_wrw_kernel<64, 128>), grid,
                       dim3(CONV_BLOCK), 0, stream, p);
  } else {
    dim3 grid(cdiv(Co, 32), cdiv(C, 128), splits);
    hipLaunchKernelGGL((gnconv1x1_wrw_kernel<32, 128>), grid,
                       dim3(CONV_BLOCK), 0, stream, p);
  }
  return splits;
}
