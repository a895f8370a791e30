// Halo-staged 3x3/stride-1/pad-1 conv kernels (gfx950) — fwd + wrw.
//
// The generic implicit-GEMM kernels (conv.hip) re-read every input pixel
// 9 times through the im2col addressing (measured: 3x3 layers are L2/HBM
// bound on that duplication).  These kernels stage the input tile WITH
// ITS HALO in LDS once per ci-chunk and synthesize all nine taps from
// LDS, cutting global input traffic to ~1.4x (fwd) / 1x (wrw).
//
// Coverage (dispatched from the launchers in conv.hip's host code):
//   fwd: R=S=3, stride=1, pad=1, Ci%32==0 — the zoo's hot 3x3s
//        (DenseNet growth convs, ResNet/RegNet body 3x3s).
//   wrw: same plus OH*OW%32==0 (>=8x8 feature maps).
// Everything else falls back to the implicit-GEMM kernels.

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define HBLOCK 256
#define LDS3H __attribute__((address_space(3)))
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 trvec_h;

// MFMA fragment via hardware transpose reads from an LDS image laid out
// [k rows][n cols] with row stride LROW (same recipe as conv.hip; lane
// semantics verified on hardware, tools/mfma_probe/).
template <int LROW>
__device__ inline bf16x8_t tr_frag_h(const __hip_bfloat16* img, int mbase,
                                     int colbase, int lane) {
  const int j15 = lane & 15, q = lane >> 4;
  const int row = mbase + q * 8 + (j15 >> 2);
  const int col = colbase + 4 * (j15 & 3);
  auto p0 = (LDS3H trvec_h*)((LDS3H __hip_bfloat16*)img + (long)row * LROW + col);
  auto p1 = (LDS3H trvec_h*)((LDS3H __hip_bfloat16*)img +
                             (long)(row + 4) * LROW + col);
  trvec_h lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
  trvec_h hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  union { struct { trvec_h a, b; } t; bf16x8_t v; } u;
  u.t.a = lo;
  u.t.b = hi;
  return u.v;
}
#define CI_CHUNK 32
#define HPAD 40  // padded ci stride in halo LDS rows (conflict-free, 16B-aligned)

// ---------------------------------------------------------------- forward
// Output tile: TH x TW pixels (8x16) for one sample, BN output channels.
// K-loop over ci chunks of 32; inner loop over the 9 taps.
// WTR=false: w is the conv weight in channels_last memory [Co][3][3][Ci]
//            (B rows staged naturally, k = ci contiguous).
// WTR=true:  computes the DATA GRADIENT of a 3x3/s1/p1 conv as a halo
//            forward over dy: roles swap (kernel Ci = original Co is the
//            reduction, kernel Co = original Ci is the output), w stays
//            in NATURAL channels_last memory [k=Co_orig][3][3][n=Ci_orig]
//            — rows are staged k-major into LDS and B fragments use
//            hardware transpose reads with the tap FLIPPED (r'=2-r,
//            s'=2-s).  This replaces the flip+copy weight transform the
//            python path used to materialize per call.
template <int BN, bool WTR>
__global__ void __launch_bounds__(HBLOCK)
conv3x3_fwd_halo(const bf16* __restrict__ x, const bf16* __restrict__ w,
                 bf16* __restrict__ y, const float* __restrict__ bias,
                 const int N, const int H, const int W, const int Ci,
                 const int Co) {
  constexpr int TH = 8, TW = 16;
  constexpr int HH = TH + 2, HW = TW + 2;  // 10 x 18 halo
  constexpr int FA = 2;               // 32 pixels per wave (4 waves x 32)
  constexpr int FB = BN / 16;
  constexpr int WK = 9 * CI_CHUNK;    // staged weight k-extent (288)
  constexpr int WLD = WK + 8;
  constexpr int LW = BN + 8;          // WTR row stride ([tap*32+k][n])
  constexpr int WSZ = WTR ? 9 * CI_CHUNK * LW : BN * WLD;

  __shared__ bf16 halo[HH * HW * HPAD];
  __shared__ bf16 wlds[WSZ];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;

  // block -> (n, oh0, ow0, co0)
  const int tiles_w = cdiv(W, TW);
  const int tiles_h = cdiv(H, TH);
  int bid = blockIdx.x;
  const int tw_i = bid % tiles_w; bid /= tiles_w;
  const int th_i = bid % tiles_h; bid /= tiles_h;
  const int n = bid;
  const int oh0 = th_i * TH, ow0 = tw_i * TW;
  const int co0 = blockIdx.y * BN;
  const int K = 9 * Ci;

  f32x4 acc[FA][FB];
#pragma unroll
  for (int i = 0; i < FA; ++i)
#pragma unroll
    for (int j = 0; j < FB; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int ci0 = 0; ci0 < Ci; ci0 += CI_CHUNK) {
    // ---- stage halo (10 x 18 x 32), vectorized over ci
    constexpr int HCH = HH * HW * (CI_CHUNK / 8);  // 720 chunks
    for (int c = t; c < HCH; c += HBLOCK) {
      const int c8 = (c % (CI_CHUNK / 8)) * 8;
      const int pix = c / (CI_CHUNK / 8);
      const int hh = pix / HW, ww = pix % HW;
      const int ih = oh0 - 1 + hh, iw = ow0 - 1 + ww;
      // branchless: always load a clamped address, mask invalid lanes
      const bool ok = (ih >= 0) & (ih < H) & (iw >= 0) & (iw < W);
      const long off = ok ? (((long)n * H + ih) * W + iw) * Ci + ci0 + c8 : 0;
      bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(x + off);
      union { bf16x8_t h; int4 q; } u2; u2.h = v;
      u2.q.x = ok ? u2.q.x : 0; u2.q.y = ok ? u2.q.y : 0;
      u2.q.z = ok ? u2.q.z : 0; u2.q.w = ok ? u2.q.w : 0;
      *reinterpret_cast<bf16x8_t*>(&halo[(hh * HW + ww) * HPAD + c8]) = u2.h;
    }
    if (WTR) {
      // ---- stage natural weight rows [tap*32 + k][n] (n contiguous)
      constexpr int WCH2 = 9 * CI_CHUNK * (BN / 8);
      for (int c = t; c < WCH2; c += HBLOCK) {
        const int c8 = (c % (BN / 8)) * 8;           // n (= original ci)
        const int rest = c / (BN / 8);
        const int tap = rest % 9;
        const int kl = rest / 9;                     // k (= original co)
        const bool ok = co0 + c8 < Co;
        const long off =
            ok ? ((long)(ci0 + kl) * 9 + tap) * Co + co0 + c8 : 0;
        bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(w + off);
        union { bf16x8_t h; int4 q; } u2; u2.h = v;
        u2.q.x = ok ? u2.q.x : 0; u2.q.y = ok ? u2.q.y : 0;
        u2.q.z = ok ? u2.q.z : 0; u2.q.w = ok ? u2.q.w : 0;
        *reinterpret_cast<bf16x8_t*>(
            &wlds[(tap * CI_CHUNK + kl) * LW + c8]) = u2.h;
      }
    } else {
    // ---- stage weights [co][tap*32+ci] from w[co][tap*Ci + ci]
    constexpr int WCH = BN * 9 * (CI_CHUNK / 8);
    for (int c = t; c < WCH; c += HBLOCK) {
      const int c8 = (c % (CI_CHUNK / 8)) * 8;
      const int rest = c / (CI_CHUNK / 8);
      const int tap = rest % 9;
      const int co = rest / 9;
      const bool ok = co0 + co < Co;
      const long off = ok ? (long)(co0 + co) * K + tap * Ci + ci0 + c8 : 0;
      bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(w + off);
      union { bf16x8_t h; int4 q; } u2; u2.h = v;
      u2.q.x = ok ? u2.q.x : 0; u2.q.y = ok ? u2.q.y : 0;
      u2.q.z = ok ? u2.q.z : 0; u2.q.w = ok ? u2.q.w : 0;
      *reinterpret_cast<bf16x8_t*>(&wlds[co * WLD + tap * CI_CHUNK + c8]) = u2.h;
    }
    }
    __syncthreads();

#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int r = tap / 3, s = tap % 3;
      bf16x8_t afrag[FA], bfrag[FB];
#pragma unroll
      for (int i = 0; i < FA; ++i) {
        const int p = wave * 32 + i * 16 + (lane & 15);
        const int py = p / TW, px = p % TW;
        afrag[i] = *reinterpret_cast<const bf16x8_t*>(
            &halo[((py + r) * HW + px + s) * HPAD + (lane >> 4) * 8]);
      }
#pragma unroll
      for (int j = 0; j < FB; ++j) {
        if (WTR) {
          bfrag[j] = tr_frag_h<LW>(wlds, (8 - tap) * CI_CHUNK, j * 16, lane);
        } else {
          const int co = j * 16 + (lane & 15);
          bfrag[j] = *reinterpret_cast<const bf16x8_t*>(
              &wlds[co * WLD + tap * CI_CHUNK + (lane >> 4) * 8]);
        }
      }
#pragma unroll
      for (int i = 0; i < FA; ++i)
#pragma unroll
        for (int j = 0; j < FB; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue through LDS (coalesced 16-byte stores; the fragment
  // layout's native stores are 2-byte at 4 rows per lane).  wlds is
  // free after the last MFMA: BN * WLD >= NPIXELS(128) * (BN + 8).
  constexpr int LDO = BN + 8;
  static_assert(WSZ >= 128 * LDO, "epilogue tile must fit wlds");
  bf16* o_lds = wlds;
  __syncthreads();
#pragma unroll
  for (int i = 0; i < FA; ++i) {
#pragma unroll
    for (int j = 0; j < FB; ++j) {
      const int col = j * 16 + (lane & 15);
      const float b = bias ? (co0 + col < Co ? bias[co0 + col] : 0.f) : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int p = wave * 32 + i * 16 + (lane >> 4) * 4 + rr;
        o_lds[p * LDO + col] = __float2bfloat16(acc[i][j][rr] + b);
      }
    }
  }
  __syncthreads();
  for (int c = t; c < 128 * (BN / 8); c += HBLOCK) {
    const int p = c / (BN / 8);
    const int c8 = (c % (BN / 8)) * 8;
    const int oh = oh0 + p / TW, ow = ow0 + p % TW;
    if (oh >= H || ow >= W) continue;
    bf16* dst = y + (((long)n * H + oh) * W + ow) * Co + co0 + c8;
    if (co0 + c8 + 7 < Co)
      *reinterpret_cast<bf16x8_t*>(dst) =
          *reinterpret_cast<const bf16x8_t*>(&o_lds[p * LDO + c8]);
    else
      for (int j = 0; j < 8 && co0 + c8 + j < Co; ++j)
        dst[j] = o_lds[p * LDO + c8 + j];
  }
}

// ------------------------------------------------------------------ wrw
// dW[co][tap*Ci+ci] += sum_m dy[m][co] * x[tap(m)][ci]
// Block: [32 co] x [9 taps x 32 ci = 288 k-cols], m-split over chunks of
// 128 output pixels (whole output rows).  Per chunk: dy staged
// transposed ([co][m]) with vector fragment reads; the x halo band is
// staged once (global traffic 1x) and B-fragments gather from it
// directly (8 scalar LDS reads per fragment — exactly the consumption,
// cheaper than materializing the im2col image).
// Requires OH*OW % 128 == 0... actually % BM == 0 handled by loop guard;
// W must be a power of two <= 32 (runtime shifts).
__global__ void __launch_bounds__(HBLOCK)
conv3x3_wrw_halo(const bf16* __restrict__ x, const bf16* __restrict__ dy,
                 float* __restrict__ dw, const int N, const int H,
                 const int W, const int Ci, const int Co,
                 const int m_per_split, const int wshift) {
  constexpr int BCO = 32;
  constexpr int BM = 128;             // reduction chunk (whole rows)
  constexpr int LMD = BCO + 8;        // natural [m][co] rows (16B-aligned)
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  // 18 k-fragments split 5/5/4/4 across the 4 waves
  const int fb_count = (wave < 2) ? 5 : 4;
  const int fb_base = (wave < 2) ? wave * 5 : 10 + (wave - 2) * 4;

  __shared__ bf16 dy_t[BM * LMD];         // natural [m][co] image
  __shared__ bf16 halo[204 * CI_CHUNK];   // band: (BM/W+2) x (W+2) pixels

  const int co0 = blockIdx.x * BCO;
  const int ci0 = blockIdx.y * CI_CHUNK;
  const int K = 9 * Ci;
  const int M = N * H * W;
  const int mstart = blockIdx.z * m_per_split;
  const int mend = min(M, mstart + m_per_split);
  const int wmask = W - 1;

  const int rows_per_chunk = BM >> wshift;        // BM/W (W<=32 => >=4)
  const int hh_rows = rows_per_chunk + 2;
  const int hw_cols = W + 2;

  f32x4 acc[2][5];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 5; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int mt = mstart; mt < mend; mt += BM) {
    // ---- stage dy transposed [co][mm]
    constexpr int DCH = BM * (BCO / 8);
    for (int c = t; c < DCH; c += HBLOCK) {
      const int mm = c / (BCO / 8);
      const int c8 = (c % (BCO / 8)) * 8;
      const int m = mt + mm;
      bf16x8_t v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (m < mend && co0 + c8 + 7 < Co)
        v = *reinterpret_cast<const bf16x8_t*>(dy + (long)m * Co + co0 + c8);
      else if (m < mend) {
        bf16* vv = reinterpret_cast<bf16*>(&v);
        for (int j = 0; j < 8 && co0 + c8 + j < Co; ++j)
          vv[j] = dy[(long)m * Co + co0 + c8 + j];
      }
      *reinterpret_cast<bf16x8_t*>(&dy_t[mm * LMD + c8]) = v;
    }
    // ---- stage x band: rows [py0-1 .. py0+rows], cols [-1..W]
    const int n = mt / (H * W);
    const int py0 = (mt % (H * W)) >> wshift;   // chunk starts at col 0
    const int band_ch = hh_rows * hw_cols * (CI_CHUNK / 8);
    for (int c = t; c < band_ch; c += HBLOCK) {
      const int c8 = (c % (CI_CHUNK / 8)) * 8;
      const int pix = c / (CI_CHUNK / 8);
      const int hh = pix / hw_cols, ww = pix % hw_cols;
      const int ih = py0 - 1 + hh, iw = ww - 1;
      bf16x8_t v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (ih >= 0 && ih < H && iw >= 0 && iw < W)
        v = *reinterpret_cast<const bf16x8_t*>(
            x + (((long)n * H + ih) * W + iw) * Ci + ci0 + c8);
      *reinterpret_cast<bf16x8_t*>(&halo[(hh * hw_cols + ww) * CI_CHUNK + c8]) = v;
    }
    __syncthreads();

    // ---- 4 MFMA sub-steps of 32 m each
#pragma unroll
    for (int sub = 0; sub < 4; ++sub) {
      const int msub = sub * 32;
      bf16x8_t afrag[2];
      {
        const int j15 = lane & 15, q = lane >> 4;
        const int mrow = msub + q * 8 + (j15 >> 2);
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          const int col = i * 16 + 4 * (j15 & 3);
          auto p0 = (LDS3H trvec_h*)((LDS3H bf16*)dy_t + mrow * LMD + col);
          auto p1 = (LDS3H trvec_h*)((LDS3H bf16*)dy_t +
                                     (mrow + 4) * LMD + col);
          trvec_h lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
          trvec_h hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
          union { struct { trvec_h a, b; } t2; bf16x8_t v; } u;
          u.t2.a = lo;
          u.t2.b = hi;
          afrag[i] = u.v;
        }
      }
#pragma unroll
      for (int j = 0; j < 5; ++j) {
        if (j >= fb_count) continue;
        // fragment columns are 16 consecutive k = (tap, cil); a 4-col
        // address run never crosses a tap boundary (16 | frag base)
        const int base16 = (fb_base + j) * 16;
        const int tap = base16 / CI_CHUNK;
        const int r = tap / 3, sxx = tap % 3;
        const int j15 = lane & 15, q = lane >> 4;
        const int mm = msub + q * 8 + (j15 >> 2);
        const int py = mm >> wshift, px = mm & wmask;
        const int cil = (base16 % CI_CHUNK) + 4 * (j15 & 3);
        const long rowoff =
            ((long)(py + r) * hw_cols + px + sxx) * CI_CHUNK + cil;
        const long rowoff4 =
            ((long)((mm + 4) >> wshift) + r) * hw_cols * CI_CHUNK +
            (long)(((mm + 4) & wmask) + sxx) * CI_CHUNK + cil;
        auto p0 = (LDS3H trvec_h*)((LDS3H bf16*)halo + rowoff);
        auto p1 = (LDS3H trvec_h*)((LDS3H bf16*)halo + rowoff4);
        trvec_h lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
        trvec_h hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
        union { struct { trvec_h a, b; } t2; bf16x8_t v; } u;
        u.t2.a = lo;
        u.t2.b = hi;
        bf16x8_t bfrag = u.v;
#pragma unroll
        for (int i = 0; i < 2; ++i)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag, acc[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- publish: dw[co][tap*Ci + ci0 + cil]
  float* slab = dw + (long)blockIdx.z * Co * K;
#pragma unroll
  for (int j = 0; j < 5; ++j) {
    if (j >= fb_count) continue;
    const int klocal = (fb_base + j) * 16 + (lane & 15);
    const int tap = klocal / CI_CHUNK, cil = klocal % CI_CHUNK;
    const int kg = tap * Ci + ci0 + cil;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int co = co0 + i * 16 + (lane >> 4) * 4 + rr;
        if (co < Co) slab[(long)co * K + kg] = acc[i][j][rr];
      }
    }
  }
}

// ---------------------------------------------------------------- launch
extern "C" bool dlb_conv3x3_fwd_halo(const void* x, const void* w, void* y,
                                     const float* bias, int N, int H, int W,
                                     int Ci, int Co, hipStream_t stream) {
  if (Ci % CI_CHUNK != 0) return false;
  const int tiles = N * cdiv(H, 8) * cdiv(W, 16);
  if (Co % 32 != 0 && Co < 32) return false;
  if (Co % 64 == 0 && Co >= 64) {
    dim3 grid(tiles, cdiv(Co, 64));
    hipLaunchKernelGGL((conv3x3_fwd_halo<64, false>), grid, dim3(HBLOCK), 0,
                       stream, (const bf16*)x, (const bf16*)w, (bf16*)y,
                       bias, N, H, W, Ci, Co);
  } else {
    dim3 grid(tiles, cdiv(Co, 32));
    hipLaunchKernelGGL((conv3x3_fwd_halo<32, false>), grid, dim3(HBLOCK), 0,
                       stream, (const bf16*)x, (const bf16*)w, (bf16*)y,
                       bias, N, H, W, Ci, Co);
  }
  return true;
}

// Data gradient of a 3x3/s1/p1 conv via the halo kernel in WTR mode:
// dx[N,H,W,Ci] from dy[N,H,W,Co] and the UNMODIFIED channels_last weight.
extern "C" bool dlb_conv3x3_bwd_halo(const void* dy, const void* w, void* dx,
                                     int N, int H, int W, int Ci, int Co,
                                     hipStream_t stream) {
  // kernel roles: "Ci" = Co (reduction), "Co" = Ci (output channels)
  if (Co % CI_CHUNK != 0) return false;
  if (Ci % 8 != 0) return false;
  const int tiles = N * cdiv(H, 8) * cdiv(W, 16);
  if (Ci % 64 == 0 && Ci >= 64) {
    dim3 grid(tiles, cdiv(Ci, 64));
    hipLaunchKernelGGL((conv3x3_fwd_halo<64, true>), grid, dim3(HBLOCK), 0,
                       stream, (const bf16*)dy, (const bf16*)w, (bf16*)dx,
                       (const float*)nullptr, N, H, W, Co, Ci);
  } else {
    dim3 grid(tiles, cdiv(Ci, 32));
    hipLaunchKernelGGL((conv3x3_fwd_halo<32, true>), grid, dim3(HBLOCK), 0,
                       stream, (const bf16*)dy, (const bf16*)w, (bf16*)dx,
                       (const float*)nullptr, N, H, W, Co, Ci);
  }
  return true;
}

extern "C" bool dlb_conv3x3_wrw_halo(const void* x, const void* dy, float* dw,
                                     int N, int H, int W, int Ci, int Co,
                                     int splits, hipStream_t stream) {
  if (Ci % CI_CHUNK != 0 || Co % 8 != 0) return false;
  if (W > 32 || (W & (W - 1)) != 0 || (H * W) % 128 != 0) return false;
  const int M = N * H * W;
  int wshift = 0;
  while ((1 << wshift) < W) ++wshift;
  int m_per_split = cdiv(cdiv(M, splits), 128) * 128;
  splits = cdiv(M, m_per_split);
  dim3 grid(cdiv(Co, 32), Ci / CI_CHUNK, splits);
  hipLaunchKernelGGL(conv3x3_wrw_halo, grid, dim3(HBLOCK), 0, stream,
                     (const bf16*)x, (const bf16*)dy, dw, N, H, W, Ci, Co,
                     m_per_split, wshift);
  return true;
}
