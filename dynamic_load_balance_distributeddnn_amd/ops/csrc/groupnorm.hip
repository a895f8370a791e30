// Fused GroupNorm(+ReLU) forward/backward for NHWC activations (gfx950).
//
// Why this kernel exists: the zoo applies GroupNorm before/after nearly
// every conv and PyTorch-ROCm's eager path runs it as
// 5+ fp32 kernels with bf16<->fp32 casts around them — measured ~55% of
// the DenseNet-121 step (profiles/).  Here it is:
//   fwd: 2 streaming passes over the activation (stats, then normalize
//        + affine + optional ReLU), bf16 in/out, fp32 statistics.
//   bwd: 2 passes (group sums + dgamma/dbeta, then dx), ReLU mask
//        recomputed from saved stats so no mask tensor is stored.
//
// Layout: NHWC ([N, HW, C] contiguous in C) — the layout the whole CV
// path runs in (channels_last).  Grid = (N, channel-chunks): GroupNorm's
// reductions are per-(sample, group), so a chunk of WHOLE groups is
// fully independent of every other chunk — slicing C (at group-aligned
// octet granularity, <= 128 octets so every thread owns exactly ONE
// channel octet) multiplies blocks without any cross-block coordination.
// Threads are (pixel, channel-octet): a thread owns a FIXED run of 8
// channels across pixels strided by TP, so partial sums live in 8
// statically-indexed registers (no scratch — CDNA guide rule 20).
//
// Reductions are DETERMINISTIC by construction: per-thread partials go
// to a [TP][cspan] LDS image with plain stores (TP*cspan <= 2048
// floats), then fixed-order tree/serial merges produce the group sums
// and per-channel dgamma/dbeta.  The earlier LDS atomicAdd merge both
// serialized under tp contention (+6..+100 us/dispatch, tools/gn_probe)
// and made results order-dependent across runs; dgamma/dbeta global
// atomics had the same two problems and are now per-sample partial rows
// reduced by reduce.hip's colsum kernel.
// All loads are 16-byte bf16x8, fully coalesced.  Requires C%8==0
// (every channel count in the zoo satisfies this).

#include "common.h"

#ifndef GN_BLOCK
#define GN_BLOCK 256
#endif
#define GN_MAXG 64      // num_groups <= 64 covers the zoo (8/16/32)
#define GN_MAXCHUNK 128  // octets per chunk (=> single octet per thread)

typedef __hip_bfloat16 bf16;

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ inline bf16 f2bf(float v) { return __float2bfloat16(v); }

struct Bf16x8 {
  bf16 v[8];
};

// Virtual-concat segment table: the DenseNet residual stream is a concat
// of up to ~50 conv outputs; reading them in place removes the cat copy
// per layer (measured ~7% of the step).  Channel octets never cross a
// segment boundary (every segment's channel count is a multiple of 8).
#define GN_MAXSEG 56
struct GnSegs {
  const bf16* p[GN_MAXSEG];
  int start[GN_MAXSEG + 1];
  int nseg;
};
struct GnSegsMut {           // backward dx outputs, same layout
  bf16* p[GN_MAXSEG];
};

// octet c0 -> (segment base for sample row, local channel, seg width)
__device__ inline const bf16* seg_locate(const GnSegs& sg, int c0, int& cloc,
                                         int& cs) {
  int si = 0;
  while (si + 1 < sg.nseg && c0 >= sg.start[si + 1]) ++si;
  cloc = c0 - sg.start[si];
  cs = sg.start[si + 1] - sg.start[si];
  return sg.p[si];
}

// This block's octet range [o0, o1) and thread mapping.  chunk_oct is
// group-aligned and <= GN_MAXCHUNK, so TCe == span and every active
// thread owns exactly one octet (single-trip).
struct ChunkMap {
  int o0, o1;    // octet range
  int TCe, TP;   // channel-octet threads, pixel stride
  int tc, tp;    // this thread's coordinates
  bool active;
};

__device__ inline ChunkMap chunk_map(int TC, int chunk_oct) {
  ChunkMap m;
  m.o0 = blockIdx.y * chunk_oct;
  m.o1 = min(m.o0 + chunk_oct, TC);
  m.TCe = m.o1 - m.o0;       // <= GN_MAXCHUNK <= GN_BLOCK
  m.TP = GN_BLOCK / m.TCe;
  const int t = threadIdx.x;
  m.tc = t % m.TCe;
  m.tp = t / m.TCe;
  m.active = t < m.TCe * m.TP;
  return m;
}

// ---------------------------------------------------------------- forward
// Dynamic LDS layout: [cspan] gamma | [cspan] beta | [TP*cspan] P0 |
// [TP*cspan] P1  (P* hold per-(tp, channel) partials; TP*cspan <= 2048).
template <bool HASRES>
__global__ void __launch_bounds__(GN_BLOCK)
gn_fwd_kernel(const GnSegs segs, bf16* __restrict__ y,
              const float* __restrict__ gamma, const float* __restrict__ beta,
              float* __restrict__ mean_out, float* __restrict__ rstd_out,
              const bf16* __restrict__ res,
              const int HW, const int C, const int G, const float eps,
              const int relu, const int chunk_oct) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const ChunkMap m = chunk_map(TC, chunk_oct);
  const int t = threadIdx.x;
  const int Cg = C / G;
  const int g0 = (m.o0 << 3) / Cg, g1 = (m.o1 << 3) / Cg;
  const int cbase = m.o0 << 3;
  const int cspan = (m.o1 - m.o0) << 3;

  __shared__ float s_mean[GN_MAXG];
  __shared__ float s_rstd[GN_MAXG];
  extern __shared__ float s_dyn[];
  float* s_ga = s_dyn;
  float* s_be = s_dyn + cspan;
  float* P0 = s_dyn + 2 * cspan;
  float* P1 = P0 + m.TP * cspan;
  for (int c = t; c < cspan; c += GN_BLOCK) {
    s_ga[c] = gamma[cbase + c];
    s_be[c] = beta[cbase + c];
  }

  const int c0 = (m.o0 + m.tc) << 3;
  int cloc, cs;
  const bf16* sb = seg_locate(segs, c0, cloc, cs);
  const bf16* xb = sb + (long)n * HW * cs + cloc;

  if (m.active) {
    float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    #pragma unroll 4
    for (int p = m.tp; p < HW; p += m.TP) {
      Bf16x8 chunk = *reinterpret_cast<const Bf16x8*>(xb + (long)p * cs);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf2f(chunk.v[j]);
        s[j] += v;
        ss[j] += v * v;
      }
    }
    const int pb = m.tp * cspan + (m.tc << 3);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      P0[pb + j] = s[j];
      P1[pb + j] = ss[j];
    }
  }
  __syncthreads();

  // fixed-order group merge (deterministic): one thread per group sums
  // its channels across the TP rows
  const float inv_m = 1.0f / ((float)HW * Cg);
  for (int g = g0 + t; g < g1; g += GN_BLOCK) {
    float su = 0.f, sq = 0.f;
    const int cl0 = g * Cg - cbase;
    for (int tp = 0; tp < m.TP; ++tp)
      for (int c = cl0; c < cl0 + Cg; ++c) {
        su += P0[tp * cspan + c];
        sq += P1[tp * cspan + c];
      }
    float mu = su * inv_m;
    float var = sq * inv_m - mu * mu;
    float r = rsqrtf(var + eps);
    s_mean[g] = mu;
    s_rstd[g] = r;
    mean_out[(long)n * G + g] = mu;
    rstd_out[(long)n * G + g] = r;
  }
  __syncthreads();

  if (!m.active) return;

  bf16* yb = y + (long)n * HW * C;
  const bf16* resb = HASRES ? res + (long)n * HW * C : nullptr;
  float ga[8], be[8], mu[8], rs[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = c0 + j, g = c / Cg;
    ga[j] = s_ga[c - cbase];
    be[j] = s_be[c - cbase];
    mu[j] = s_mean[g];
    rs[j] = s_rstd[g];
  }
  #pragma unroll 4
  for (int p = m.tp; p < HW; p += m.TP) {
    Bf16x8 chunk = *reinterpret_cast<const Bf16x8*>(xb + (long)p * cs);
    Bf16x8 rv;
    if (HASRES)
      rv = *reinterpret_cast<const Bf16x8*>(resb + (long)p * C + c0);
    Bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = (bf2f(chunk.v[j]) - mu[j]) * rs[j] * ga[j] + be[j];
      if (HASRES) v += bf2f(rv.v[j]);
      if (relu) v = fmaxf(v, 0.f);
      out.v[j] = f2bf(v);
    }
    *reinterpret_cast<Bf16x8*>(yb + (long)p * C + c0) = out;
  }
}

// ------------------------------------------------------------- stats-only
// First half of the fused forward: per-(sample, group) mean/rstd over the
// virtual concat, with NO normalize pass.  Used by the fused
// GN->1x1-conv kernels (conv_gn.hip), which normalize at operand-load
// time — the packed y activation is never materialized.
// Dynamic LDS: [TP*cspan] P0 | [TP*cspan] P1.
extern "C" __global__ void __launch_bounds__(GN_BLOCK)
gn_stats_kernel(const GnSegs segs, float* __restrict__ mean_out,
                float* __restrict__ rstd_out, const int HW, const int C,
                const int G, const float eps, const int chunk_oct) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const ChunkMap m = chunk_map(TC, chunk_oct);
  const int t = threadIdx.x;
  const int Cg = C / G;
  const int g0 = (m.o0 << 3) / Cg, g1 = (m.o1 << 3) / Cg;
  const int cbase = m.o0 << 3;
  const int cspan = (m.o1 - m.o0) << 3;

  extern __shared__ float s_dyn[];
  float* P0 = s_dyn;
  float* P1 = s_dyn + m.TP * cspan;

  if (m.active) {
    const int c0 = (m.o0 + m.tc) << 3;
    int cloc, cs;
    const bf16* sb = seg_locate(segs, c0, cloc, cs);
    const bf16* xb = sb + (long)n * HW * cs + cloc;
    float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    #pragma unroll 4
    for (int p = m.tp; p < HW; p += m.TP) {
      Bf16x8 chunk = *reinterpret_cast<const Bf16x8*>(xb + (long)p * cs);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf2f(chunk.v[j]);
        s[j] += v;
        ss[j] += v * v;
      }
    }
    const int pb = m.tp * cspan + (m.tc << 3);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      P0[pb + j] = s[j];
      P1[pb + j] = ss[j];
    }
  }
  __syncthreads();

  const float inv_m = 1.0f / ((float)HW * Cg);
  for (int g = g0 + t; g < g1; g += GN_BLOCK) {
    float su = 0.f, sq = 0.f;
    const int cl0 = g * Cg - cbase;
    for (int tp = 0; tp < m.TP; ++tp)
      for (int c = cl0; c < cl0 + Cg; ++c) {
        su += P0[tp * cspan + c];
        sq += P1[tp * cspan + c];
      }
    float mu = su * inv_m;
    float var = sq * inv_m - mu * mu;
    mean_out[(long)n * G + g] = mu;
    rstd_out[(long)n * G + g] = rsqrtf(var + eps);
  }
}

// ---------------------------------------------------------------- backward
// dx_i = r * (g_c*dy_i - (s1 + xhat_i*s2) / m)   with per-group sums
//   s1 = sum(g_c * dy),  s2 = sum(g_c * dy * xhat)
// dgamma_c/dbeta_c partials go to dgb_part[n][2C] rows (plain stores);
// reduce.hip's colsum kernel folds them deterministically.
// ReLU mask recomputed as (xhat*g+b) > 0.
// Dynamic LDS: [2*cspan] dgb | [cspan] gamma | [cspan] beta |
// [TP*cspan] P0 | [TP*cspan] P1.
template <bool HASRES>
__global__ void __launch_bounds__(GN_BLOCK)
gn_bwd_kernel(const GnSegs segs, const bf16* __restrict__ dz,
              const GnSegsMut dxs, const float* __restrict__ gamma,
              const float* __restrict__ beta,
              const float* __restrict__ mean_in,
              const float* __restrict__ rstd_in,
              float* __restrict__ dgb_part,
              const bf16* __restrict__ res, bf16* __restrict__ dres,
              const int HW, const int C, const int G, const int relu,
              const int accumulate, const int chunk_oct) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const ChunkMap m = chunk_map(TC, chunk_oct);
  const int t = threadIdx.x;
  const int Cg = C / G;
  const int g0 = (m.o0 << 3) / Cg, g1 = (m.o1 << 3) / Cg;
  const int cbase = m.o0 << 3;
  const int cspan = (m.o1 - m.o0) << 3;

  __shared__ float s_s1[GN_MAXG];
  __shared__ float s_s2[GN_MAXG];
  __shared__ float s_mu[GN_MAXG];
  __shared__ float s_rs[GN_MAXG];
  extern __shared__ float s_dyn[];
  float* s_dgb = s_dyn;              // [2*cspan] merged dgamma/dbeta
  float* s_ga = s_dyn + 2 * cspan;
  float* s_be = s_ga + cspan;
  float* P0 = s_be + cspan;          // [TP*cspan]
  float* P1 = P0 + m.TP * cspan;
  for (int g = g0 + t; g < g1; g += GN_BLOCK) {
    s_mu[g] = mean_in[(long)n * G + g];
    s_rs[g] = rstd_in[(long)n * G + g];
  }
  for (int c = t; c < cspan; c += GN_BLOCK) {
    s_ga[c] = gamma[cbase + c];
    s_be[c] = beta[cbase + c];
  }
  __syncthreads();

  const bf16* db = dz + (long)n * HW * C;
  const bf16* resb = HASRES ? res + (long)n * HW * C : nullptr;
  bf16* dresb = HASRES ? dres + (long)n * HW * C : nullptr;
  const int c0 = (m.o0 + m.tc) << 3;
  int cloc, cs;
  const bf16* sb = seg_locate(segs, c0, cloc, cs);
  const bf16* xb = sb + (long)n * HW * cs + cloc;

  float ga[8], be[8], mu[8], rs[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = c0 + j, g = c / Cg;
    ga[j] = s_ga[c - cbase];
    be[j] = s_be[c - cbase];
    mu[j] = s_mu[g];
    rs[j] = s_rs[g];
  }

  float a1[8] = {0}, a2[8] = {0}, adg[8] = {0}, adb[8] = {0};
  if (m.active) {
    #pragma unroll 4
    for (int p = m.tp; p < HW; p += m.TP) {
      Bf16x8 xc = *reinterpret_cast<const Bf16x8*>(xb + (long)p * cs);
      Bf16x8 dc = *reinterpret_cast<const Bf16x8*>(db + (long)p * C + c0);
      Bf16x8 rv;
      if (HASRES)
        rv = *reinterpret_cast<const Bf16x8*>(resb + (long)p * C + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = (bf2f(xc.v[j]) - mu[j]) * rs[j];
        float dy = bf2f(dc.v[j]);
        if (relu) {
          float yv = xhat * ga[j] + be[j];
          if (HASRES) yv += bf2f(rv.v[j]);
          dy = yv > 0.f ? dy : 0.f;
        }
        a1[j] += ga[j] * dy;
        a2[j] += ga[j] * dy * xhat;
        adg[j] += dy * xhat;
        adb[j] += dy;
      }
    }
    // phase 1: group sums
    const int pb = m.tp * cspan + (m.tc << 3);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      P0[pb + j] = a1[j];
      P1[pb + j] = a2[j];
    }
  }
  __syncthreads();
  for (int g = g0 + t; g < g1; g += GN_BLOCK) {
    float v1 = 0.f, v2 = 0.f;
    const int cl0 = g * Cg - cbase;
    for (int tp = 0; tp < m.TP; ++tp)
      for (int c = cl0; c < cl0 + Cg; ++c) {
        v1 += P0[tp * cspan + c];
        v2 += P1[tp * cspan + c];
      }
    s_s1[g] = v1;
    s_s2[g] = v2;
  }
  __syncthreads();

  // phase 2: per-channel dgamma/dbeta partials (reuse P0/P1)
  if (m.active) {
    const int pb = m.tp * cspan + (m.tc << 3);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      P0[pb + j] = adg[j];
      P1[pb + j] = adb[j];
    }
  }
  __syncthreads();
  for (int c = t; c < cspan; c += GN_BLOCK) {
    float dg = 0.f, db2 = 0.f;
    for (int tp = 0; tp < m.TP; ++tp) {
      dg += P0[tp * cspan + c];
      db2 += P1[tp * cspan + c];
    }
    s_dgb[c] = dg;
    s_dgb[cspan + c] = db2;
  }
  __syncthreads();

  // publish per-(sample, channel) partials with plain stores — the
  // deterministic column reduction (reduce.hip colsum) follows
  for (int c = t; c < cspan; c += GN_BLOCK) {
    float* row = dgb_part + (long)n * 2 * C;
    row[cbase + c] = s_dgb[c];
    row[C + cbase + c] = s_dgb[cspan + c];
  }

  if (!m.active) return;

  const float inv_m = 1.0f / ((float)HW * Cg);
  bf16* dxb;
  {
    int si = 0;
    while (si + 1 < segs.nseg && c0 >= segs.start[si + 1]) ++si;
    dxb = dxs.p[si] + (long)n * HW * cs + cloc;
  }
  float k1[8], k2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int g = (c0 + j) / Cg;
    k1[j] = s_s1[g] * inv_m;
    k2[j] = s_s2[g] * inv_m;
  }
  #pragma unroll 4
  for (int p = m.tp; p < HW; p += m.TP) {
    Bf16x8 xc = *reinterpret_cast<const Bf16x8*>(xb + (long)p * cs);
    Bf16x8 dc = *reinterpret_cast<const Bf16x8*>(db + (long)p * C + c0);
    Bf16x8 rv;
    if (HASRES)
      rv = *reinterpret_cast<const Bf16x8*>(resb + (long)p * C + c0);
    Bf16x8 out;
    Bf16x8 drv;
    Bf16x8 prev;
    if (accumulate)
      prev = *reinterpret_cast<const Bf16x8*>(dxb + (long)p * cs);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xhat = (bf2f(xc.v[j]) - mu[j]) * rs[j];
      float dy = bf2f(dc.v[j]);
      if (relu) {
        float yv = xhat * ga[j] + be[j];
        if (HASRES) yv += bf2f(rv.v[j]);
        dy = yv > 0.f ? dy : 0.f;
      }
      if (HASRES) drv.v[j] = f2bf(dy);  // residual grad = masked dy
      float v = rs[j] * (ga[j] * dy - (k1[j] + xhat * k2[j]));
      if (accumulate) v += bf2f(prev.v[j]);
      out.v[j] = f2bf(v);
    }
    *reinterpret_cast<Bf16x8*>(dxb + (long)p * cs) = out;
    if (HASRES)
      *reinterpret_cast<Bf16x8*>(dresb + (long)p * C + c0) = drv;
  }
}

// ---------------------------------------------------------------- launchers
// Block target for the (N, chunks) grid.  512 measured best at the
// flagship N=512 (nchunks=1): every grid size beyond ~512 blocks pays a
// fixed ~10-25 us per 512-block round that chunk-splitting cannot buy
// back (tools/gn_probe).  Chunking still engages for small per-rank
// batches (N < 512), where it fills CUs.
static int gn_target_blocks(int bwd) {
  static int cached[2] = {-1, -1};
  if (cached[bwd] < 0) {
    const char* e = getenv(bwd ? "DLB_GN_TARGET_BWD" : "DLB_GN_TARGET");
    cached[bwd] = e ? atoi(e) : 512;
    if (cached[bwd] < 1) cached[bwd] = 1;
  }
  return cached[bwd];
}

static int gcd_i(int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; }

// Chunk size in octets: group-aligned (a chunk covers whole groups),
// <= GN_MAXCHUNK (single octet per thread), sized so N * nchunks ~ the
// block target.  Returns (chunk_oct, nchunks).
static void gn_chunking(int N, int C, int G, int bwd, int* chunk_oct,
                        int* nchunks) {
  const int TC = C >> 3;
  const int Cg = C / G;
  const int u = Cg / gcd_i(Cg, 8);       // octets per group-aligned unit
  const int units = TC / u;              // always exact (C = G*Cg, C%8==0)
  int want = gn_target_blocks(bwd) / (N > 0 ? N : 1);
  if (want < 1) want = 1;
  if (want > units) want = units;
  int cu = (units + want - 1) / want;    // units per chunk
  while (cu * u > GN_MAXCHUNK && cu > 1) --cu;
  *chunk_oct = cu * u;
  *nchunks = (TC + *chunk_oct - 1) / *chunk_oct;
}

static size_t gn_dyn_shmem(int chunk_oct, int fixed_cspans) {
  // P0/P1 hold [TP][cspan] partials; TP*cspan = (GN_BLOCK/TCe)*TCe*8
  // <= GN_BLOCK*8 = 2048 floats each — and a SMALLER last chunk has a
  // LARGER TP, so size them at the bound, not at this chunk's TP.
  const int cspan = chunk_oct * 8;
  return (size_t)(fixed_cspans * cspan + 2 * (GN_BLOCK * 8)) * sizeof(float);
}

extern "C" void dlb_gn_fwd_segs(const void* const* xs, const int* starts,
                                int nseg, void* y, const float* gamma,
                                const float* beta, float* mean, float* rstd,
                                const void* res,
                                int N, int HW, int C, int G,
                                float eps, int relu, hipStream_t stream) {
  GnSegs sg{};
  sg.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    sg.p[i] = (const bf16*)xs[i];
    sg.start[i] = starts[i];
  }
  sg.start[nseg] = starts[nseg];
  int chunk_oct, nchunks;
  gn_chunking(N, C, G, 0, &chunk_oct, &nchunks);
  if (res)
    hipLaunchKernelGGL((gn_fwd_kernel<true>), dim3(N, nchunks),
                       dim3(GN_BLOCK), gn_dyn_shmem(chunk_oct, 2), stream,
                       sg, (bf16*)y, gamma, beta, mean, rstd,
                       (const bf16*)res, HW, C, G, eps, relu, chunk_oct);
  else
    hipLaunchKernelGGL((gn_fwd_kernel<false>), dim3(N, nchunks),
                       dim3(GN_BLOCK), gn_dyn_shmem(chunk_oct, 2), stream,
                       sg, (bf16*)y, gamma, beta, mean, rstd, nullptr,
                       HW, C, G, eps, relu, chunk_oct);
}

extern "C" void dlb_gn_bwd_segs(const void* const* xs, const int* starts,
                                int nseg, const void* dz, void* const* dxs,
                                const float* gamma, const float* beta,
                                const float* mean, const float* rstd,
                                float* dgb_part,
                                const void* res, void* dres,
                                int N, int HW, int C, int G, int relu,
                                int accumulate, hipStream_t stream) {
  GnSegs sg{};
  GnSegsMut dsg{};
  sg.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    sg.p[i] = (const bf16*)xs[i];
    sg.start[i] = starts[i];
    dsg.p[i] = (bf16*)dxs[i];
  }
  sg.start[nseg] = starts[nseg];
  int chunk_oct, nchunks;
  gn_chunking(N, C, G, 1, &chunk_oct, &nchunks);
  if (res)
    hipLaunchKernelGGL((gn_bwd_kernel<true>), dim3(N, nchunks),
                       dim3(GN_BLOCK), gn_dyn_shmem(chunk_oct, 4), stream,
                       sg, (const bf16*)dz, dsg, gamma, beta, mean, rstd,
                       dgb_part, (const bf16*)res, (bf16*)dres,
                       HW, C, G, relu, accumulate, chunk_oct);
  else
    hipLaunchKernelGGL((gn_bwd_kernel<false>), dim3(N, nchunks),
                       dim3(GN_BLOCK), gn_dyn_shmem(chunk_oct, 4), stream,
                       sg, (const bf16*)dz, dsg, gamma, beta, mean, rstd,
                       dgb_part, nullptr, nullptr,
                       HW, C, G, relu, accumulate, chunk_oct);
}

extern "C" void dlb_gn_stats_segs(const void* const* xs, const int* starts,
                                  int nseg, float* mean, float* rstd, int N,
                                  int HW, int C, int G, float eps,
                                  hipStream_t stream) {
  GnSegs sg{};
  sg.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    sg.p[i] = (const bf16*)xs[i];
    sg.start[i] = starts[i];
  }
  sg.start[nseg] = starts[nseg];
  int chunk_oct, nchunks;
  gn_chunking(N, C, G, 0, &chunk_oct, &nchunks);
  hipLaunchKernelGGL(gn_stats_kernel, dim3(N, nchunks), dim3(GN_BLOCK),
                     gn_dyn_shmem(chunk_oct, 0), stream, sg, mean, rstd,
                     HW, C, G, eps, chunk_oct);
}
