// Deterministic column reductions for split-K weight-grad slabs and the
// GroupNorm dgamma/dbeta partials (gfx950).
//
//   out[c] (+)= sum_{s<S} part[s*len + c]
//
// Profiled history: torch::sum_out ran these shapes at ~11 us/call
// (generic-reducer config overhead, 4.4% of the DenseNet step); a flat
// one-thread-per-column kernel was WORSE (~42 us: ceil(len/256) blocks
// is 32 blocks at typical wrw sizes, with a serial split loop —
// latency-bound).  This version parallelizes both dimensions inside a
// block: 256 threads as 32 columns x 8 row-groups, each thread sums a
// fixed row range with a 4-deep unrolled (ILP) loop, then an LDS tree
// folds the 8 row-group partials in a FIXED order — bitwise
// deterministic across runs, unlike fp32 atomics.
#include "common.h"

#define CS_COLS 32
#define CS_ROWG 8

// Csplit >= 0 routes columns: c < Csplit -> outA[c], else outB[c-Csplit]
// (the dgamma/dbeta pair); Csplit < 0 -> everything to outA.
// add != 0 accumulates into pre-zeroed outputs, else overwrites.
extern "C" __global__ void __launch_bounds__(CS_COLS* CS_ROWG)
colsum_kernel(const float* __restrict__ part, const int S, const long len,
              float* __restrict__ outA, float* __restrict__ outB,
              const long Csplit, const int add) {
  const int tc = threadIdx.x % CS_COLS;
  const int tr = threadIdx.x / CS_COLS;
  const long c = (long)blockIdx.x * CS_COLS + tc;
  __shared__ float fold[CS_ROWG][CS_COLS];

  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  if (c < len) {
    const int gs = (S + CS_ROWG - 1) / CS_ROWG;
    const int s0 = tr * gs;
    const int s1 = min(S, s0 + gs);
    const float* p = part + c;
    int s = s0;
    for (; s + 4 <= s1; s += 4) {
      a0 += p[(long)s * len];
      a1 += p[(long)(s + 1) * len];
      a2 += p[(long)(s + 2) * len];
      a3 += p[(long)(s + 3) * len];
    }
    for (; s < s1; ++s) a0 += p[(long)s * len];
  }
  fold[tr][tc] = (a0 + a1) + (a2 + a3);
  __syncthreads();
#pragma unroll
  for (int off = CS_ROWG / 2; off > 0; off >>= 1) {
    if (tr < off) fold[tr][tc] += fold[tr + off][tc];
    __syncthreads();
  }
  if (tr == 0 && c < len) {
    float v = fold[0][tc];
    float* dst = (Csplit >= 0 && c >= Csplit) ? outB + (c - Csplit)
                                              : outA + c;
    if (add)
      *dst += v;
    else
      *dst = v;
  }
}

// Split-K slab reduce: out[0:len] = sum over slabs (overwrite).
extern "C" void dlb_slab_sum(const float* part, float* out, int splits,
                             long len, hipStream_t stream) {
  const unsigned grid = (unsigned)((len + CS_COLS - 1) / CS_COLS);
  hipLaunchKernelGGL(colsum_kernel, dim3(grid), dim3(CS_COLS * CS_ROWG), 0,
                     stream, part, splits, len, out, nullptr, -1L, 0);
}

// GroupNorm dgamma/dbeta: part is [N, 2C] per-sample partial rows;
// accumulates into the (pre-zeroed / accumulating) dgamma and dbeta.
extern "C" void dlb_gn_dgb_reduce(const float* part, int N, int C,
                                  float* dgamma, float* dbeta,
                                  hipStream_t stream) {
  const long len2 = 2L * C;
  const unsigned grid = (unsigned)((len2 + CS_COLS - 1) / CS_COLS);
  hipLaunchKernelGGL(colsum_kernel, dim3(grid), dim3(CS_COLS * CS_ROWG), 0,
                     stream, part, N, len2, dgamma, dbeta, (long)C, 1);
}

// --------------------- batched multi-layer variant ----------------------
// A dense-block backward produces up to ~50 per-layer dgamma/dbeta
// partial buffers; reducing each with its own launch costs ~4-6 us of
// latency+boundary apiece.  This folds a whole block's reductions into
// ONE launch: a descriptor table maps block ranges to layers.
#define CSM_MAX 52
struct CsmDesc {
  const float* part[CSM_MAX];
  float* dg[CSM_MAX];
  float* db[CSM_MAX];
  int C[CSM_MAX];
  int blk0[CSM_MAX + 1];
  int nl;
  int N;
};

extern "C" __global__ void __launch_bounds__(CS_COLS* CS_ROWG)
colsum_multi_kernel(const CsmDesc d) {
  int li = 0;
  while (li + 1 < d.nl && (int)blockIdx.x >= d.blk0[li + 1]) ++li;
  const int lb = blockIdx.x - d.blk0[li];
  const long Cl = d.C[li];
  const long len2 = 2 * Cl;
  const int tc = threadIdx.x % CS_COLS;
  const int tr = threadIdx.x / CS_COLS;
  const long c = (long)lb * CS_COLS + tc;
  __shared__ float fold[CS_ROWG][CS_COLS];

  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  if (c < len2) {
    const int gs = (d.N + CS_ROWG - 1) / CS_ROWG;
    const int s0 = tr * gs;
    const int s1 = min(d.N, s0 + gs);
    const float* p = d.part[li] + c;
    int s = s0;
    for (; s + 4 <= s1; s += 4) {
      a0 += p[(long)s * len2];
      a1 += p[(long)(s + 1) * len2];
      a2 += p[(long)(s + 2) * len2];
      a3 += p[(long)(s + 3) * len2];
    }
    for (; s < s1; ++s) a0 += p[(long)s * len2];
  }
  fold[tr][tc] = (a0 + a1) + (a2 + a3);
  __syncthreads();
#pragma unroll
  for (int off = CS_ROWG / 2; off > 0; off >>= 1) {
    if (tr < off) fold[tr][tc] += fold[tr + off][tc];
    __syncthreads();
  }
  if (tr == 0 && c < len2) {
    float v = fold[0][tc];
    if (c < Cl)
      d.dg[li][c] += v;
    else
      d.db[li][c - Cl] += v;
  }
}

extern "C" void dlb_gn_dgb_reduce_multi(const void* const* parts,
                                        void* const* dgs, void* const* dbs,
                                        const int* Cs, int nl, int N,
                                        hipStream_t stream) {
  CsmDesc d{};
  d.nl = nl;
  d.N = N;
  int b = 0;
  for (int i = 0; i < nl; ++i) {
    d.part[i] = (const float*)parts[i];
    d.dg[i] = (float*)dgs[i];
    d.db[i] = (float*)dbs[i];
    d.C[i] = Cs[i];
    d.blk0[i] = b;
    b += (int)((2L * Cs[i] + CS_COLS - 1) / CS_COLS);
  }
  d.blk0[nl] = b;
  hipLaunchKernelGGL(colsum_multi_kernel, dim3(b), dim3(CS_COLS * CS_ROWG),
                     0, stream, d);
}

// ----------------- incremental GroupNorm statistics --------------------
// DenseNet's norm1 statistics cover the whole virtual-concat stream;
// recomputing them per layer re-reads O(L^2) activation data
// (gn_stats_kernel measured 1.26 ms/step on the flagship).  Instead:
// per fresh segment, reduce per-channel (sum, sum-of-squares) ONCE
// (chansum), and derive any later layer's per-(sample, group) mean/rstd
// from the accumulated per-channel sums — group boundaries move as C
// grows, so the channel sums (not group sums) are the reusable unit.

typedef __hip_bfloat16 csbf16;

// x [N, HW, Cs] bf16 -> sum/ssq [N, Cs] fp32.  One thread per (n, c),
// fixed-order serial over HW (deterministic); adjacent-c threads stay
// coalesced at the channel stride.
extern "C" __global__ void __launch_bounds__(256)
chansum_kernel(const csbf16* __restrict__ x, float* __restrict__ sum,
               float* __restrict__ ssq, const int N, const int HW,
               const int Cs) {
  const long e = (long)blockIdx.x * 256 + threadIdx.x;
  if (e >= (long)N * Cs) return;
  const long n = e / Cs;
  const int c = (int)(e - n * Cs);
  const csbf16* xb = x + n * (long)HW * Cs + c;
  float a0 = 0.f, q0 = 0.f;
  #pragma unroll 4
  for (int p = 0; p < HW; ++p) {
    const float v = __bfloat162float(xb[(long)p * Cs]);
    a0 += v;
    q0 += v * v;
  }
  sum[e] = a0;
  ssq[e] = q0;
}

extern "C" void dlb_chansum(const void* x, float* sum, float* ssq, int N,
                            int HW, int Cs, hipStream_t stream) {
  const long grid = ((long)N * Cs + 255) / 256;
  hipLaunchKernelGGL(chansum_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const csbf16*)x, sum, ssq, N, HW, Cs);
}

// Per-channel sum segments (newest-first, like the activation segments)
// -> per-(sample, group) mean/rstd.  One thread per (n, g), walking the
// group's channels through the segment table (<= 32 channels).
#define CSS_MAXSEG 56
struct CsSegs {
  const float* sum[CSS_MAXSEG];
  const float* ssq[CSS_MAXSEG];
  int start[CSS_MAXSEG + 1];
  int nseg;
};

extern "C" __global__ void __launch_bounds__(256)
gn_stats_sums_kernel(const CsSegs segs, float* __restrict__ mean,
                     float* __restrict__ rstd, const int N, const int HW,
                     const int C, const int G, const float eps) {
  const long e = (long)blockIdx.x * 256 + threadIdx.x;
  if (e >= (long)N * G) return;
  const long n = e / G;
  const int g = (int)(e - n * G);
  const int Cg = C / G;
  float su = 0.f, sq = 0.f;
  int si = 0;
  for (int c = g * Cg; c < (g + 1) * Cg; ++c) {
    while (si + 1 < segs.nseg && c >= segs.start[si + 1]) ++si;
    const int cs = segs.start[si + 1] - segs.start[si];
    const long off = n * (long)cs + (c - segs.start[si]);
    su += segs.sum[si][off];
    sq += segs.ssq[si][off];
  }
  const float inv_m = 1.0f / ((float)HW * Cg);
  const float mu = su * inv_m;
  const float var = sq * inv_m - mu * mu;
  mean[e] = mu;
  rstd[e] = rsqrtf(var + eps);
}

extern "C" void dlb_gn_stats_sums(const void* const* sums,
                                  const void* const* ssqs,
                                  const int* starts, int nseg, float* mean,
                                  float* rstd, int N, int HW, int C, int G,
                                  float eps, hipStream_t stream) {
  CsSegs sg{};
  sg.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    sg.sum[i] = (const float*)sums[i];
    sg.ssq[i] = (const float*)ssqs[i];
    sg.start[i] = starts[i];
  }
  sg.start[nseg] = starts[nseg];
  const long grid = ((long)N * G + 255) / 256;
  hipLaunchKernelGGL(gn_stats_sums_kernel, dim3((unsigned)grid), dim3(256),
                     0, stream, sg, mean, rstd, N, HW, C, G, eps);
}
