// Elementwise / lookup kernels completing the SURVEY K-set (gfx950):
//   K11  embedding lookup x sqrt(d) fwd/bwd  (Net/Transformer.py:91)
//   K15  dropout with philox mask, mask recomputed in backward
//   K10  SE tail: sigmoid(gate) broadcast-multiply fwd/bwd
//        (Net/RegNet.py:21-22)
// All bf16 streaming; fp32 accumulation where grads add.

#include "common.h"

typedef __hip_bfloat16 bf16;
struct Bf16x8e { bf16 v[8]; };

__device__ inline unsigned ew_philox(unsigned long long seed, unsigned idx) {
  unsigned c0 = idx, c1 = (unsigned)(seed >> 32);
  unsigned key = (unsigned)seed;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    const unsigned long long prod = 0xD256D193ull * c0;
    c0 = (unsigned)(prod >> 32) ^ c1 ^ key;
    c1 = (unsigned)prod;
    key += 0x9E3779B9u;
  }
  return c0;
}

// ------------------------------- K15 dropout ---------------------------
extern "C" __global__ void __launch_bounds__(256)
dropout_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
               const long n, const float pd, const unsigned long long seed) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= n) return;
  const unsigned thresh = (unsigned)(pd * 4294967296.0);
  const float scale = 1.f / (1.f - pd);
  const float keep =
      ew_philox(seed, (unsigned)i) >= thresh ? scale : 0.f;
  y[i] = __float2bfloat16(__bfloat162float(x[i]) * keep);
}

extern "C" void dlb_dropout(const void* x, void* y, long n, float pd,
                            unsigned long long seed, hipStream_t stream) {
  const long grid = (n + 255) / 256;
  hipLaunchKernelGGL(dropout_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const bf16*)x, (bf16*)y, n, pd, seed);
}

// ---------------------------- K11 embedding ----------------------------
// fwd: out[t, :] = table[idx[t], :] * scale   ([T] rows, d columns)
extern "C" __global__ void __launch_bounds__(256)
embed_fwd_kernel(const bf16* __restrict__ table, const int* __restrict__ idx,
                 bf16* __restrict__ out, const long T, const int d,
                 const float scale) {
  const long e = ((long)blockIdx.x * 256 + threadIdx.x) * 8;
  if (e >= T * (long)d) return;
  const long t = e / d;
  const int c = (int)(e - t * d);
  const Bf16x8e v = *reinterpret_cast<const Bf16x8e*>(
      table + (long)idx[t] * d + c);
  Bf16x8e o;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    o.v[j] = __float2bfloat16(__bfloat162float(v.v[j]) * scale);
  *reinterpret_cast<Bf16x8e*>(out + e) = o;
}

// bwd: dtable[idx[t], :] += dy[t, :] * scale — fp32 atomics (repeated
// tokens make a deterministic scatter a sort problem; torch's own
// embedding backward is atomic too)
extern "C" __global__ void __launch_bounds__(256)
embed_bwd_kernel(const bf16* __restrict__ dy, const int* __restrict__ idx,
                 float* __restrict__ dtable, const long T, const int d,
                 const float scale) {
  const long e = (long)blockIdx.x * 256 + threadIdx.x;
  if (e >= T * (long)d) return;
  const long t = e / d;
  const int c = (int)(e - t * d);
  atomicAdd(&dtable[(long)idx[t] * d + c],
            __bfloat162float(dy[e]) * scale);
}

extern "C" void dlb_embed_fwd(const void* table, const int* idx, void* out,
                              long T, int d, float scale,
                              hipStream_t stream) {
  const long grid = (T * (long)d / 8 + 255) / 256;
  hipLaunchKernelGGL(embed_fwd_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const bf16*)table, idx, (bf16*)out, T, d,
                     scale);
}

extern "C" void dlb_embed_bwd(const void* dy, const int* idx, float* dtable,
                              long T, int d, float scale,
                              hipStream_t stream) {
  const long grid = (T * (long)d + 255) / 256;
  hipLaunchKernelGGL(embed_bwd_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const bf16*)dy, idx, dtable, T, d, scale);
}

// ------------------------------ K10 SE tail ----------------------------
// y[n,p,c] = x[n,p,c] * sigmoid(g[n,c])    (NHWC; gate is [N, C])
extern "C" __global__ void __launch_bounds__(256)
se_fwd_kernel(const bf16* __restrict__ x, const bf16* __restrict__ g,
              bf16* __restrict__ y, const long NHW, const int HW,
              const int C) {
  const long e = ((long)blockIdx.x * 256 + threadIdx.x) * 8;
  if (e >= NHW * (long)C) return;
  const long p = e / C;          // (n, pix) row
  const int c = (int)(e - p * C);
  const long n = p / HW;
  const Bf16x8e xv = *reinterpret_cast<const Bf16x8e*>(x + e);
  const Bf16x8e gv = *reinterpret_cast<const Bf16x8e*>(g + n * C + c);
  Bf16x8e o;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float s = 1.f / (1.f + __expf(-__bfloat162float(gv.v[j])));
    o.v[j] = __float2bfloat16(__bfloat162float(xv.v[j]) * s);
  }
  *reinterpret_cast<Bf16x8e*>(y + e) = o;
}

extern "C" void dlb_se_fwd(const void* x, const void* g, void* y, long NHW,
                           int HW, int C, hipStream_t stream) {
  const long grid = (NHW * (long)C / 8 + 255) / 256;
  hipLaunchKernelGGL(se_fwd_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const bf16*)x, (const bf16*)g, (bf16*)y, NHW,
                     HW, C);
}

// dx = dy * sig(g);  dg[n,c] = sum_p dy*x * sig*(1-sig).
// One thread per (n, c), serial over the HW pixels: dy/x/dx accesses at
// channel stride stay coalesced across adjacent-c threads and the gate
// grad is a plain per-thread store (deterministic, no atomics).
extern "C" __global__ void __launch_bounds__(256)
se_bwd_kernel(const bf16* __restrict__ x, const bf16* __restrict__ g,
              const bf16* __restrict__ dy, bf16* __restrict__ dx,
              float* __restrict__ dg, const int N, const int HW,
              const int C) {
  const long e = (long)blockIdx.x * 256 + threadIdx.x;
  if (e >= (long)N * C) return;
  const long n = e / C;
  const int c = (int)(e - n * C);
  const float s = 1.f / (1.f + __expf(-__bfloat162float(g[n * C + c])));
  const bf16* xb = x + n * (long)HW * C + c;
  const bf16* db = dy + n * (long)HW * C + c;
  bf16* ob = dx + n * (long)HW * C + c;
  float acc = 0.f;
  #pragma unroll 4
  for (int p = 0; p < HW; ++p) {
    const float d = __bfloat162float(db[(long)p * C]);
    acc += d * __bfloat162float(xb[(long)p * C]);
    ob[(long)p * C] = __float2bfloat16(d * s);
  }
  dg[e] = acc * s * (1.f - s);
}

extern "C" void dlb_se_bwd(const void* x, const void* g, const void* dy,
                           void* dx, float* dg, int N, int HW, int C,
                           hipStream_t stream) {
  const long grid = ((long)N * C + 255) / 256;
  hipLaunchKernelGGL(se_bwd_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, (const bf16*)x, (const bf16*)g,
                     (const bf16*)dy, (bf16*)dx, dg, N, HW, C);
}
