// Standalone probe: where does the GroupNorm kernel's fixed ~15-30 us
// per-block-round cost come from?  Times kernel variants of increasing
// complexity at grid sizes 512..8192 on one shape.
//   k0: empty kernel, small args
//   k1: empty kernel, GnSegs-sized (1.1 KB) by-value args
//   k2: x-read sweep + y write (streaming only, gn thread mapping)
//   k3: k2 + LDS atomics + barrier + second sweep (gn_bwd shape)
//   k4: k3 + global atomicAdd publish (dgamma/dbeta analog)
// Build: hipcc -O3 --offload-arch=gfx950 probe.hip -o probe
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

typedef __hip_bfloat16 bf16;
struct Bf16x8 { bf16 v[8]; };

struct BigArgs {               // mimics GnSegs + GnSegsMut kernarg bulk
  const bf16* p[56];
  int start[57];
  bf16* q[56];
  int pad[3];
};

__global__ void __launch_bounds__(256) k0(bf16* y) {
  if (blockIdx.x == 123456) y[0] = __float2bfloat16(0.f);
}

__global__ void __launch_bounds__(256) k1(BigArgs a, bf16* y) {
  if (blockIdx.x == 123456) y[0] = a.p[0][0];
}

// gn-style mapping: block (n, chunk); thread (tc, tp) over span octets
__device__ void sweep_sum(const bf16* xb, int HW, int C, int span8,
                          int TCe, int TP, int tc, int tp, int c0,
                          float* s8) {
  for (int oct = tc; oct < span8; oct += TCe) {
    const bf16* xo = xb + c0 + (oct << 3);
    for (int p = tp; p < HW; p += TP) {
      Bf16x8 ch = *reinterpret_cast<const Bf16x8*>(xo + (long)p * C);
#pragma unroll
      for (int j = 0; j < 8; ++j) s8[j] += __bfloat162float(ch.v[j]);
    }
  }
}

__global__ void __launch_bounds__(256) k2(const bf16* x, bf16* y, int HW,
                                          int C, int chunk_oct) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const int o0 = blockIdx.y * chunk_oct;
  const int span = min(chunk_oct, TC - o0);
  const int TCe = span < 256 ? span : 256;
  const int TP = 256 / TCe;
  const int tc = threadIdx.x % TCe, tp = threadIdx.x / TCe;
  if (threadIdx.x >= TCe * TP) return;
  const bf16* xb = x + (long)n * HW * C;
  bf16* yb = y + (long)n * HW * C;
  float s8[8] = {0};
  sweep_sum(xb, HW, C, span, TCe, TP, tc, tp, o0 << 3, s8);
  // write-back sweep using s8 (keeps the value live)
  for (int oct = tc; oct < span; oct += TCe) {
    const int c0 = (o0 + oct) << 3;
    for (int p = tp; p < HW; p += TP) {
      Bf16x8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) out.v[j] = __float2bfloat16(s8[j]);
      *reinterpret_cast<Bf16x8*>(yb + (long)p * C + c0) = out;
    }
  }
}

__global__ void __launch_bounds__(256) k3(const bf16* x, bf16* y, int HW,
                                          int C, int chunk_oct, int G) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const int o0 = blockIdx.y * chunk_oct;
  const int span = min(chunk_oct, TC - o0);
  const int TCe = span < 256 ? span : 256;
  const int TP = 256 / TCe;
  const int tc = threadIdx.x % TCe, tp = threadIdx.x / TCe;
  __shared__ float s_sum[64];
  for (int g = threadIdx.x; g < G; g += 256) s_sum[g] = 0.f;
  __syncthreads();
  const bool act = threadIdx.x < TCe * TP;
  const bf16* xb = x + (long)n * HW * C;
  bf16* yb = y + (long)n * HW * C;
  float s8[8] = {0};
  const int Cg = C / G;
  if (act) {
    sweep_sum(xb, HW, C, span, TCe, TP, tc, tp, o0 << 3, s8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&s_sum[(((o0 + tc) << 3) + j) / Cg % 64], s8[j]);
  }
  __syncthreads();
  if (!act) return;
  for (int oct = tc; oct < span; oct += TCe) {
    const int c0 = (o0 + oct) << 3;
    for (int p = tp; p < HW; p += TP) {
      Bf16x8 in = *reinterpret_cast<const Bf16x8*>(xb + (long)p * C + c0);
      Bf16x8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out.v[j] = __float2bfloat16(__bfloat162float(in.v[j]) +
                                    s_sum[(c0 + j) / Cg % 64]);
      *reinterpret_cast<Bf16x8*>(yb + (long)p * C + c0) = out;
    }
  }
}

__global__ void __launch_bounds__(256) k4(const bf16* x, bf16* y,
                                          float* dgb, int HW, int C,
                                          int chunk_oct, int G) {
  const int n = blockIdx.x;
  const int TC = C >> 3;
  const int o0 = blockIdx.y * chunk_oct;
  const int span = min(chunk_oct, TC - o0);
  const int TCe = span < 256 ? span : 256;
  const int TP = 256 / TCe;
  const int tc = threadIdx.x % TCe, tp = threadIdx.x / TCe;
  __shared__ float s_sum[64];
  for (int g = threadIdx.x; g < G; g += 256) s_sum[g] = 0.f;
  __syncthreads();
  const bool act = threadIdx.x < TCe * TP;
  const bf16* xb = x + (long)n * HW * C;
  bf16* yb = y + (long)n * HW * C;
  float s8[8] = {0};
  const int Cg = C / G;
  if (act) {
    sweep_sum(xb, HW, C, span, TCe, TP, tc, tp, o0 << 3, s8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&s_sum[(((o0 + tc) << 3) + j) / Cg % 64], s8[j]);
  }
  __syncthreads();
  const int cspan = span << 3;
  for (int c = threadIdx.x; c < cspan; c += 256) {
    atomicAdd(&dgb[(o0 << 3) + c], s_sum[c / Cg % 64]);
    atomicAdd(&dgb[C + (o0 << 3) + c], 1.f);
  }
  if (!act) return;
  for (int oct = tc; oct < span; oct += TCe) {
    const int c0 = (o0 + oct) << 3;
    for (int p = tp; p < HW; p += TP) {
      Bf16x8 in = *reinterpret_cast<const Bf16x8*>(xb + (long)p * C + c0);
      Bf16x8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out.v[j] = __float2bfloat16(__bfloat162float(in.v[j]) +
                                    s_sum[(c0 + j) / Cg % 64]);
      *reinterpret_cast<Bf16x8*>(yb + (long)p * C + c0) = out;
    }
  }
}

#define CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("ERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

template <typename F>
float timeit(F f, int iters) {
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  for (int i = 0; i < 3; ++i) f();
  hipEventRecord(a);
  for (int i = 0; i < iters; ++i) f();
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms;
  hipEventElapsedTime(&ms, a, b);
  hipEventDestroy(a);
  hipEventDestroy(b);
  return ms * 1000.f / iters;  // us
}

int main(int argc, char** argv) {
  const int N = 512, HW = (argc > 1) ? atoi(argv[1]) : 64, C = 128, G = 32;
  const int iters = 50;
  bf16 *x, *y;
  float* dgb;
  CHECK(hipMalloc(&x, (long)N * HW * C * 2));
  CHECK(hipMalloc(&y, (long)N * HW * C * 2));
  CHECK(hipMalloc(&dgb, 2L * C * 4));
  CHECK(hipMemset(x, 0x3c, (long)N * HW * C * 2));
  BigArgs big{};
  big.p[0] = x;
  printf("shape N%d HW%d C%d; us per launch\n", N, HW, C);
  for (int nchunks = 1; nchunks <= 16; nchunks *= 2) {
    const int TC = C >> 3;
    if (nchunks > TC) break;
    const int chunk = (TC + nchunks - 1) / nchunks;
    dim3 grid(N, nchunks), blk(256);
    float t0 = timeit([&] { hipLaunchKernelGGL(k0, grid, blk, 0, 0, y); }, iters);
    float t1 = timeit([&] { hipLaunchKernelGGL(k1, grid, blk, 0, 0, big, y); }, iters);
    float t2 = timeit([&] { hipLaunchKernelGGL(k2, grid, blk, 0, 0, x, y, HW, C, chunk); }, iters);
    float t3 = timeit([&] { hipLaunchKernelGGL(k3, grid, blk, 0, 0, x, y, HW, C, chunk, G); }, iters);
    float t4 = timeit([&] { hipLaunchKernelGGL(k4, grid, blk, 0, 0, x, y, dgb, HW, C, chunk, G); }, iters);
    printf("chunks %2d (blocks %5d): k0 %7.1f  k1(bigarg) %7.1f  "
           "k2(stream) %7.1f  k3(+lds+bar) %7.1f  k4(+glatomic) %7.1f\n",
           nchunks, N * nchunks, t0, t1, t2, t3, t4);
  }
  return 0;
}
