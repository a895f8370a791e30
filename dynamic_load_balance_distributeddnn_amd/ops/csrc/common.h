// Common helpers for the gfx950 (MI355X / CDNA4) kernel set.
// Built exclusively for --offload-arch=gfx950; wave size is 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DLB_WAVE 64

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      throw std::runtime_error(std::string("HIP error: ") +               \
                               hipGetErrorString(_e) + " at " __FILE__);  \
    }                                                                     \
  } while (0)

// ceil-div
static inline __host__ __device__ int cdiv(int a, int b) { return (a + b - 1) / b; }

// Branchless division by a runtime constant (libdivide-style magic
// multiply; valid for dividends < 2^31, which covers every index here).
// Host fills via init(); device divides with one 64-bit mul + shift.
struct FastDiv {
  unsigned long long mul;
  unsigned shift;
  unsigned d;
  __host__ void init(unsigned d_) {
    d = d_;
    unsigned L = 0;
    while ((1ull << L) < d) ++L;
    shift = 32 + L;
    mul = ((1ull << shift) + d - 1) / d;  // ceil(2^(32+L)/d)
  }
  __device__ inline unsigned div(unsigned m) const {
    return (unsigned)((m * mul) >> shift);
  }
  __device__ inline void divmod(unsigned m, unsigned& q, unsigned& r) const {
    q = div(m);
    r = m - q * d;
  }
};

// Wave-level reduction over all 64 lanes (sum).
template <typename T>
__device__ inline T wave_reduce_sum(T v) {
#pragma unroll
  for (int off = DLB_WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, DLB_WAVE);
  return v;
}
