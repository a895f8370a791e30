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

// Wave-level reduction over all 64 lanes (sum).
template <typename T>
__device__ inline T wave_reduce_sum(T v) {
#pragma unroll
  for (int off = DLB_WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, DLB_WAVE);
  return v;
}
