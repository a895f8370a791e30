// Probe __builtin_amdgcn_ds_read_tr16_b64_v4bf16 lane semantics on
// gfx950: fill LDS with a ramp, read with several per-lane address
// patterns, dump each lane's 4 returned elements (as raw indices).
#include <hip/hip_runtime.h>
#include <cstdio>

#define LDS3 __attribute__((address_space(3)))
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 trvec;

__global__ void probe(unsigned short* out, int mode) {
  __shared__ unsigned short lds[1024];
  const int l = threadIdx.x;
  for (int i = l; i < 1024; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  unsigned byte_off;
  switch (mode) {
    case 0: byte_off = 0; break;                       // uniform
    case 1: byte_off = (l & 15) * 2; break;            // lane-in-group x2B
    case 2: byte_off = l * 8; break;                   // lane-linear 8B
    default: byte_off = (l & 15) * 2 + (l >> 4) * 128; // group-strided
  }
  auto p3 = (LDS3 trvec*)((LDS3 char*)lds + byte_off);
  trvec r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3);
  union { trvec v; unsigned short u[4]; } u;
  u.v = r;
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = u.u[j];
}

int main() {
  unsigned short* out;
  hipMallocManaged(&out, 64 * 4 * sizeof(unsigned short));
  for (int mode = 0; mode < 4; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, mode);
    hipDeviceSynchronize();
    printf("mode %d:\n", mode);
    for (int g = 0; g < 4; ++g)
      for (int i = 0; i < 4; ++i) {
        int l = g * 16 + i;
        printf("  l%02d: %4d %4d %4d %4d\n", l, out[l * 4], out[l * 4 + 1],
               out[l * 4 + 2], out[l * 4 + 3]);
      }
  }
  return 0;
}
