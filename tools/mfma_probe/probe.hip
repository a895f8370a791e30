// Empirical probe of the gfx950 v_mfma_f32_16x16x32_bf16 A/B fragment
// lane->element mapping.  Two candidates are computed; the host checks
// which reproduces the reference GEMM on asymmetric inputs.
//   CAND0: lane l, elem j -> k = (l/16)*8 + j          (contiguous 8)
//   CAND1: lane l, elem j -> k = (l/16)*4 + j%4 + 16*(j/4)  (two x16 halves)
// C/D map (documented): col = l&15, row = (l>>4)*4 + reg.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void probe(const bf16* A /*16x32 row-major*/,
                      const bf16* B /*32x16 row-major*/, float* D0, float* D1) {
  int l = threadIdx.x;
  int i = l % 16;     // A row / B col / D col
  int kg = l / 16;

  bf16x8_t a0, b0, a1, b1;
  for (int j = 0; j < 8; ++j) {
    int k0 = kg * 8 + j;
    int k1 = kg * 4 + (j % 4) + 16 * (j / 4);
    ((bf16*)&a0)[j] = A[i * 32 + k0];
    ((bf16*)&b0)[j] = B[k0 * 16 + i];
    ((bf16*)&a1)[j] = A[i * 32 + k1];
    ((bf16*)&b1)[j] = B[k1 * 16 + i];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  f32x4 d0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, c, 0, 0, 0);
  f32x4 d1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r) {
    int row = (l / 16) * 4 + r, col = l % 16;
    D0[row * 16 + col] = d0[r];
    D1[row * 16 + col] = d1[r];
  }
}

int main() {
  bf16 *A, *B;
  float *D0, *D1;
  hipMallocManaged(&A, 16 * 32 * sizeof(bf16));
  hipMallocManaged(&B, 32 * 16 * sizeof(bf16));
  hipMallocManaged(&D0, 256 * sizeof(float));
  hipMallocManaged(&D1, 256 * sizeof(float));
  float Af[16 * 32], Bf[32 * 16];
  for (int i = 0; i < 16 * 32; ++i) {
    Af[i] = (float)((i * 37 % 23) - 11) / 4.0f;  // asymmetric
    A[i] = __float2bfloat16(Af[i]);
    Af[i] = __bfloat162float(A[i]);
  }
  for (int i = 0; i < 32 * 16; ++i) {
    Bf[i] = (float)((i * 53 % 29) - 13) / 8.0f;
    B[i] = __float2bfloat16(Bf[i]);
    Bf[i] = __bfloat162float(B[i]);
  }
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, A, B, D0, D1);
  hipDeviceSynchronize();
  double e0 = 0, e1 = 0;
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      float ref = 0;
      for (int k = 0; k < 32; ++k) ref += Af[i * 32 + k] * Bf[k * 16 + j];
      e0 += fabs(D0[i * 16 + j] - ref);
      e1 += fabs(D1[i * 16 + j] - ref);
    }
  printf("CAND0(contig8) err=%g  CAND1(split4+16) err=%g\n", e0, e1);
  printf("winner: %s\n", e0 < e1 ? "CAND0" : "CAND1");
  return 0;
}
