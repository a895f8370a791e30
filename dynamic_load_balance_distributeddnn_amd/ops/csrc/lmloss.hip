// Fused LM loss head (gfx950): decoder GEMM -> log_softmax -> NLL,
// forward + backward, with the [T, V] logits never materialized.
//
// Reference computation (dbs.py:371-374 criterion over
// Net/Transformer.py:95 log_softmax of the 200->33278 decoder): eager
// materializes ~120 MB of bf16 logits per step at the flagship shape and
// reads them 5+ times across log_softmax fwd/bwd and NLL.  Here:
//
//   fwd:   z[t,v] = h[t,:] . W[v,:] + bias[v] streamed in [64,64] MFMA
//          tiles with an ONLINE (max, sumexp) accumulator per row;
//          partials per vocab partition -> combine kernel gives
//          lse[t], loss = mean(lse - z_target).
//   bwd:   dZ = go/T * (softmax(z) - onehot); z is RECOMPUTED tile-wise:
//          - dh kernel:  dh[t,:]  = dZ[t,:] @ W    (tiles over V)
//          - dw kernel:  dW[v,:]  = dZ[:,v]^T @ h, db[v] = sum_t dZ[t,v]
//          Three streaming GEMM passes instead of one cached-logits
//          pass; HBM traffic drops from ~6 logits-sized arrays to zero.
//
// MFMA v_mfma_f32_16x16x32_bf16 with the same fragment conventions as
// conv.hip (contiguous k-map, C/D col=lane&15 row=(lane>>4)*4+r), and
// ds_read_b64_tr_b16 transpose-read fragments for the dP^T / W^T / h^T
// operands (lane map verified by tools/mfma_probe/tr_probe.hip).
//
// Constraints: d <= 224 and d % 8 == 0 (the zoo's decoder is d=200).

#include "common.h"

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define LM_BLOCK 256
#define LM_BM 64          // rows (tokens) per block
#define LM_BN 64          // vocab columns per tile
#define LM_DPAD 224       // padded d (7 k-steps of 32)
#define LM_LDH (LM_DPAD + 8)
#define LM_LDP (LM_BN + 8)

__device__ inline bf16x8_t lm_zero8() {
  bf16x8_t z = {0, 0, 0, 0, 0, 0, 0, 0};
  return z;
}

#define LDS3 __attribute__((address_space(3)))
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 trvec;

// transpose-read fragment from a row-major [m][cols] LDS image (stride
// LROW): element j of the return = img[mbase + (lane>>4)*8 + j][colbase
// + (lane&15)] — a [n=col][k=m] MFMA operand (see conv.hip tr_frag).
template <int LROW>
__device__ inline bf16x8_t lm_tr_frag(const bf16* img, int mbase,
                                      int colbase, int lane) {
  const int j15 = lane & 15, q = lane >> 4;
  const int row = mbase + q * 8 + (j15 >> 2);
  const int col = colbase + 4 * (j15 & 3);
  auto p0 = (LDS3 trvec*)((LDS3 bf16*)img + (long)row * LROW + col);
  auto p1 = (LDS3 trvec*)((LDS3 bf16*)img + (long)(row + 4) * LROW + col);
  trvec lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
  trvec hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  union { struct { trvec a, b; } t; bf16x8_t v; } u;
  u.t.a = lo;
  u.t.b = hi;
  return u.v;
}

// Stage a [rows, d] bf16 global block into a [rows][LM_LDH] LDS image,
// zero-padding cols [d, LM_DPAD) and out-of-range rows.
__device__ inline void lm_stage(const bf16* __restrict__ src, long row0,
                                long rmax, int d, bf16* dst, int rows) {
  const int chunks_per_row = LM_DPAD / 8;
  const int total = rows * chunks_per_row;
  for (int c = threadIdx.x; c < total; c += LM_BLOCK) {
    const int r = c / chunks_per_row;
    const int k8 = (c % chunks_per_row) * 8;
    bf16x8_t v = lm_zero8();
    if (row0 + r < rmax && k8 < d)
      v = *reinterpret_cast<const bf16x8_t*>(src + (row0 + r) * d + k8);
    *reinterpret_cast<bf16x8_t*>(dst + r * LM_LDH + k8) = v;
  }
}

// Register-staged variant: the serialized stage->barrier->MFMA loop
// measured ~3 ms per dispatch (5x the traffic bound) — prefetching the
// NEXT tile's 7 bf16x8 loads per thread while the current tile's MFMAs
// run hides the L2/L3 latency (conv.hip's pipeline idiom).
// 64 rows * (LM_DPAD/8) chunks = 7 * LM_BLOCK exactly.
#define LM_SREG (LM_BM * (LM_DPAD / 8) / LM_BLOCK)
struct LmStage {
  bf16x8_t v[LM_SREG];
  float b;  // bias[n0 + tid] for tid < LM_BN (vocab-tile bias slice)
};

__device__ inline void lm_sload(const bf16* __restrict__ src, long row0,
                                long rmax, int d, LmStage& s,
                                const float* __restrict__ bias) {
  const int cpr = LM_DPAD / 8;
#pragma unroll
  for (int u = 0; u < LM_SREG; ++u) {
    const int c = threadIdx.x + u * LM_BLOCK;
    const int r = c / cpr;
    const int k8 = (c % cpr) * 8;
    s.v[u] = lm_zero8();
    if (row0 + r < rmax && k8 < d)
      s.v[u] = *reinterpret_cast<const bf16x8_t*>(src + (row0 + r) * d + k8);
  }
  if (bias) {
    s.b = 0.f;
    if (threadIdx.x < LM_BN && row0 + threadIdx.x < rmax)
      s.b = bias[row0 + threadIdx.x];
  }
}

__device__ inline void lm_swrite(const LmStage& s, bf16* dst,
                                 float* s_bias) {
  const int cpr = LM_DPAD / 8;
#pragma unroll
  for (int u = 0; u < LM_SREG; ++u) {
    const int c = threadIdx.x + u * LM_BLOCK;
    const int r = c / cpr;
    const int k8 = (c % cpr) * 8;
    *reinterpret_cast<bf16x8_t*>(dst + r * LM_LDH + k8) = s.v[u];
  }
  if (s_bias && threadIdx.x < LM_BN) s_bias[threadIdx.x] = s.b;
}

// z tile [64 t-rows x 64 v-cols] via MFMA from staged h and W images.
// Wave w computes rows [w*16, w*16+16).  acc[j] covers cols [j*16,+16).
__device__ inline void lm_ztile(const bf16* h_lds, const bf16* w_lds,
                                f32x4 acc[4], int lane, int wave) {
#pragma unroll
  for (int j = 0; j < 4; ++j) acc[j] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < LM_DPAD / 32; ++ks) {
    const int k8 = ks * 32 + (lane >> 4) * 8;
    bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
        &h_lds[(wave * 16 + (lane & 15)) * LM_LDH + k8]);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
          &w_lds[(j * 16 + (lane & 15)) * LM_LDH + k8]);
      acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[j],
                                                       0, 0, 0);
    }
  }
}

// ---------------------------------------------------------------- forward
// grid (ceil(T/64), P).  part[row*P + p] = (running max, running sumexp)
// over partition p's vocab range; ztgt[row] written by the partition
// holding target[row].
extern "C" __global__ void __launch_bounds__(LM_BLOCK)
lmloss_fwd_kernel(const bf16* __restrict__ h, const bf16* __restrict__ w,
                  const float* __restrict__ bias,
                  const int* __restrict__ tgt, float2* __restrict__ part,
                  float* __restrict__ ztgt, const int T, const int d,
                  const int V, const int P, const int tiles_per_p) {
  __shared__ bf16 h_lds[LM_BM * LM_LDH];
  __shared__ bf16 w_lds[LM_BN * LM_LDH];
  __shared__ float s_bias[LM_BN];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const long m0 = (long)blockIdx.x * LM_BM;
  const int p = blockIdx.y;
  const int v_tile0 = p * tiles_per_p;
  const int v_tile1 = min(v_tile0 + tiles_per_p, (V + LM_BN - 1) / LM_BN);

  lm_stage(h, m0, T, d, h_lds, LM_BM);

  // per-lane online accumulators for row slots r=0..3
  float m_run[4] = {-3.4e38f, -3.4e38f, -3.4e38f, -3.4e38f};
  float s_run[4] = {0.f, 0.f, 0.f, 0.f};
  // this lane's rows: wave*16 + (lane>>4)*4 + r
  const int rbase = wave * 16 + (lane >> 4) * 4;
  int my_tgt[4];
#pragma unroll
  for (int r = 0; r < 4; ++r)
    my_tgt[r] = (m0 + rbase + r < T) ? tgt[m0 + rbase + r] : -1;

  LmStage ws;
  if (v_tile0 < v_tile1)
    lm_sload(w, (long)v_tile0 * LM_BN, V, d, ws, bias);
  __syncthreads();
  if (v_tile0 < v_tile1) lm_swrite(ws, w_lds, s_bias);
  __syncthreads();

  for (int vt = v_tile0; vt < v_tile1; ++vt) {
    const int n0 = vt * LM_BN;
    if (vt + 1 < v_tile1)  // prefetch next W tile during the MFMAs
      lm_sload(w, (long)(vt + 1) * LM_BN, V, d, ws, bias);
    f32x4 acc[4];
    lm_ztile(h_lds, w_lds, acc, lane, wave);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float z[4];
      float tile_max = -3.4e38f;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int cl = j * 16 + (lane & 15);
        const int col = n0 + cl;
        z[j] = (col < V) ? acc[j][r] + s_bias[cl] : -3.4e38f;
        tile_max = fmaxf(tile_max, z[j]);
        if (col == my_tgt[r]) ztgt[m0 + rbase + r] = z[j];
      }
      const float newm = fmaxf(m_run[r], tile_max);
      if (newm > -3.4e38f) {
        float s = s_run[r] * __expf(m_run[r] - newm);
#pragma unroll
        for (int j = 0; j < 4; ++j) s += __expf(z[j] - newm);
        m_run[r] = newm;
        s_run[r] = s;
      }
    }
    __syncthreads();          // all MFMA reads of w_lds complete
    if (vt + 1 < v_tile1) lm_swrite(ws, w_lds, s_bias);
    __syncthreads();
  }

  // merge the 16 lanes of each (lane>>4) group: cols -> one (m, s)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float m = m_run[r], s = s_run[r];
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) {
      const float om = __shfl_xor(m, off, 64);
      const float os = __shfl_xor(s, off, 64);
      const float nm = fmaxf(m, om);
      s = s * __expf(m - nm) + os * __expf(om - nm);
      m = nm;
    }
    if ((lane & 15) == 0 && m0 + rbase + r < T) {
      part[(m0 + rbase + r) * P + p] = make_float2(m, s);
    }
  }
}

// combine: lse[t] = m + log s over partitions; loss += (lse - ztgt)/T
extern "C" __global__ void __launch_bounds__(LM_BLOCK)
lmloss_combine_kernel(const float2* __restrict__ part,
                      const float* __restrict__ ztgt,
                      float* __restrict__ lse, float* __restrict__ loss,
                      const int T, const int P) {
  const long row = (long)blockIdx.x * LM_BLOCK + threadIdx.x;
  float contrib = 0.f;
  if (row < T) {
    float m = -3.4e38f;
    for (int p = 0; p < P; ++p) m = fmaxf(m, part[row * P + p].x);
    float s = 0.f;
    for (int p = 0; p < P; ++p) {
      const float2 v = part[row * P + p];
      s += v.y * __expf(v.x - m);
    }
    const float l = m + __logf(s);
    lse[row] = l;
    contrib = (l - ztgt[row]) / (float)T;
  }
  contrib = wave_reduce_sum(contrib);
  __shared__ float s_sum[LM_BLOCK / 64];
  if ((threadIdx.x & 63) == 0) s_sum[threadIdx.x >> 6] = contrib;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < LM_BLOCK / 64; ++i) tot += s_sum[i];
    atomicAdd(loss, tot);
  }
}

// dP tile epilogue: write go/T * (softmax - onehot) to a [64][LM_LDP]
// bf16 LDS image from the z accumulators.  bias comes from the staged
// LDS slice and lse/tgt from per-lane registers — per-tile 4-byte
// global gathers here re-serialized every tile on waitcnt chains.
__device__ inline void lm_dp_tile(f32x4 acc[4],
                                  const float* __restrict__ s_bias,
                                  const int my_tgt[4],
                                  const float my_lse[4], long m0,
                                  int n0, int T, int V, float scale,
                                  bf16* dp_lds, int lane, int wave,
                                  float db_part[4]) {
  const int rbase = wave * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long row = m0 + rbase + r;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int cl = j * 16 + (lane & 15);
      const int col = n0 + cl;
      float v = 0.f;
      if (row < T && col < V) {
        v = __expf(acc[j][r] + s_bias[cl] - my_lse[r]);
        if (col == my_tgt[r]) v -= 1.f;
        v *= scale;
      }
      db_part[j] += v;
      dp_lds[(rbase + r) * LM_LDP + cl] = __float2bfloat16(v);
    }
  }
}

// ------------------------------------------------------------ backward dh
// grid (ceil(T/64), P): dh[t,:] += sum_{v in p} dP[t,v] W[v,:]
// (fp32 atomics into a zeroed dh; P=1 keeps them contention-free).
extern "C" __global__ void __launch_bounds__(LM_BLOCK)
lmloss_bwd_dh_kernel(const bf16* __restrict__ h, const bf16* __restrict__ w,
                     const float* __restrict__ bias,
                     const int* __restrict__ tgt,
                     const float* __restrict__ lse,
                     const float* __restrict__ go, float* __restrict__ dh,
                     const int T, const int d, const int V, const int P,
                     const int tiles_per_p) {
  __shared__ bf16 h_lds[LM_BM * LM_LDH];
  __shared__ bf16 w_lds[LM_BN * LM_LDH];
  __shared__ bf16 dp_lds[LM_BM * LM_LDP];
  __shared__ float s_bias[LM_BN];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const long m0 = (long)blockIdx.x * LM_BM;
  const int p = blockIdx.y;
  const int v_tile0 = p * tiles_per_p;
  const int v_tile1 = min(v_tile0 + tiles_per_p, (V + LM_BN - 1) / LM_BN);
  const float scale = go[0] / (float)T;

  lm_stage(h, m0, T, d, h_lds, LM_BM);

  f32x4 dacc[LM_DPAD / 16];
#pragma unroll
  for (int j = 0; j < LM_DPAD / 16; ++j) dacc[j] = {0.f, 0.f, 0.f, 0.f};
  float db_dummy[4] = {0.f, 0.f, 0.f, 0.f};
  const int rbase = wave * 16 + (lane >> 4) * 4;
  float my_lse[4];
  int my_tgt[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long row = m0 + rbase + r;
    my_lse[r] = (row < T) ? lse[row] : 0.f;
    my_tgt[r] = (row < T) ? tgt[row] : -1;
  }

  LmStage ws;
  if (v_tile0 < v_tile1)
    lm_sload(w, (long)v_tile0 * LM_BN, V, d, ws, bias);
  __syncthreads();
  if (v_tile0 < v_tile1) lm_swrite(ws, w_lds, s_bias);
  __syncthreads();

  for (int vt = v_tile0; vt < v_tile1; ++vt) {
    const int n0 = vt * LM_BN;
    if (vt + 1 < v_tile1)  // prefetch next W tile during this tile's math
      lm_sload(w, (long)(vt + 1) * LM_BN, V, d, ws, bias);
    f32x4 acc[4];
    lm_ztile(h_lds, w_lds, acc, lane, wave);
    lm_dp_tile(acc, s_bias, my_tgt, my_lse, m0, n0, T, V, scale, dp_lds,
               lane, wave, db_dummy);
    __syncthreads();
    // dh[64, d] += dP[64, 64] @ W_tile[64, d]
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
          &dp_lds[(wave * 16 + (lane & 15)) * LM_LDP + ks * 32 +
                  (lane >> 4) * 8]);
#pragma unroll
      for (int j = 0; j < LM_DPAD / 16; ++j) {
        bf16x8_t bfrag = lm_tr_frag<LM_LDH>(w_lds, ks * 32, j * 16, lane);
        dacc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                          dacc[j], 0, 0, 0);
      }
    }
    __syncthreads();          // w_lds reads complete before overwrite
    if (vt + 1 < v_tile1) lm_swrite(ws, w_lds, s_bias);
    __syncthreads();
  }

#pragma unroll
  for (int j = 0; j < LM_DPAD / 16; ++j) {
    const int col = j * 16 + (lane & 15);
    if (col >= d) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = m0 + rbase + r;
      if (row < T) atomicAdd(&dh[row * d + col], dacc[j][r]);
    }
  }
}

// ------------------------------------------------------------ backward dW
// grid (ceil(V/64)): block owns W rows [n0, n0+64): stages its W tile
// once, loops T chunks recomputing dP, accumulates
//   dW[v,:] = sum_t dP[t,v] h[t,:]   and   db[v] = sum_t dP[t,v].
// Full-T loop per block -> direct stores, no atomics.
extern "C" __global__ void __launch_bounds__(LM_BLOCK)
lmloss_bwd_dw_kernel(const bf16* __restrict__ h, const bf16* __restrict__ w,
                     const float* __restrict__ bias,
                     const int* __restrict__ tgt,
                     const float* __restrict__ lse,
                     const float* __restrict__ go, float* __restrict__ dw,
                     float* __restrict__ db, const int T, const int d,
                     const int V) {
  __shared__ bf16 h_lds[LM_BM * LM_LDH];
  __shared__ bf16 w_lds[LM_BN * LM_LDH];
  __shared__ bf16 dp_lds[LM_BM * LM_LDP];
  __shared__ float db_lds[LM_BN];
  __shared__ float s_bias[LM_BN];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int n0 = blockIdx.x * LM_BN;
  const float scale = go[0] / (float)T;

  lm_stage(w, n0, V, d, w_lds, LM_BN);
  if (t < LM_BN) {
    db_lds[t] = 0.f;
    s_bias[t] = (n0 + t < V) ? bias[n0 + t] : 0.f;
  }

  f32x4 wacc[LM_DPAD / 16];
#pragma unroll
  for (int j = 0; j < LM_DPAD / 16; ++j) wacc[j] = {0.f, 0.f, 0.f, 0.f};
  float db_part[4] = {0.f, 0.f, 0.f, 0.f};
  const int rbase = wave * 16 + (lane >> 4) * 4;

  LmStage hs;
  lm_sload(h, 0, T, d, hs, nullptr);
  __syncthreads();
  lm_swrite(hs, h_lds, nullptr);
  __syncthreads();

  for (long m0 = 0; m0 < T; m0 += LM_BM) {
    if (m0 + LM_BM < T)  // prefetch next h chunk during this chunk's math
      lm_sload(h, m0 + LM_BM, T, d, hs, nullptr);
    float my_lse[4];
    int my_tgt[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = m0 + rbase + r;
      my_lse[r] = (row < T) ? lse[row] : 0.f;
      my_tgt[r] = (row < T) ? tgt[row] : -1;
    }
    f32x4 acc[4];
    lm_ztile(h_lds, w_lds, acc, lane, wave);
    lm_dp_tile(acc, s_bias, my_tgt, my_lse, m0, n0, T, V, scale, dp_lds,
               lane, wave, db_part);
    __syncthreads();
    // dW[64 v, d] += dP^T[64 v, 64 t] @ h[64 t, d]
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t afrag = lm_tr_frag<LM_LDP>(dp_lds, ks * 32,
                                          wave * 16, lane);
#pragma unroll
      for (int j = 0; j < LM_DPAD / 16; ++j) {
        bf16x8_t bfrag = lm_tr_frag<LM_LDH>(h_lds, ks * 32, j * 16, lane);
        wacc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                          wacc[j], 0, 0, 0);
      }
    }
    __syncthreads();          // h_lds reads complete before overwrite
    if (m0 + LM_BM < T) lm_swrite(hs, h_lds, nullptr);
    __syncthreads();
  }

  // db: this lane's partials cover cols j*16+(lane&15) summed over its
  // rows; fold the 4 row-groups (lane>>4) and 4 waves via LDS.
#pragma unroll
  for (int j = 0; j < 4; ++j)
    atomicAdd(&db_lds[j * 16 + (lane & 15)], db_part[j]);

  const int vbase = wave * 16 + (lane >> 4) * 4;  // dW acc row (v-local)
#pragma unroll
  for (int j = 0; j < LM_DPAD / 16; ++j) {
    const int col = j * 16 + (lane & 15);
    if (col >= d) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int v = n0 + vbase + r;
      if (v < V) dw[(long)v * d + col] = wacc[j][r];
    }
  }
  __syncthreads();
  if (t < LM_BN && n0 + t < V) db[n0 + t] = db_lds[t];
}

// ---------------------------------------------------------------- launchers
extern "C" void dlb_lmloss_fwd(const void* h, const void* w,
                               const float* bias, const int* tgt,
                               float* part, float* ztgt, float* lse,
                               float* loss, long T, int d, int V, int P,
                               hipStream_t stream) {
  const int rb = (int)((T + LM_BM - 1) / LM_BM);
  const int vtiles = (V + LM_BN - 1) / LM_BN;
  const int tiles_per_p = (vtiles + P - 1) / P;
  hipLaunchKernelGGL(lmloss_fwd_kernel, dim3(rb, P), dim3(LM_BLOCK), 0,
                     stream, (const bf16*)h, (const bf16*)w, bias, tgt,
                     (float2*)part, ztgt, (int)T, d, V, P, tiles_per_p);
  const int cb = (int)((T + LM_BLOCK - 1) / LM_BLOCK);
  hipLaunchKernelGGL(lmloss_combine_kernel, dim3(cb), dim3(LM_BLOCK), 0,
                     stream, (const float2*)part, ztgt, lse, loss, (int)T,
                     P);
}

extern "C" void dlb_lmloss_bwd(const void* h, const void* w,
                               const float* bias, const int* tgt,
                               const float* lse, const float* go, float* dh,
                               float* dw, float* db, long T, int d, int V,
                               int P, hipStream_t stream) {
  const int rb = (int)((T + LM_BM - 1) / LM_BM);
  const int vtiles = (V + LM_BN - 1) / LM_BN;
  const int tiles_per_p = (vtiles + P - 1) / P;
  hipLaunchKernelGGL(lmloss_bwd_dh_kernel, dim3(rb, P), dim3(LM_BLOCK), 0,
                     stream, (const bf16*)h, (const bf16*)w, bias, tgt, lse,
                     go, dh, (int)T, d, V, P, tiles_per_p);
  hipLaunchKernelGGL(lmloss_bwd_dw_kernel, dim3(vtiles), dim3(LM_BLOCK), 0,
                     stream, (const bf16*)h, (const bf16*)w, bias, tgt, lse,
                     go, dw, db, (int)T, d, V);
}
