// Shared MFMA tile helpers for the f32 policy-network kernels.
//
// K-permutation for float4 fragment reads: element k of a K-vector is
// stored at kperm(k) = (k%4)*(K/4) + k/4 (bijective when K % 16 == 0).
// A lane with MFMA k-subindex kk = lane>>4 then reads its four k's for a
// 16-deep macro-step as ONE aligned float4 at [kk*(K/4) + 4*k16]. Packed
// weights (models/hip_actor.py / hip_train.py): wt[K,N] ->
// view(K/16, 4(s), 4(kk), N).permute(0,2,3,1).contiguous(), so a lane's B
// values for a macro-step are one float4 at (((k16*4+kk)*N + col)*4):
// 16 consecutive lanes read 16 consecutive float4s (coalesced 1 KiB).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64

typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ int kperm(int k, int K) {
  return (k & 3) * (K >> 2) + (k >> 2);
}

// bf16 fragment scheme: both operands share the CONTIGUOUS per-lane k-map
// k = kk*8 + j (any bijective shared map yields the correct dot product),
// so a lane reads ONE bf16x8 (16 B) per 32-deep K step. Packed weights:
// wt[K,N].view(K/32, 4(kk), 8(j), N).permute(0,1,3,2).contiguous().
typedef __bf16 rls_bf16;
typedef rls_bf16 bf16x8 __attribute__((ext_vector_type(8)));

// C[16 x 64] for one wave on mfma_f32_16x16x32_bf16: A is a PLAIN
// row-major bf16 LDS tile (contiguous k; lda in elements, keep
// (lda/2) % 64 != 0 for bank spread), B packed global bf16.
__device__ __forceinline__ void mfma_bf16_core_16x64(
    const rls_bf16* __restrict__ a_lds, int lda, int K,
    const rls_bf16* __restrict__ b_pk, int ldb, int col0, int lane,
    f32x4 acc[4]) {
  const int ar = lane & 15;
  const int kk = lane >> 4;
  const int bc = lane & 15;
  const rls_bf16* arow = a_lds + ar * lda + kk * 8;
  const rls_bf16* bbase = b_pk + (int64_t)8 * (kk * ldb + col0 + bc);
  for (int k32 = 0; k32 < (K >> 5); ++k32) {
    bf16x8 av = *(const bf16x8*)(arow + 32 * k32);
    const rls_bf16* bk = bbase + (int64_t)32 * k32 * ldb;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      bf16x8 bv = *(const bf16x8*)(bk + 128 * n);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n],
                                                       0, 0, 0);
    }
  }
}

// Core: C[16 x 64] += A[16 x K] * B[K x 64] for one wave.
// A: LDS tile in kperm layout (row stride lda, floats). B: packed global
// (logical row length ldb). acc: 4 n-subtiles of f32x4.
// C/D map (16x16x4 f32): col = lane&15, row = (lane>>4)*4 + i.
__device__ __forceinline__ void mfma_core_16x64(
    const float* __restrict__ a_lds, int lda, int K,
    const float* __restrict__ b_pk, int ldb, int col0, int lane,
    f32x4 acc[4]) {
  const int ar = lane & 15;
  const int kk = lane >> 4;
  const int bc = lane & 15;
  const float* arow = a_lds + ar * lda + kk * (K >> 2);
  const float* bbase = b_pk + (int64_t)4 * (kk * ldb + col0 + bc);
  const int nk16 = K >> 4;
  for (int k16 = 0; k16 < nk16; ++k16) {
    f32x4 av = *(const f32x4*)(arow + 4 * k16);
    const float* bk = bbase + (int64_t)16 * k16 * ldb;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      f32x4 bv = *(const f32x4*)(bk + 64 * n);
      #pragma unroll
      for (int s = 0; s < 4; ++s)
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[s], bv[s], acc[n],
                                                      0, 0, 0);
    }
  }
}
