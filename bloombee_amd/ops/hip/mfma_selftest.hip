// MFMA fragment-layout self-test for gfx950.
//
// Computes C = A @ B for one v_mfma_f32_16x16x32_bf16 using the fragment
// maps the attention/GEMM kernels assume:
//   A (16x32): lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B (32x16): lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C (16x16): lane l reg r holds C[row = (l>>4)*4 + r][col = l&15]
// The host test feeds ASYMMETRIC A/B (guide §3: symmetric B hides a
// row<->col swap) and compares against a CPU matmul. A mismatch here
// localizes any layout error before it hides inside a fused kernel.

#include "common.h"


__global__ void mfma_selftest_kernel(const unsigned short* __restrict__ A,
                                     const unsigned short* __restrict__ B,
                                     float* __restrict__ C) {
  const int l = threadIdx.x;
  short8 as, bs;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    as[j] = (short)A[(l & 15) * 32 + (l >> 4) * 8 + j];
    bs[j] = (short)B[((l >> 4) * 8 + j) * 16 + (l & 15)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(bf16x8, as), __builtin_bit_cast(bf16x8, bs), c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}
