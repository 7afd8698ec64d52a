// 4-bit-weight skinny-M GEMM for gfx950: C(M,N) = A(M,K) @ dequant(Wq)^T,
// M <= 32. Decode GEMMs are pure weight streams (gemm_skinny.hip); quantizing
// W to 4-bit group-min/max codes (quant4.hip format: 2 codes/byte along K,
// fp16 scale/zero per group of 64) cuts the streamed bytes ~3.5x, which is
// the whole speedup — dequant runs as VALU work between the MFMAs.
//
// This is what makes W4 weight-compressed DECODE a compute path rather than
// a storage trick (the reference's FlexGen compression only compresses
// weights at rest, flexgen_utils/compression.py:94-210), and it makes
// same-weights self-drafting for speculative decoding pay for itself: the
// draft streams a quarter of the target's bytes.
//
// Memory pattern: lane (li, hi) loads a 4-byte code word (8 nibbles) per
// 32-k slice; the four hi lanes of one W row touch one 16 B window per
// slice and adjacent slices walk the row sequentially, so L2 turns the
// 16 B requests into full DRAM bursts (same locality argument as the bf16
// kernel's 16 B fragments).

#include "common.h"
#include <hip/hip_fp16.h>

DEVINL bf16x8 dq8_q4(unsigned int c, float sc, float zp) {
  short8 s;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    s[j] = (short)f2bf((float)((c >> (4 * j)) & 0xFu) * sc + zp);
  return as_bf16x8(s);
}

template <int MT>  // 16-row M-tiles (1: M<=16, 2: M<=32)
__global__ __launch_bounds__(256) void gemm_skinny_w4_kernel(
    const unsigned short* __restrict__ A,     // (M, K) bf16
    const unsigned char* __restrict__ Wq,     // (N, K/2) packed nibbles
    const __half* __restrict__ scale,         // (N, K/64) f16
    const __half* __restrict__ zero,          // (N, K/64) f16
    const unsigned short* __restrict__ R,     // (M, N) residual or null
    const unsigned short* __restrict__ bias,  // (N,) or null
    unsigned short* __restrict__ C,           // (M, N)       (ksplit == 1)
    float* __restrict__ Cpart,                // (ksplit, M, N) (ksplit > 1)
    int M, int N, int K, int kchunk, int ksplit) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  // XCD-aware remap (see gemm_skinny_v2): keep each XCD on one contiguous
  // W region
  int tile = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int xcd = tile % 8, orig8 = tile / 8;
    const int q = nwg / 8, r = nwg % 8;
    if (nwg >= 8)
      tile = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig8;
  }
  const int n0 = tile * 64 + wave * 16;
  if (n0 >= N) return;
  const int split = blockIdx.y;
  const int k0 = split * kchunk;
  const int k1 = min(K, k0 + kchunk);

  const unsigned char* wrow = Wq + (long)(n0 + li) * (K / 2);
  const __half* srow = scale + (long)(n0 + li) * (K / 64);
  const __half* zrow = zero + (long)(n0 + li) * (K / 64);

  int arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) arow[t] = min(t * 16 + li, M - 1);

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // two code words (2 x 32-k slices) per group of 64: load scale/zero once
  // per 64-k step. The 4-deep unroll keeps ~8 loads in flight per lane.
  for (int k = k0; k < k1; k += 128) {
#pragma unroll
    for (int g = 0; g < 2; ++g) {                 // two 64-k groups
      const int kg = k + g * 64;
      const float sc = __half2float(srow[kg / 64]);
      const float zp = __half2float(zrow[kg / 64]);
#pragma unroll
      for (int u = 0; u < 2; ++u) {               // two 32-k slices per group
        const int kk = kg + u * 32 + hi * 8;
        const unsigned int cw =
            *reinterpret_cast<const unsigned int*>(wrow + kk / 2);
        bf16x8 bfrag = dq8_q4(cw, sc, zp);
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          bf16x8 afrag = as_bf16x8(
              *reinterpret_cast<const short8*>(A + (long)arow[t] * K + kk));
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                           acc[t], 0, 0, 0);
        }
      }
    }
  }

  if (ksplit == 1) {
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = t * 16 + hi * 4 + reg;
        if (m >= M) continue;
        float v = acc[t][reg];
        if (bias) v += bf2f(bias[n0 + li]);
        if (R) v += bf2f(R[(long)m * N + n0 + li]);
        C[(long)m * N + n0 + li] = f2bf(v);
      }
  } else {
    float* dst = Cpart + (long)split * M * N;
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = t * 16 + hi * 4 + reg;
        if (m >= M) continue;
        dst[(long)m * N + n0 + li] = acc[t][reg];
      }
  }
}
