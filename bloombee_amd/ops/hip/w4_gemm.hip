// 4-bit-weight skinny-M GEMM for gfx950: C(M,N) = A(M,K) @ dequant(Wq)^T,
// M <= 32. Decode GEMMs are pure weight streams (gemm_skinny.hip); quantizing
// W to 4-bit group-min/max codes (quant4.hip format: 2 codes/byte along K,
// fp16 scale/zero per group of 64) cuts the streamed bytes ~3.5x, which is
// the whole speedup — so the in-kernel dequant must be nearly free or the
// kernel turns VALU-bound (the first, scalar-f2bf version measured 0.6 TB/s
// effective = SLOWER than bf16).
//
// Fast dequant (the classic w4 fp16 bit-trick):
//   half(0x6400 | q) == 1024 + q        for q in 0..15
// so two nibbles OR'd into a half2 give {1024+lo, 1024+hi} with 3 bit ops,
// and one v_pk_fma_f16 applies w = q*sc + (zp - 1024*sc). A rides as fp16
// (the host converts the tiny activation matrix) and the MFMA is the f16
// variant — no per-element float conversions anywhere. ~5 VALU per pair
// instead of ~20.
//
// This puts FlexGen-style weight compression ON the compute path (the
// reference compresses weights only at rest, flexgen_utils/compression.py:
// 94-210) and makes same-weights self-drafting for speculative decoding
// pay for itself: the draft streams a quarter of the target's bytes.

#include "common.h"
#include <hip/hip_fp16.h>

typedef __attribute__((ext_vector_type(8))) _Float16 half8;
typedef __attribute__((ext_vector_type(2))) _Float16 half2v;

DEVINL half8 dq8_q4_f16(unsigned int c, half2v sc2, half2v zp2) {
  half8 out;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const unsigned byte = (c >> (8 * p)) & 0xFFu;
    const unsigned h2 = 0x64006400u | (byte & 0xFu) | ((byte & 0xF0u) << 12);
    half2v v = __builtin_bit_cast(half2v, h2);
    v = v * sc2 + zp2;  // v_pk_fma_f16
    out[2 * p] = v[0];
    out[2 * p + 1] = v[1];
  }
  return out;
}

template <int MT>  // 16-row M-tiles (1: M<=16, 2: M<=32)
__global__ __launch_bounds__(256) void gemm_skinny_w4_kernel(
    const _Float16* __restrict__ A,           // (M, K) fp16
    const unsigned char* __restrict__ Wq,     // (N, K/2) packed nibbles
    const __half* __restrict__ scale,         // (N, K/64) f16
    const __half* __restrict__ zero,          // (N, K/64) f16
    const unsigned short* __restrict__ R,     // (M, N) bf16 residual or null
    const unsigned short* __restrict__ bias,  // (N,) bf16 or null
    unsigned short* __restrict__ C,           // (M, N) bf16   (ksplit == 1)
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

  // two code words (2 x 32-k slices) per group of 64: scale/zero loaded
  // once per group; the 128-k step keeps ~4 code loads in flight per lane
  for (int k = k0; k < k1; k += 128) {
#pragma unroll
    for (int g = 0; g < 2; ++g) {                 // two 64-k groups
      const int kg = k + g * 64;
      const float scf = __half2float(srow[kg / 64]);
      const float zpf = __half2float(zrow[kg / 64]) - 1024.f * scf;
      const _Float16 sch = (_Float16)scf;
      const _Float16 zph = (_Float16)zpf;
      const half2v sc2 = {sch, sch};
      const half2v zp2 = {zph, zph};
#pragma unroll
      for (int u = 0; u < 2; ++u) {               // two 32-k slices per group
        const int kk = kg + u * 32 + hi * 8;
        const unsigned int cw =
            *reinterpret_cast<const unsigned int*>(wrow + kk / 2);
        half8 bfrag = dq8_q4_f16(cw, sc2, zp2);
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          half8 afrag = *reinterpret_cast<const half8*>(
              A + (long)arow[t] * K + kk);
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_f16(afrag, bfrag,
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
