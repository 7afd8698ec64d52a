// 4-bit-weight skinny-M GEMM for gfx950: C(M,N) = A(M,K) @ dequant(Wq)^T,
// M <= 32. Decode GEMMs are pure weight streams (gemm_skinny.hip); quantizing
// W to 4-bit group-min/max codes (quant4.hip format: 2 codes/byte along K,
// fp16 scale/zero per group of 64) cuts the streamed bytes ~3.5x.
//
// Probe-driven design (benchmarks/w4_probe.hip, profiles/):
//  * dequant must be packed-f16: half(0x6400|q)==1024+q builds a half2 from
//    two nibbles in 3 bit ops; subtract the magic FIRST (exact in f16),
//    then one v_pk_fma applies scale/zero. Scalar f32 dequant measured
//    0.6 TB/s effective (VALU-bound, slower than streaming bf16).
//  * M==1 (the speculative-draft shape) runs a GEMV: a wave reads ONE row's
//    codes as contiguous 1 KB loads (the pure-stream access pattern) and
//    accumulates with v_dot2_f32_f16 — measured 2.0x over the bf16 kernel
//    (20 us vs 40 us on the 28672x4096 gate_up shape).
//  * M 2..32 runs the MFMA form: 16 B code words distributed to fragment
//    lanes by predicated shfl, next iteration's codes prefetched, and the
//    host splits K so the grid reaches ~1800 blocks — at the bf16 kernel's
//    448-block grid the w4 stream is latency-bound at 1.2 TB/s; ksplit=4
//    measured 2.25 TB/s (1.35x bf16).
//
// This puts FlexGen-style weight compression ON the compute path (the
// reference compresses weights only at rest, flexgen_utils/compression.py:
// 94-210) and makes same-weights self-drafting for speculative decoding
// pay for itself: the draft streams a quarter of the target's bytes.

#include "common.h"
#include <hip/hip_fp16.h>

typedef __attribute__((ext_vector_type(8))) _Float16 half8;
typedef __attribute__((ext_vector_type(2))) _Float16 half2v;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

DEVINL half8 dq8_q4_f16(unsigned int c, half2v sc2, half2v zp2) {
  const half2v magic = {(_Float16)1024.f, (_Float16)1024.f};
  half8 out;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const unsigned byte = (c >> (8 * p)) & 0xFFu;
    const unsigned h2 = 0x64006400u | (byte & 0xFu) | ((byte & 0xF0u) << 12);
    half2v v = __builtin_bit_cast(half2v, h2);
    // subtract the magic FIRST: (1024+q)-1024 == q exactly in f16; folding
    // 1024*sc into the zero-point rounds at ulp(1024*sc)
    v = (v - magic) * sc2 + zp2;
    out[2 * p] = v[0];
    out[2 * p + 1] = v[1];
  }
  return out;
}

// ---------------------------------------------------------------------------
// M == 1: GEMV. grid = (ceil(N / (4*ROWS)),) x 256 threads; wave w covers
// ROWS consecutive rows with the next row's 2 KB of codes prefetched.
// PRELOAD: K <= 4096 keeps the whole A row in registers (K/64 half2/lane);
// larger K re-reads A fragments from L1 inside the loop.
// ---------------------------------------------------------------------------
// MROWS (1..4): batch rows sharing one code stream — the dequant (the
// issue-bound part, see profiles/r02 §4 PMC) is amortized over MROWS
// v_dot2 chains, so M<=4 runs near the M=1 byte rate instead of falling
// back to the 2.25 TB/s MFMA form (ROUND3 item 4).
template <int NLOADS, int MROWS = 1>  // NLOADS = K / 2048
__global__ __launch_bounds__(256) void gemm_w4_gemv_kernel(
    const _Float16* __restrict__ A,           // (MROWS, K) fp16
    const unsigned char* __restrict__ Wq,     // (N, K/2)
    const __half* __restrict__ scale,         // (N, K/64)
    const __half* __restrict__ zero,          // (N, K/64)
    const unsigned short* __restrict__ R,     // (MROWS, N) bf16 or null
    const unsigned short* __restrict__ bias,  // (N,) bf16 or null
    unsigned short* __restrict__ C,           // (MROWS, N) bf16
    int N, int K, int rows_per_wave) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_base = (blockIdx.x * 4 + wave) * rows_per_wave;
  if (n_base >= N) return;
  constexpr bool PRELOAD = (MROWS == 1 ? NLOADS <= 2 : MROWS * NLOADS <= 8);
  const half2v magic = {(_Float16)1024.f, (_Float16)1024.f};

  half2v areg[PRELOAD ? MROWS * NLOADS * 16 : 1];
  if (PRELOAD) {
#pragma unroll
    for (int m = 0; m < MROWS; ++m)
#pragma unroll
      for (int h = 0; h < NLOADS; ++h)
#pragma unroll
        for (int j = 0; j < 16; ++j)
          areg[(m * NLOADS + h) * 16 + j] = *reinterpret_cast<const half2v*>(
              A + (long)m * K + h * 2048 + lane * 32 + 2 * j);
  }

  uint4v buf[2][NLOADS];  // [row parity][k-chunk]
  auto ldrow = [&](int slot, int n) {
    const unsigned char* wrow = Wq + (long)n * (K / 2);
#pragma unroll
    for (int h = 0; h < NLOADS; ++h)
      buf[slot][h] =
          *reinterpret_cast<const uint4v*>(wrow + h * 1024 + lane * 16);
  };
  ldrow(0, n_base);
  for (int r = 0; r < rows_per_wave; ++r) {
    const int n = n_base + r;
    if (n >= N) break;
    if (r + 1 < rows_per_wave && n + 1 < N) ldrow((r + 1) & 1, n + 1);
    const __half* srow = scale + (long)n * (K / 64);
    const __half* zrow = zero + (long)n * (K / 64);
    float acc[MROWS];
#pragma unroll
    for (int m = 0; m < MROWS; ++m) acc[m] = 0.f;
#pragma unroll
    for (int h = 0; h < NLOADS; ++h) {
      const int g = (h * 2048 + lane * 32) / 64;
      const _Float16 sch = (_Float16)__half2float(srow[g]);
      const _Float16 zph = (_Float16)__half2float(zrow[g]);
      const half2v sc2 = {sch, sch};
      const half2v zp2 = {zph, zph};
      const uint4v cw = buf[r & 1][h];
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        const unsigned int c = cw[d];
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          const unsigned byte = (c >> (8 * p)) & 0xFFu;
          const unsigned hh =
              0x64006400u | (byte & 0xFu) | ((byte & 0xF0u) << 12);
          half2v v = __builtin_bit_cast(half2v, hh);
          v = (v - magic) * sc2 + zp2;
#pragma unroll
          for (int m = 0; m < MROWS; ++m) {
            half2v a;
            if constexpr (PRELOAD) {
              a = areg[(m * NLOADS + h) * 16 + d * 4 + p];
            } else {
              a = *reinterpret_cast<const half2v*>(
                  A + (long)m * K + h * 2048 + lane * 32 + (d * 4 + p) * 2);
            }
            acc[m] = __builtin_amdgcn_fdot2(a, v, acc[m], false);
          }
        }
      }
    }
#pragma unroll
    for (int m = 0; m < MROWS; ++m) {
#pragma unroll
      for (int x = 1; x < WAVE; x <<= 1) acc[m] += __shfl_xor(acc[m], x);
      if (lane == 0) {
        float av = acc[m];
        if (bias) av += bf2f(bias[n]);
        if (R) av += bf2f(R[(long)m * N + n]);
        C[(long)m * N + n] = f2bf(av);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// M 2..32: MFMA form. 16 B code words per lane (1/4 the request rate of
// per-fragment 4 B loads), distributed to fragment lanes by predicated
// shfl; next 128-k iteration's codes + scales prefetched.
// ---------------------------------------------------------------------------
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
  // XCD-aware remap (see gemm_skinny_v2)
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
  if (k0 >= K) return;

  const unsigned char* wrow = Wq + (long)(n0 + li) * (K / 2);
  const __half* srow = scale + (long)(n0 + li) * (K / 64);
  const __half* zrow = zero + (long)(n0 + li) * (K / 64);

  int arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) arow[t] = min(t * 16 + li, M - 1);

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // software pipeline: this 128-k iteration's 16 B code word + 4 B scale
  // words were loaded last iteration
  uint4v cwA = *reinterpret_cast<const uint4v*>(wrow + (k0 + hi * 32) / 2);
  unsigned int svA = *reinterpret_cast<const unsigned int*>(srow + k0 / 64);
  unsigned int zvA = *reinterpret_cast<const unsigned int*>(zrow + k0 / 64);
  for (int k = k0; k < k1; k += 128) {
    uint4v cwB;
    unsigned int svB = 0, zvB = 0;
    if (k + 128 < k1) {
      cwB = *reinterpret_cast<const uint4v*>(wrow + (k + 128 + hi * 32) / 2);
      svB = *reinterpret_cast<const unsigned int*>(srow + (k + 128) / 64);
      zvB = *reinterpret_cast<const unsigned int*>(zrow + (k + 128) / 64);
    }
    const half2v sh = __builtin_bit_cast(half2v, svA);
    const half2v zh = __builtin_bit_cast(half2v, zvA);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const half2v sc2 = {sh[s / 2], sh[s / 2]};
      const half2v zp2 = {zh[s / 2], zh[s / 2]};
      // slice s's fragment codes live in dword hi of lane (s*16 + li)
      unsigned int cw = 0;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        unsigned int v = __shfl(cwA[j], s * 16 + li);
        if (hi == j) cw = v;
      }
      half8 bfrag = dq8_q4_f16(cw, sc2, zp2);
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        half8 afrag = *reinterpret_cast<const half8*>(
            A + (long)arow[t] * K + k + s * 32 + hi * 8);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_f16(afrag, bfrag,
                                                        acc[t], 0, 0, 0);
      }
    }
    cwA = cwB;
    svA = svB;
    zvA = zvB;
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
