// Skinny-M GEMM for gfx950: C(M,N) = A(M,K) @ W(N,K)^T [+ residual] [+ bias],
// M <= 32 — the decode-step projection shape (batch tokens x weight matrix).
//
// Decode GEMMs are pure weight streams: the job is to read W (N*K*2 bytes)
// once at full HBM bandwidth. hipBLASLt's skinny-M solutions measure
// ~2.6-4.4 TB/s on these shapes (profiles/r01_decode_kernel_stats.md); this
// kernel exists to close that gap. Replaces the reference's decode-path
// torch.matmul projections (flexgen_utils/pytorch_backend.py:733-916).
//
// Decomposition: grid = (N/64, ksplit); 4 waves per 256-thread workgroup,
// wave w owns 16 N-rows of W. B-fragments (8 consecutive k at fixed n) are
// direct contiguous 16 B loads from W; with the k-loop unrolled 4x each W row
// is streamed in 256 B sequential pieces. A (<= 32xK bf16, a few hundred KB)
// is re-read per wave but L2-resident after the first touch. No LDS at all.
//
// Split-K (ksplit > 1) for small-N shapes that would otherwise leave the
// chip idle (N=4096 -> only 64 workgroups): each split writes an f32 partial
// (ksplit, M, N); gemm_skinny_combine_kernel folds them + residual/bias and
// converts to bf16.

#include "common.h"

template <int MT>  // number of 16-row M-tiles (1: M<=16, 2: M<=32)
__global__ __launch_bounds__(256) void gemm_skinny_kernel(
    const unsigned short* __restrict__ A,   // (M, K) bf16
    const unsigned short* __restrict__ W,   // (N, K) bf16
    const unsigned short* __restrict__ R,   // (M, N) bf16 residual or null
    const unsigned short* __restrict__ bias,  // (N,) bf16 or null
    unsigned short* __restrict__ C,         // (M, N) bf16   (ksplit == 1)
    float* __restrict__ Cpart,              // (ksplit, M, N) f32 (ksplit > 1)
    int M, int N, int K, int ksplit) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  const int n0 = blockIdx.x * 64 + wave * 16;
  if (n0 >= N) return;
  const int split = blockIdx.y;
  const int kchunk = ((K / 32 + ksplit - 1) / ksplit) * 32;
  const int k0 = split * kchunk;
  const int k1 = min(K, k0 + kchunk);

  const unsigned short* wrow = W + (long)(n0 + li) * K;
  // A rows this lane feeds (garbage-row trick for M not a multiple of 16:
  // clamped rows compute garbage C rows that are simply not stored).
  int arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) arow[t] = min(t * 16 + li, M - 1);

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  int k = k0;
  for (; k + 128 <= k1; k += 128) {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int kk = k + u * 32 + hi * 8;
      bf16x8 bfrag = as_bf16x8(*reinterpret_cast<const short8*>(wrow + kk));
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        bf16x8 afrag = as_bf16x8(
            *reinterpret_cast<const short8*>(A + (long)arow[t] * K + kk));
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t],
                                                         0, 0, 0);
      }
    }
  }
  for (; k < k1; k += 32) {
    const int kk = k + hi * 8;
    bf16x8 bfrag = as_bf16x8(*reinterpret_cast<const short8*>(wrow + kk));
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      bf16x8 afrag = as_bf16x8(
          *reinterpret_cast<const short8*>(A + (long)arow[t] * K + kk));
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t],
                                                       0, 0, 0);
    }
  }

  // C layout: row = t*16 + hi*4 + reg, col = n0 + li.
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

// v2: software-pipelined direct-VGPR W streaming. The guide's verdict for
// M<=32 decode-weight GEMMs (cdna_hip_programming.md "GEMV / M <= 16 decode
// weights" row): the W operand is streamed once and not shared across waves,
// so an LDS round trip is pure overhead — load straight to VGPRs with a deep
// unroll and late vmcnt. Two register sets alternate: while set s's MFMAs
// consume, set s^1's 128-k-deep loads (4 KB of W per wave) are in flight, so
// each wave keeps ~one HBM-latency's worth of bytes outstanding.
// Host guarantees (k1 - k0) % 256 == 0 so the set pairing needs no tail.
template <int MT>
__global__ __launch_bounds__(256) void gemm_skinny_v2_kernel(
    const unsigned short* __restrict__ A,   // (M, K) bf16
    const unsigned short* __restrict__ W,   // (N, K) bf16
    const unsigned short* __restrict__ R,   // (M, N) bf16 residual or null
    const unsigned short* __restrict__ bias,  // (N,) bf16 or null
    unsigned short* __restrict__ C,         // (M, N) bf16   (ksplit == 1)
    float* __restrict__ Cpart,              // (ksplit, M, N) f32 (ksplit > 1)
    int M, int N, int K, int kchunk, int ksplit) {
  constexpr int U = 4;                     // k-slices per pipeline set (128 k)
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  const int n0 = blockIdx.x * 64 + wave * 16;
  if (n0 >= N) return;
  const int split = blockIdx.y;
  const int k0 = split * kchunk;
  const int k1 = min(K, k0 + kchunk);

  const unsigned short* wrow = W + (long)(n0 + li) * K;
  int arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) arow[t] = min(t * 16 + li, M - 1);

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  bf16x8 bB0[U], bB1[U], bA0[U][MT], bA1[U][MT];
  auto issue0 = [&](int k) {
#pragma unroll
    for (int u = 0; u < U; ++u) {
      const int kk = k + u * 32 + hi * 8;
      bB0[u] = as_bf16x8(*reinterpret_cast<const short8*>(wrow + kk));
#pragma unroll
      for (int t = 0; t < MT; ++t)
        bA0[u][t] = as_bf16x8(
            *reinterpret_cast<const short8*>(A + (long)arow[t] * K + kk));
    }
  };
  auto issue1 = [&](int k) {
#pragma unroll
    for (int u = 0; u < U; ++u) {
      const int kk = k + u * 32 + hi * 8;
      bB1[u] = as_bf16x8(*reinterpret_cast<const short8*>(wrow + kk));
#pragma unroll
      for (int t = 0; t < MT; ++t)
        bA1[u][t] = as_bf16x8(
            *reinterpret_cast<const short8*>(A + (long)arow[t] * K + kk));
    }
  };
  auto mfma0 = [&]() {
#pragma unroll
    for (int u = 0; u < U; ++u)
#pragma unroll
      for (int t = 0; t < MT; ++t)
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bA0[u][t], bB0[u],
                                                         acc[t], 0, 0, 0);
  };
  auto mfma1 = [&]() {
#pragma unroll
    for (int u = 0; u < U; ++u)
#pragma unroll
      for (int t = 0; t < MT; ++t)
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(bA1[u][t], bB1[u],
                                                         acc[t], 0, 0, 0);
  };

  // (k1 - k0) is a multiple of 256 = two sets; pipeline pairs of sets.
  issue0(k0);
  for (int k = k0; k < k1; k += 256) {
    issue1(k + 128);
    mfma0();
    if (k + 256 < k1) issue0(k + 256);
    mfma1();
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

__global__ void gemm_skinny_combine_kernel(
    const float* __restrict__ Cpart, const unsigned short* __restrict__ R,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ C,
    int M, int N, int ksplit) {
  const long total = (long)M * N;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = 0.f;
    for (int s = 0; s < ksplit; ++s) v += Cpart[(long)s * total + i];
    if (bias) v += bf2f(bias[i % N]);
    if (R) v += bf2f(R[i]);
    C[i] = f2bf(v);
  }
}
