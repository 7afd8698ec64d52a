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

// v2: software-pipelined direct-VGPR W streaming + A through LDS.
// The guide's verdicts for this shape class (cdna_hip_programming.md):
//  * "GEMV / M <= 16 decode weights" — W is streamed once and not shared
//    across waves: no LDS round trip, load straight to VGPRs, deep unroll,
//    late vmcnt. Two register sets alternate so each wave keeps a
//    HBM-latency's worth of W bytes in flight.
//  * "the x operand ... through LDS in full 128-B lines, NOT fragment-shaped
//    loads straight to VGPRs" — per-fragment A loads double TA pressure
//    (+18..45% measured in the guide); here the (M x 256) A stage is copied
//    to LDS once per stage with full-line loads and A fragments come from
//    conflict-free padded ds_read_b128.
// Host guarantees (k1 - k0) % 256 == 0 so the set pairing needs no tail.
template <int MT>
__global__ __launch_bounds__(256) void gemm_skinny_v2_kernel(
    const unsigned short* __restrict__ A,   // (M, K) bf16
    const unsigned short* __restrict__ W,   // (N, K) bf16
    const unsigned short* __restrict__ R,   // (M, N) bf16 residual or null
    const unsigned short* __restrict__ bias,  // (N,) bf16 or null
    unsigned short* __restrict__ C,         // (M, N) bf16   (ksplit == 1)
    float* __restrict__ Cpart,              // (ksplit, M, N) f32 (ksplit > 1)
    int M, int N, int K, int kchunk, int ksplit,
    const unsigned short* __restrict__ nw,  // (K,) rmsnorm weight (mode 1)
    float eps, int norm_mode,  // 0: none; 1: scale A by invr*nw (stage path);
                               // 2: weight pre-folded into W — scale the
                               //    ACCUMULATOR by invr in the epilogue
                               //    (zero per-stage cost; the production path)
    const float* __restrict__ ssin,   // (nstripes, 32) producer row sum-sq
    int nstripes,
    float* __restrict__ ssout) {      // (N/64, 32) this GEMM's row sum-sq
  constexpr int U = 4;                     // k-slices per pipeline set (128 k)
  constexpr int KSTEP = 256;               // A elements staged per stage
  constexpr int RSTRIDE = KSTEP + 8;       // padded LDS row stride (elements)
  __shared__ __attribute__((aligned(16))) unsigned short atile[32 * RSTRIDE];
  // Fused rmsnorm of A (nw != null): removes the separate rms_norm launch +
  // its read/write from the decode step (launch-count reduction — the PMC
  // work in profiles/r02 §11 showed the in-context GEMM cost is boundary/
  // drain, not cache, so fewer kernels is the lever). Each WG redundantly
  // streams A (M*K bf16, L2-resident after the first WG) to compute per-row
  // inv-rms, then scales A fragments by invr[m]*nw[k] as they stage to LDS.
  __shared__ float ssp[256];
  __shared__ float invr[32];

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  // Bijective XCD-aware remap: the dispatcher places block b on XCD b%8, so
  // without the remap memory-adjacent N-tiles land on different XCDs. Group
  // them instead: each XCD streams one contiguous W region (guide: +10-12%
  // when HBM-bound).
  int tile = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int xcd = tile % 8, orig8 = tile / 8;
    const int q = nwg / 8, r = nwg % 8;
    if (nwg >= 8)
      tile = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig8;
  }
  const int n0 = tile * 64 + wave * 16;
  const int split = blockIdx.y;
  const int k0 = split * kchunk;
  const int k1 = min(K, k0 + kchunk);

  const unsigned short* wrow = W + (long)(n0 + li) * K;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // ---- A staging: the whole workgroup copies A[0:M][ks:ks+256] with
  // full-line loads; rows beyond M-1 are clamped (their C rows are dropped).
  // Thread piece j: row = (tid + j*256)/32 clamped, 8 elements at
  // ((tid + j*256)%32)*8 within the stage.
  short8 aregs[(32 * KSTEP) / (256 * 8)];  // 4 pieces per thread
  short8 nregs[(32 * KSTEP) / (256 * 8)];  // matching norm-weight pieces
  auto load_a = [&](int ks) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int flat = threadIdx.x + j * 256;
      const int row = min(flat / 32, M - 1);
      const int off = (flat % 32) * 8;
      aregs[j] = *reinterpret_cast<const short8*>(A + (long)row * K + ks + off);
      if (norm_mode == 1)
        nregs[j] = *reinterpret_cast<const short8*>(nw + ks + off);
    }
  };
  auto store_a = [&]() {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int flat = threadIdx.x + j * 256;
      short8 v = aregs[j];
      if (norm_mode == 1) {
        const float ir = invr[flat / 32 >= M ? M - 1 : flat / 32];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = (short)f2bf(bf2f((unsigned short)v[e]) * ir *
                             bf2f((unsigned short)nregs[j][e]));
      }
      *reinterpret_cast<short8*>(atile + (flat / 32) * RSTRIDE + (flat % 32) * 8) =
          v;
    }
  };

  bf16x8 bB0[U], bB1[U];
  auto issue0 = [&](int k) {
#pragma unroll
    for (int u = 0; u < U; ++u)
      bB0[u] = as_bf16x8(
          *reinterpret_cast<const short8*>(wrow + k + u * 32 + hi * 8));
  };
  auto issue1 = [&](int k) {
#pragma unroll
    for (int u = 0; u < U; ++u)
      bB1[u] = as_bf16x8(
          *reinterpret_cast<const short8*>(wrow + k + u * 32 + hi * 8));
  };
  // A fragment for m-tile t, k-slice u of the staged 256: ds_read_b128 at
  // row (t*16 + li), element (u*32 + hi*8) — row stride 528 B => lanes hit
  // distinct banks.
  auto mfma_set = [&](bf16x8* bB, int half) {
#pragma unroll
    for (int u = 0; u < U; ++u)
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        bf16x8 afrag = as_bf16x8(*reinterpret_cast<const short8*>(
            atile + (t * 16 + li) * RSTRIDE + half * 128 + u * 32 + hi * 8));
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bB[u], acc[t],
                                                         0, 0, 0);
      }
  };

  // invr reduce from producer stats (coalesced: lanes sweep each stripe's
  // contiguous 32 floats; 4 accumulator chains keep loads in flight) or
  // from re-streaming A (fallback for callers with no producer stats —
  // +18-60 us/GEMM under split-K, profiles/r02 §13).
  auto reduce_invr = [&]() {
    if (ssin) {
      const int row = threadIdx.x & 31, grp = threadIdx.x >> 5;
      float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
      int s = grp;
      for (; s + 24 < nstripes; s += 32) {
        s0 += ssin[(long)s * 32 + row];
        s1 += ssin[(long)(s + 8) * 32 + row];
        s2 += ssin[(long)(s + 16) * 32 + row];
        s3 += ssin[(long)(s + 24) * 32 + row];
      }
      for (; s < nstripes; s += 8) s0 += ssin[(long)s * 32 + row];
      ssp[threadIdx.x] = (s0 + s1) + (s2 + s3);
      __syncthreads();
      if (threadIdx.x < 32) {
        float t = 0.f;
#pragma unroll
        for (int g = 0; g < 8; ++g) t += ssp[g * 32 + threadIdx.x];
        invr[threadIdx.x] = rsqrtf(t / K + eps);
      }
      __syncthreads();
    } else {
      const int row = threadIdx.x >> 3, sub = threadIdx.x & 7;
      float ss = 0.f;
      if (row < M) {
        const unsigned short* p = A + (long)row * K + sub * (K / 8);
        for (int x = 0; x < K / 8; x += 8) {
          short8 v = *reinterpret_cast<const short8*>(p + x);
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const float f = bf2f((unsigned short)v[e]);
            ss += f * f;
          }
        }
      }
      ssp[threadIdx.x] = ss;
      __syncthreads();
      if (sub == 0 && row < M) {
        float t = 0.f;
#pragma unroll
        for (int e = 0; e < 8; ++e) t += ssp[(row << 3) + e];
        invr[row] = rsqrtf(t / K + eps);
      }
      __syncthreads();
    }
  };

  if (norm_mode == 1) {
    // mode 1 scales A as it stages: invr must exist BEFORE the pipeline.
    // Issue the first W-register set first so the stream is in flight.
    issue0(k0);
    reduce_invr();
    load_a(k0);
    store_a();
    __syncthreads();
  } else {
    // Pipeline: stage s's A sits in LDS while its two B register sets
    // stream; stage s+1's A loads issue right after the barrier frees them.
    // mode 2 defers its invr reduce to just before the epilogue — the
    // weight stream starts with zero added latency.
    load_a(k0);
    issue0(k0);
    store_a();
    __syncthreads();
  }
  for (int k = k0; k < k1; k += KSTEP) {
    if (k + KSTEP < k1) load_a(k + KSTEP);
    issue1(k + 128);
    mfma_set(bB0, 0);
    if (k + KSTEP < k1) issue0(k + KSTEP);
    mfma_set(bB1, 1);
    if (k + KSTEP < k1) {
      __syncthreads();  // everyone done reading stage s
      store_a();
      __syncthreads();
    }
  }

  if (norm_mode == 2) reduce_invr();

  if (ksplit == 1) {
    if (ssout == nullptr) {
#pragma unroll
      for (int t = 0; t < MT; ++t)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int m = t * 16 + hi * 4 + reg;
          if (m >= M) continue;
          float v = acc[t][reg];
          if (norm_mode == 2) v *= invr[m];
          if (bias) v += bf2f(bias[n0 + li]);
          if (R) v += bf2f(R[(long)m * N + n0 + li]);
          C[(long)m * N + n0 + li] = f2bf(v);
        }
    } else {
      // Emit per-stripe row sum-of-squares alongside the store: li-group
      // shfl reduce (16 cols per wave), cross-wave fold via ssp. The next
      // GEMM's fused rmsnorm sums these instead of re-streaming A.
      __syncthreads();  // atile is dead; reuse ssp[wave*32 + m]
#pragma unroll
      for (int t = 0; t < MT; ++t)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int m = t * 16 + hi * 4 + reg;
          float vs = 0.f;
          if (m < M) {
            float v = acc[t][reg];
            if (norm_mode == 2) v *= invr[m];
            if (bias) v += bf2f(bias[n0 + li]);
            if (R) v += bf2f(R[(long)m * N + n0 + li]);
            C[(long)m * N + n0 + li] = f2bf(v);
            vs = v * v;
          }
#pragma unroll
          for (int msk = 1; msk < 16; msk <<= 1) vs += __shfl_xor(vs, msk);
          if (li == 0) ssp[wave * 32 + m] = vs;
        }
      __syncthreads();
      if (threadIdx.x < 32)
        ssout[(long)tile * 32 + threadIdx.x] =
            ssp[threadIdx.x] + ssp[32 + threadIdx.x] +
            ssp[64 + threadIdx.x] + ssp[96 + threadIdx.x];
    }
  } else {
    float* dst = Cpart + (long)split * M * N;
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = t * 16 + hi * 4 + reg;
        if (m >= M) continue;
        float v = acc[t][reg];
        if (norm_mode == 2) v *= invr[m];
        dst[(long)m * N + n0 + li] = v;
      }
  }
}

// Split-K combine that also emits per-stripe row sum-of-squares (ssout
// layout (N/64, 32)): grid (N/64, ceil(M/16)); a wave covers 4 rows x 16
// float4 column-groups of the 64-col stripe, so partial loads are 16 B
// vectors (16 lanes x 16 B = one coalesced 256 B line run per split per
// row), the row sum is an in-lane fold + a 16-lane shfl reduce, and each
// (stripe, m) is produced by exactly one lane — no LDS, no atomics.
__global__ __launch_bounds__(256) void gemm_skinny_combine_ss_kernel(
    const float* __restrict__ Cpart, const unsigned short* __restrict__ R,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ C,
    float* __restrict__ ssout, int M, int N, int ksplit) {
  const int n0 = blockIdx.x * 64;
  const int lane = threadIdx.x & (WAVE - 1);
  const int c4 = (lane & 15) * 4;            // first of this lane's 4 cols
  const int m = blockIdx.y * 16 + (threadIdx.x >> 6) * 4 + (lane >> 4);
  if (m >= M) return;
  const long total = (long)M * N;
  const long i = (long)m * N + n0 + c4;
  f32x4 v = {0.f, 0.f, 0.f, 0.f};
  for (int s = 0; s < ksplit; ++s) {
    const f32x4 p = *reinterpret_cast<const f32x4*>(Cpart + s * total + i);
#pragma unroll
    for (int e = 0; e < 4; ++e) v[e] += p[e];
  }
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    if (bias) v[e] += bf2f(bias[n0 + c4 + e]);
    if (R) v[e] += bf2f(R[i + e]);
  }
  short4v o4;
#pragma unroll
  for (int e = 0; e < 4; ++e) o4[e] = (short)f2bf(v[e]);
  *reinterpret_cast<short4v*>(C + i) = o4;
  if (ssout) {
    float vs = v[0] * v[0] + v[1] * v[1] + v[2] * v[2] + v[3] * v[3];
#pragma unroll
    for (int msk = 1; msk < 16; msk <<= 1) vs += __shfl_xor(vs, msk);
    if ((lane & 15) == 0) ssout[(long)blockIdx.x * 32 + m] = vs;
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
