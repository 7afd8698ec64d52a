// Grouped MoE expert GEMM for gfx950: ONE launch over all experts.
//
//   C[s, :] = A[row(s), :] @ W[expert(s), :, :]^T        s = 0..S-1
//
// Sorted slots are grouped by expert: slots [off[e], off[e+1]) belong to
// expert e (host sorts token->expert assignments; off is the exclusive
// prefix sum of per-expert counts, all ON DEVICE — no host sync, so the
// whole MoE step stays hipGraph-capturable). rowmap maps a slot to its A
// row (token gather for the gate_up GEMM); null means identity (the down
// GEMM consumes the already-sorted activation rows). scale, when given,
// multiplies C row s by scale[s] in the epilogue — the routing weight fold
// for the down projection.
//
// Shape regime: decode MoE. Per-expert token groups are skinny (avg
// S*topk/E rows); the job is streaming each ACTIVE expert's weight matrix
// once at full HBM bandwidth, exactly like gemm_skinny.hip (same MFMA
// 16x16x32 decomposition, same garbage-row clamp for ragged group sizes).
// Blocks of empty (expert, m-chunk) pairs exit before touching W.
//
// Replaces the per-expert skinny-GEMM loop (BASELINE.json config 4 "MoE
// grouped GEMM"; reference runs stock HF experts, models/mixtral/block.py:
// 13-137, one matmul pair per expert plus a host-synced nonzero per
// expert).
//
// Grid: (ceil(N/64) XCD-remapped, E * mchunks); 4 waves x 16 N-rows each.

#include "common.h"

template <int MT>  // 16-row m-tiles per block (2 => 32-row chunks)
__global__ __launch_bounds__(256) void moe_gemm_kernel(
    const unsigned short* __restrict__ A,   // (Ta, K) bf16
    const unsigned short* __restrict__ W,   // (E, N, K) bf16
    const int* __restrict__ off,            // (E+1,) exclusive prefix sum
    const int* __restrict__ rowmap,         // (S,) slot -> A row, or null
    const float* __restrict__ scale,        // (S,) per-slot scale, or null
    unsigned short* __restrict__ C,         // (S, N) bf16
    int N, int K, int mchunks) {
  const int e = blockIdx.y / mchunks;
  const int m0 = (blockIdx.y % mchunks) * (MT * 16);
  const int base = off[e];
  const int cnt = off[e + 1] - base;
  if (m0 >= cnt) return;  // empty chunk: no W traffic

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;

  // XCD-aware tile remap (see gemm_skinny_v2): group memory-adjacent
  // N-tiles on one XCD so each XCD streams a contiguous W region.
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

  const unsigned short* wrow = W + ((long)e * N + n0 + li) * K;

  // A rows this lane feeds; slots past the group end clamp to the last row
  // (their C rows are simply not stored).
  int arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    int s = base + min(m0 + t * 16 + li, cnt - 1);
    arow[t] = (rowmap != nullptr) ? rowmap[s] : s;
  }

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  int k = 0;
  for (; k + 128 <= K; k += 128) {
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
  for (; k < K; k += 32) {
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

  // C row = slot base + m0 + t*16 + hi*4 + reg, col = n0 + li.
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int m = m0 + t * 16 + hi * 4 + reg;
      if (m >= cnt) continue;
      float v = acc[t][reg];
      if (scale != nullptr) v *= scale[base + m];
      C[(long)(base + m) * N + n0 + li] = f2bf(v);
    }
}

// v2: the gemm_skinny_v2 software pipeline applied to the grouped kernel —
// two W register sets streamed ahead of the MFMAs, A staged through LDS in
// full 128-B lines (indexed token rows resolved once per stage). Requires
// K % 256 == 0 (both Mixtral shapes: H=4096, I=14336 qualify); the host
// falls back to v1 otherwise.
template <int MT>
__global__ __launch_bounds__(256) void moe_gemm_v2_kernel(
    const unsigned short* __restrict__ A,   // (Ta, K) bf16
    const unsigned short* __restrict__ W,   // (E, N, K) bf16
    const int* __restrict__ off,            // (E+1,)
    const int* __restrict__ rowmap,         // (S,) or null
    const float* __restrict__ scale,        // (S,) or null
    unsigned short* __restrict__ C,         // (S, N) bf16
    int N, int K, int mchunks) {
  constexpr int U = 4;
  constexpr int KSTEP = 256;
  constexpr int RSTRIDE = KSTEP + 8;
  __shared__ __attribute__((aligned(16))) unsigned short atile[32 * RSTRIDE];

  const int e = blockIdx.y / mchunks;
  const int m0 = (blockIdx.y % mchunks) * (MT * 16);
  const int base = off[e];
  const int cnt = off[e + 1] - base;
  if (m0 >= cnt) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;

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

  const unsigned short* wrow = W + ((long)e * N + n0 + li) * K;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // A staging: thread piece j covers staged row (tid + j*256)/32; that
  // row's A row is the expert group's slot m0 + r (clamped), via rowmap.
  short8 aregs[4];
  auto src_row = [&](int r) {
    int s = base + min(m0 + r, cnt - 1);
    return (rowmap != nullptr) ? rowmap[s] : s;
  };
  auto load_a = [&](int ks) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int flat = threadIdx.x + j * 256;
      const int row = src_row(flat / 32);
      const int o = (flat % 32) * 8;
      aregs[j] = *reinterpret_cast<const short8*>(A + (long)row * K + ks + o);
    }
  };
  auto store_a = [&]() {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int flat = threadIdx.x + j * 256;
      *reinterpret_cast<short8*>(atile + (flat / 32) * RSTRIDE + (flat % 32) * 8) =
          aregs[j];
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

  load_a(0);
  issue0(0);
  store_a();
  __syncthreads();
  for (int k = 0; k < K; k += KSTEP) {
    if (k + KSTEP < K) load_a(k + KSTEP);
    issue1(k + 128);
    mfma_set(bB0, 0);
    if (k + KSTEP < K) issue0(k + KSTEP);
    mfma_set(bB1, 1);
    if (k + KSTEP < K) {
      __syncthreads();
      store_a();
      __syncthreads();
    }
  }

#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int m = m0 + t * 16 + hi * 4 + reg;
      if (m >= cnt) continue;
      float v = acc[t][reg];
      if (scale != nullptr) v *= scale[base + m];
      C[(long)(base + m) * N + n0 + li] = f2bf(v);
    }
}
