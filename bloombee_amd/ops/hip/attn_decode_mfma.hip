// MFMA-based single-token paged attention for gfx950.
//
// Score dots and the P.V accumulation run on the matrix cores: one 16x16x32
// MFMA computes 16 positions x 16 rows at once. The "q tile" is the G query
// heads of one (b, kv_head) GQA group — rows >= G are garbage lanes that only
// pollute garbage output rows (C row r depends only on A row r), so a 4-head
// group costs 4/16 MFMA efficiency and the kernel is HBM-bound, which is the
// point.
//
// BOTH operand streams read straight from the paged pool — no LDS staging:
//  * K pages are token-major (np, Hkv, P, D): a QK^T B-fragment (8 consecutive
//    d at fixed position) is a contiguous 16 B load.
//  * V pages are d-major (np, Hkv, D, P): a P.V B-fragment (8 consecutive
//    positions at fixed d) is a contiguous 16 B load. The transpose happened
//    once at kv-write time.
// LDS is only used for the P C->A fragment relayout (tiny) and the final
// 4-wave merge, so occupancy is VGPR-limited, not LDS-limited.
//
// Work decomposition: grid = (B*Hkv, n_split) flash-decode splits, 4 waves
// per workgroup; wave w owns KV tiles (32 positions) w, w+4, w+8, ... of its
// split, with a final 4-way LDS merge of (m, l, acc). V fragment loads are
// issued BEFORE the QK^T/softmax phase so their HBM latency is covered.

#include "common.h"

static constexpr int DKVBLK = 32;  // positions per KV tile

// nch: GQA group chunks per kv head (ceil(G/16)) — MQA groups wider than the
// 16-row MFMA q-tile (falcon-7b G=71) split into chunks sharing the kv head.
// alibi: per-query-head slopes (bloom family), or null.
// min 2 workgroups/CU: at the natural 232-VGPR allocation the kernel sat at
// 1 wave/SIMD with no latency hiding (measured ~67 us vs the ~19 us KV-stream
// floor); capping registers doubles resident waves.
template <int D, int MAXG>
__global__ __launch_bounds__(256, 2) void attn_decode_mfma_kernel(
    const unsigned short* __restrict__ q,        // strided, see q_sb/q_sh
    const unsigned short* __restrict__ k_pages,  // (np, Hkv, P, D)
    const unsigned short* __restrict__ v_pages,  // (np, Hkv, D, P) d-major
    const int* __restrict__ page_table,          // (B, maxp)
    const int* __restrict__ ctx_lens,            // (B,)
    const float* __restrict__ alibi,             // (Hq,) slopes or null
    unsigned short* __restrict__ out,
    float* __restrict__ part_ml,                 // (B*Hkv*nch*n_split, MAXG, 2)
    float* __restrict__ part_acc,                // (..., MAXG, D)
    int B, int Hkv, int G, int nch, int P, int maxp, int n_split, int window,
    float scale, long q_sb, long q_sh, long out_sb, long out_sh) {
  constexpr int NKK = D / 32;   // QK^T k-slices
  constexpr int NDT = D / 16;   // PV d-tiles
  constexpr int PROW_B = DKVBLK * 2 + 16;

  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / (Hkv * nch);
  const int rem = bh % (Hkv * nch);
  const int kvh = rem / nch;
  const int g0 = (rem % nch) * MAXG;   // this chunk's first group row
  const int Gl = min(G - g0, MAXG);    // live rows in this chunk
  const int ctx = ctx_lens[b];

  const int pages_total = (ctx + P - 1) / P;
  const int pages_per_split = (pages_total + n_split - 1) / n_split;
  const int c0 = split * pages_per_split * P;
  const int c1 = min(ctx, c0 + pages_per_split * P);
  int lo = 0;
  if (window > 0) lo = max(0, ctx - window);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  const float sc2 = scale * LOG2E;

  // per-wave LDS: P tile (C->A relayout) + merge scratch
  __shared__ __attribute__((aligned(16))) unsigned char lds[
      4 * 16 * PROW_B + 3 * (MAXG * (D + 2) * 4)];
  unsigned char* p_lds = lds + wave * 16 * PROW_B;
  float* merge_lds = (float*)(lds + 4 * 16 * PROW_B);

  // Q fragments: A[row][k] with row = head g (rows >= Gl harmless garbage)
  const int qh = g0 + ((li < Gl) ? li : Gl - 1);
  bf16x8 qfrag[NKK];
#pragma unroll
  for (int kk = 0; kk < NKK; ++kk)
    qfrag[kk] = as_bf16x8(*reinterpret_cast<const short8*>(
        q + (long)b * q_sb + (long)(kvh * G + qh) * q_sh + hi * 8 + 32 * kk));
  // per-row alibi slopes (row r = hi*4 + reg -> head kvh*G + g0 + r)
  float aslope[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int g = g0 + hi * 4 + r;
    aslope[r] = (alibi != nullptr && g < G) ? alibi[kvh * G + g] : 0.f;
  }

  float m2[4], l[4];
  f32x4 acc_o[NDT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m2[r] = NEG_BIG; l[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < NDT; ++n) acc_o[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const long head_slab_k = (long)kvh * P * D;  // same bytes, both layouts
  const int tile0 = (max(c0, lo) - c0) / DKVBLK;  // window skip, tile-aligned

  for (int tb = c0 + (tile0 + wave) * DKVBLK; tb < c1; tb += 4 * DKVBLK) {
    // ---- V fragments for this tile: direct B-layout loads, issued FIRST so
    // the QK^T phase covers their latency. B[k=pos][j=d]: lane (li -> d tile
    // col, hi -> position octet); 8 consecutive positions at fixed d are
    // contiguous in the d-major pool. tb is 32-aligned and P | 32, so each
    // octet sits in one page.
    const int vpos = tb + hi * 8;
    int vpage = -1;
    if (vpos < c1) vpage = page_table[b * maxp + vpos / P];
    const long vrow = ((long)vpage * Hkv + kvh) * D;
    bf16x8 vfrag[NDT];
#pragma unroll
    for (int n = 0; n < NDT; ++n) {
      vfrag[n] = as_bf16x8(short8{});
      if (vpage >= 0)
        vfrag[n] = as_bf16x8(__builtin_nontemporal_load(
            reinterpret_cast<const short8*>(
                v_pages + (vrow + li + 16 * n) * P + (vpos % P))));
    }

    // ---- S = Q K^T : K fragments straight from the paged pool ----
    f32x4 s[DKVBLK / 16];
#pragma unroll
    for (int n = 0; n < DKVBLK / 16; ++n) {
      const int pos = tb + li + 16 * n;
      const int cpos = min(pos, c1 - 1);
      const int page = page_table[b * maxp + cpos / P];
      const unsigned short* krow = k_pages + ((long)page * Hkv) * P * D +
                                   head_slab_k + (long)(cpos % P) * D;
      s[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        // KV is read once per decode step: nontemporal keeps L2 for the
        // GEMM weight streams that follow in the same step
        bf16x8 bfrag = as_bf16x8(__builtin_nontemporal_load(
            reinterpret_cast<const short8*>(krow + hi * 8 + 32 * kk)));
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], bfrag, s[n],
                                                       0, 0, 0);
      }
    }

    // ---- mask + online softmax (all 16 rows share the same position) ----
    float p[DKVBLK / 16][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float rm = NEG_BIG;
#pragma unroll
      for (int n = 0; n < DKVBLK / 16; ++n) {
        const int kpos = tb + li + 16 * n;
        float sv = s[n][reg] * sc2 + aslope[reg] * LOG2E * (float)kpos;
        const bool dead = (kpos >= c1) | (kpos < lo);
        sv = dead ? NEG_BIG : sv;
        p[n][reg] = sv;
        rm = fmaxf(rm, sv);
      }
      rm = group16_reduce_max(rm);
      const float mn = fmaxf(m2[reg], rm);
      const float corr = fast_exp2(m2[reg] - mn);
      float psum = 0.f;
#pragma unroll
      for (int n = 0; n < DKVBLK / 16; ++n) {
        p[n][reg] = fast_exp2(p[n][reg] - mn);
        psum += p[n][reg];
      }
      psum = group16_reduce_sum(psum);
      l[reg] = l[reg] * corr + psum;
      m2[reg] = mn;
      // steady-state decode rarely raises the max after the first tiles:
      // skip the NDT-wide rescale when no lane needs it (wave-uniform branch)
      if (__builtin_amdgcn_ballot_w64(corr < 0.9999f)) {
#pragma unroll
        for (int n = 0; n < NDT; ++n) acc_o[n][reg] *= corr;
      }
    }

    // ---- P through LDS into A-fragment layout ----
#pragma unroll
    for (int n = 0; n < DKVBLK / 16; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        *(unsigned short*)(p_lds + (hi * 4 + reg) * PROW_B + (li + 16 * n) * 2) =
            f2bf(p[n][reg]);
    bf16x8 pfrag = as_bf16x8(
        *reinterpret_cast<const short8*>(p_lds + li * PROW_B + hi * 16));

    // ---- acc += P V (B-fragments were loaded at tile start) ----
#pragma unroll
    for (int n = 0; n < NDT; ++n)
      acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag[n],
                                                         acc_o[n], 0, 0, 0);
  }

  // ---- merge the 4 waves ----
  // C row r = hi*4 + reg holds chunk-local head g = r (valid for g < Gl).
  // Waves 1..3 publish (acc rows, m, l) to LDS; wave 0 folds them.
  auto wave_slot = [&](int slot) { return merge_lds + slot * MAXG * (D + 2); };
  __syncthreads();
  if (wave > 0) {
    float* dst = wave_slot(wave - 1);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g < Gl) {
#pragma unroll
        for (int n = 0; n < NDT; ++n) dst[g * (D + 2) + li + 16 * n] = acc_o[n][reg];
        if (li == 0) {
          dst[g * (D + 2) + D] = m2[reg];
          dst[g * (D + 2) + D + 1] = l[reg];
        }
      }
    }
  }
  __syncthreads();
  if (wave != 0) return;
  for (int slot = 0; slot < 3; ++slot) {
    const float* src = wave_slot(slot);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g >= Gl) continue;
      const float mo = src[g * (D + 2) + D];
      const float lo2 = src[g * (D + 2) + D + 1];
      const float mn = fmaxf(m2[reg], mo);
      const float c1f = fast_exp2(m2[reg] - mn);
      const float c2f = fast_exp2(mo - mn);
      l[reg] = l[reg] * c1f + lo2 * c2f;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        acc_o[n][reg] = acc_o[n][reg] * c1f + src[g * (D + 2) + li + 16 * n] * c2f;
      m2[reg] = mn;
    }
  }

  if (n_split == 1) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g >= Gl) continue;
      const float inv = (l[reg] > 0.f) ? 1.f / l[reg] : 0.f;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        out[(long)b * out_sb + (long)(kvh * G + g0 + g) * out_sh + li + 16 * n] =
            f2bf(acc_o[n][reg] * inv);
    }
  } else {
    const long pbase = ((long)bh * n_split + split);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g >= Gl) continue;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        part_acc[(pbase * MAXG + g) * D + li + 16 * n] = acc_o[n][reg];
      if (li == 0) {
        part_ml[(pbase * MAXG + g) * 2 + 0] = m2[reg];
        part_ml[(pbase * MAXG + g) * 2 + 1] = l[reg];
      }
    }
  }
}
