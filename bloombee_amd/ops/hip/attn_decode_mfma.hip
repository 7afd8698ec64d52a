// MFMA-based single-token paged attention for gfx950.
//
// The VALU formulation (attn_decode.hip) spends ~210 VALU instructions per
// position pair (dots, shuffles, exp2, accumulate) and measures VALU-bound at
// ~3 TB/s of KV streaming. Here the score dots and the P.V accumulation run
// on the matrix cores: one 16x16x32 MFMA computes 16 positions x 16 rows at
// once. The "q tile" is the G query heads of one (b, kv_head) GQA group —
// rows >= G are garbage lanes that only pollute garbage output rows (C row r
// depends only on A row r), so a 4-head group costs 4/16 MFMA efficiency and
// the kernel is HBM-bound again, which is the point.
//
// Work decomposition: grid = (B*Hkv, n_split), 4 waves per workgroup; wave w
// owns KV tiles (32 positions) w, w+4, w+8, ... of the chunk, with a final
// 4-way LDS merge of (m, l, acc). K fragments are read DIRECTLY from the
// paged pool (B-fragment address pattern covers each (page, head) slab in
// aligned 64 B pieces exactly once — no LDS round trip); V is transpose-
// staged through per-wave LDS for the P.V operand.

#include "common.h"

static constexpr int DKVBLK = 32;  // positions per KV tile

template <int D, int MAXG>
__global__ __launch_bounds__(256) void attn_decode_mfma_kernel(
    const unsigned short* __restrict__ q,        // strided, see q_sb/q_sh
    const unsigned short* __restrict__ k_pages,  // (np, Hkv, P, D)
    const unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table,          // (B, maxp)
    const int* __restrict__ ctx_lens,            // (B,)
    unsigned short* __restrict__ out,
    float* __restrict__ part_ml,                 // (B*Hkv*n_split, G, 2)
    float* __restrict__ part_acc,                // (B*Hkv*n_split, G, D)
    int B, int Hkv, int G, int P, int maxp, int n_split, int window,
    float scale, long q_sb, long q_sh, long out_sb, long out_sh) {
  constexpr int NKK = D / 32;   // QK^T k-slices
  constexpr int NDT = D / 16;   // PV d-tiles
  constexpr int VROW_B = DKVBLK * 2 + 16;  // transposed V row stride (80 B)
  constexpr int PROW_B = DKVBLK * 2 + 16;

  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / Hkv, kvh = bh % Hkv;
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

  // per-wave LDS: transposed V tile + P tile + merge scratch
  __shared__ __attribute__((aligned(16))) unsigned char lds[
      4 * (D * VROW_B + 16 * PROW_B) + 3 * (MAXG * (D + 2) * 4)];
  unsigned char* v_lds = lds + wave * (D * VROW_B + 16 * PROW_B);
  unsigned char* p_lds = v_lds + D * VROW_B;
  float* merge_lds = (float*)(lds + 4 * (D * VROW_B + 16 * PROW_B));

  // Q fragments: A[row][k] with row = head g (rows >= G harmless garbage)
  const int qh = (li < G) ? li : G - 1;
  bf16x8 qfrag[NKK];
#pragma unroll
  for (int kk = 0; kk < NKK; ++kk)
    qfrag[kk] = as_bf16x8(*reinterpret_cast<const short8*>(
        q + (long)b * q_sb + (long)(kvh * G + qh) * q_sh + hi * 8 + 32 * kk));

  float m2[4], l[4];
  f32x4 acc_o[NDT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m2[r] = NEG_BIG; l[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < NDT; ++n) acc_o[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const long head_slab = (long)kvh * P * D;
  const int tile0 = (max(c0, lo) - c0) / DKVBLK;  // window skip, tile-aligned

  constexpr int VPASS = DKVBLK * D / (64 * 8);  // V loads per lane per tile
  for (int tb = c0 + (tile0 + wave) * DKVBLK; tb < c1; tb += 4 * DKVBLK) {
    // ---- T14 split: issue this tile's V loads FIRST (write to LDS later,
    // after the QK^T phase has covered their HBM latency) ----
    const int vrow0 = lane / (D / 8);
    const int vd8 = (lane % (D / 8)) * 8;
    short8 v_raw[VPASS];
#pragma unroll
    for (int pass = 0; pass < VPASS; ++pass) {
      const int row = pass * (64 * 8 / D) + vrow0;
      const int pos = tb + row;
      v_raw[pass] = short8{};
      if (pos < c1) {
        const int page = page_table[b * maxp + pos / P];
        v_raw[pass] = *reinterpret_cast<const short8*>(
            v_pages + ((long)page * Hkv) * P * D + head_slab +
            (long)(pos % P) * D + vd8);
      }
    }

    // ---- S = Q K^T : K fragments straight from the paged pool ----
    f32x4 s[DKVBLK / 16];
#pragma unroll
    for (int n = 0; n < DKVBLK / 16; ++n) {
      const int pos = tb + li + 16 * n;
      const int cpos = min(pos, c1 - 1);
      const int page = page_table[b * maxp + cpos / P];
      const unsigned short* krow = k_pages + ((long)page * Hkv) * P * D +
                                   head_slab + (long)(cpos % P) * D;
      s[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        bf16x8 bfrag = as_bf16x8(
            *reinterpret_cast<const short8*>(krow + hi * 8 + 32 * kk));
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], bfrag, s[n],
                                                       0, 0, 0);
      }
    }

    // ---- V raw regs -> transposed LDS image (loads have landed by now) ----
#pragma unroll
    for (int pass = 0; pass < VPASS; ++pass) {
      const int row = pass * (64 * 8 / D) + vrow0;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *(unsigned short*)(v_lds + (vd8 + j) * VROW_B + row * 2) =
            (unsigned short)v_raw[pass][j];
    }

    // ---- mask + online softmax (all 16 rows share the same position) ----
    float p[DKVBLK / 16][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float rm = NEG_BIG;
#pragma unroll
      for (int n = 0; n < DKVBLK / 16; ++n) {
        const int kpos = tb + li + 16 * n;
        float sv = s[n][reg] * sc2;
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
#pragma unroll
      for (int n = 0; n < NDT; ++n) acc_o[n][reg] *= corr;
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

    // ---- acc += P V ----
#pragma unroll
    for (int n = 0; n < NDT; ++n) {
      bf16x8 vfrag = as_bf16x8(*reinterpret_cast<const short8*>(
          v_lds + (li + 16 * n) * VROW_B + hi * 16));
      acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag,
                                                         acc_o[n], 0, 0, 0);
    }
  }

  // ---- merge the 4 waves ----
  // C row r = hi*4 + reg holds head g = r (valid for g < G, so MQA G up to
  // 16 works). Waves 1..3 publish (acc rows, m, l) to LDS; wave 0 folds them.
  auto wave_slot = [&](int slot) { return merge_lds + slot * MAXG * (D + 2); };
  __syncthreads();
  if (wave > 0) {
    float* dst = wave_slot(wave - 1);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g < G) {
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
      if (g >= G) continue;
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
      if (g >= G) continue;
      const float inv = (l[reg] > 0.f) ? 1.f / l[reg] : 0.f;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        out[(long)b * out_sb + (long)(kvh * G + g) * out_sh + li + 16 * n] =
            f2bf(acc_o[n][reg] * inv);
    }
  } else {
    const long pbase = ((long)bh * n_split + split);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int g = hi * 4 + reg;
      if (g >= G) continue;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        part_acc[(pbase * G + g) * D + li + 16 * n] = acc_o[n][reg];
      if (li == 0) {
        part_ml[(pbase * G + g) * 2 + 0] = m2[reg];
        part_ml[(pbase * G + g) * 2 + 1] = l[reg];
      }
    }
  }
}
