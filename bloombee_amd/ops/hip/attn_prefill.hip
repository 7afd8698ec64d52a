// Prefill (multi-token) paged attention for gfx950 — flash-style on MFMA.
//
// Replaces the reference's mha_llama prefill path
// (flexgen_utils/pytorch_backend.py:665-731), reading K/V directly from the
// paged cache (new tokens are kv_write'n first, then attended — one code path
// for first-prefill, multi-turn continuation and chunked prefill).
//
// Structure (v2): grid = (ceil(Tq/128), B*Hq); block = 512 (8 waves), each
//   wave owns 16 q rows — 128 rows per workgroup halves the KV re-streaming
//   of the 64-row v1 (every staged tile feeds twice the MFMA work).
//   KV tiles of 32 keys stage through LDS
//   (K row-major padded +16 B -> conflict-free 16-lane ds_read_b128;
//    V transposed [d][key] at 80 B row stride — same property)
//   QK^T and P.V on v_mfma_f32_16x16x32_bf16; online softmax in f32
//   (exp2 domain, scale folded), P round-trips LDS to reach A-fragment
//   layout.
//
// Numerics: fp32 softmax/accum; bf16 operands — matches reference tolerance
// class (test_block_exact_match.py atol 1e-4 forward on fp32 inputs).

#include "common.h"


static constexpr int KVBLK = 32;


template <int D>
__global__ __launch_bounds__(512) void attn_prefill_kernel(
    const unsigned short* __restrict__ q,        // (B, Hq, Tq, D)
    const unsigned short* __restrict__ k_pages,  // (np, Hkv, P, D)
    const unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table,          // (B, maxp)
    const int* __restrict__ q_start,             // (B,)
    const float* __restrict__ alibi,             // (Hq,) slopes or null
    const unsigned char* __restrict__ tree_mask, // (B, Tq, Tq) bool or null:
                                                 // spec-verify ancestor mask
                                                 // over the NEW tokens (ref
                                                 // backend.py:944-1047);
                                                 // committed positions < qs
                                                 // stay fully visible
    unsigned short* __restrict__ out,            // (B, Hq, Tq, D)
    int B, int Hq, int G, int Tq, int P, int maxp, int window, float scale,
    long q_sb, long q_st, long q_sh, long o_sb, long o_st, long o_sh) {
  // q element (b, h, row, d) at b*q_sb + row*q_st + h*q_sh + d (same for
  // out) — reads the q section of a fused (B, T, X, D) QKV output directly.
  constexpr int KROW_B = D * 2 + 16;  // K LDS row stride (pad kills conflicts)
  constexpr int VROW_B = KVBLK * 2 + 16;  // V^T LDS row stride (80 B)
  constexpr int PROW_B = KVBLK * 2 + 16;
  constexpr int NKK = D / 32;   // QK^T k-slices
  constexpr int NDT = D / 16;   // PV d-tiles

  __shared__ __attribute__((aligned(16))) unsigned char lds[
      KVBLK * KROW_B + D * VROW_B + 8 * 16 * PROW_B];
  unsigned short* k_lds = (unsigned short*)lds;
  unsigned short* v_lds = (unsigned short*)(lds + KVBLK * KROW_B);
  unsigned short* p_lds = (unsigned short*)(lds + KVBLK * KROW_B + D * VROW_B);

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq, h = bh % Hq;
  const int kvh = h / G;
  const int qbase = qtile * 128;
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int li = lane & 15;      // fragment row/col index
  const int hi = lane >> 4;      // fragment k-quad
  const int qs = q_start[b];
  const int ctx = qs + Tq;
  const float sc2 = scale * LOG2E;
  const float aslope = (alibi != nullptr) ? alibi[h] * LOG2E : 0.f;

  // ---- Q fragments (load once): row = li, k = hi*8 + j (+32*kk) ----
  const int my_qrow = min(qbase + wave * 16 + li, Tq - 1);
  bf16x8 qfrag[NKK];
#pragma unroll
  for (int kk = 0; kk < NKK; ++kk) {
    qfrag[kk] = as_bf16x8(*reinterpret_cast<const short8*>(
        q + (long)b * q_sb + (long)my_qrow * q_st + (long)h * q_sh + hi * 8 + 32 * kk));
  }

  // online-softmax state: this lane's 4 C-rows (r = hi*4 + reg)
  float m2[4], l[4];
  f32x4 acc_o[NDT];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m2[r] = NEG_BIG; l[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < NDT; ++n) acc_o[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // key range this q-tile can see: [lo_min, kmax). Tree queries are not
  // causal among themselves — every node may see any masked-visible node.
  const int qpos_max = qs + min(qbase + 127, Tq - 1);
  const int kmax = (tree_mask != nullptr) ? ctx : min(ctx, qpos_max + 1);
  int kstart = 0;
  if (window > 0) kstart = max(0, ((qs + qbase) - window + 1) / KVBLK * KVBLK);

  for (int kbase = kstart; kbase < kmax; kbase += KVBLK) {
    __syncthreads();  // everyone done reading previous tile
    // ---- stage K tile (row-major, padded) ----
    for (int idx = tid; idx < KVBLK * (D / 8); idx += 512) {
      const int t = idx / (D / 8);
      const int d8 = (idx % (D / 8)) * 8;
      const int kpos = kbase + t;
      short8 kv{};
      if (kpos < ctx) {
        const int page = page_table[b * maxp + kpos / P];
        const long off =
            (((long)page * (long)(Hq / G) + kvh) * P + (kpos % P)) * D + d8;
        kv = *reinterpret_cast<const short8*>(k_pages + off);
      }
      *reinterpret_cast<short8*>((unsigned char*)k_lds + t * KROW_B + d8 * 2) = kv;
    }
    // ---- stage V tile: pool is d-major (np, Hkv, D, P), so an 8-position
    // piece at fixed d is one contiguous 16 B load AND one contiguous 16 B
    // LDS store (kbase is KVBLK-aligned => each octet sits in one page) ----
    for (int idx = tid; idx < D * (KVBLK / 8); idx += 512) {
      const int d = idx / (KVBLK / 8);
      const int o8 = (idx % (KVBLK / 8)) * 8;
      const int kpos = kbase + o8;
      short8 vv{};
      if (kpos < ctx) {
        const int page = page_table[b * maxp + kpos / P];
        vv = *reinterpret_cast<const short8*>(
            v_pages + (((long)page * (long)(Hq / G) + kvh) * D + d) * P +
            (kpos % P));
      }
      *reinterpret_cast<short8*>((unsigned char*)v_lds + d * VROW_B + o8 * 2) = vv;
    }
    __syncthreads();

    // ---- S = Q K^T over this tile: 2 n-tiles of 16 keys ----
    f32x4 s[KVBLK / 16];
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n) {
      s[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        bf16x8 bfrag = as_bf16x8(*reinterpret_cast<const short8*>(
            (unsigned char*)k_lds + (li + 16 * n) * KROW_B + hi * 16 + 64 * kk));
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], bfrag, s[n], 0, 0, 0);
      }
    }

    // ---- mask + online softmax (C layout: row=hi*4+reg, col=li+16n) ----
    float p[KVBLK / 16][4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int qrow = qbase + wave * 16 + hi * 4 + reg;
      const int qpos = qs + qrow;
      float rm = NEG_BIG;
#pragma unroll
      for (int n = 0; n < KVBLK / 16; ++n) {
        const int kpos = kbase + li + 16 * n;
        float sv = s[n][reg] * sc2 + aslope * (float)kpos;
        bool dead = (qrow >= Tq) | (kpos >= ctx);
        if (!dead) {
          if (tree_mask != nullptr) {
            if (kpos >= qs)
              dead = !tree_mask[((long)b * Tq + qrow) * Tq + (kpos - qs)];
          } else {
            dead = (kpos > qpos) |
                   (window > 0 && kpos <= qpos - window);
          }
        }
        sv = dead ? NEG_BIG : sv;
        p[n][reg] = sv;
        rm = fmaxf(rm, sv);
      }
      rm = group16_reduce_max(rm);
      const float mn = fmaxf(m2[reg], rm);
      const float corr = fast_exp2(m2[reg] - mn);
      float psum = 0.f;
#pragma unroll
      for (int n = 0; n < KVBLK / 16; ++n) {
        p[n][reg] = fast_exp2(p[n][reg] - mn);
        psum += p[n][reg];
      }
      psum = group16_reduce_sum(psum);
      l[reg] = l[reg] * corr + psum;
      m2[reg] = mn;
#pragma unroll
      for (int n = 0; n < NDT; ++n) acc_o[n][reg] *= corr;
    }

    // ---- P -> LDS (C layout scatter), read back as A fragments ----
    unsigned char* pw = (unsigned char*)p_lds + wave * 16 * PROW_B;
#pragma unroll
    for (int n = 0; n < KVBLK / 16; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        *(unsigned short*)(pw + (hi * 4 + reg) * PROW_B + (li + 16 * n) * 2) =
            f2bf(p[n][reg]);
    // wave-local LDS; per-wave ds ordering is enforced by lgkmcnt waits

    bf16x8 pfrag = as_bf16x8(
        *reinterpret_cast<const short8*>(pw + li * PROW_B + hi * 16));

    // ---- acc_o += P V : NDT n-tiles of 16 d-cols ----
#pragma unroll
    for (int n = 0; n < NDT; ++n) {
      bf16x8 vfrag = as_bf16x8(*reinterpret_cast<const short8*>(
          (unsigned char*)v_lds + (li + 16 * n) * VROW_B + hi * 16));
      acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, acc_o[n], 0, 0, 0);
    }
  }

  // ---- epilogue: normalize and store ----
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int qrow = qbase + wave * 16 + hi * 4 + reg;
    if (qrow >= Tq) continue;
    const float inv = (l[reg] > 0.f) ? 1.f / l[reg] : 0.f;
#pragma unroll
    for (int n = 0; n < NDT; ++n) {
      out[(long)b * o_sb + (long)qrow * o_st + (long)h * o_sh + li + 16 * n] =
          f2bf(acc_o[n][reg] * inv);
    }
  }
}
