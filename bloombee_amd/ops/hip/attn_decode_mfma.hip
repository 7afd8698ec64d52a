// MFMA-based single-token paged attention for gfx950.
//
// S^T formulation: scores are computed TRANSPOSED — the MFMA A-operand is K
// (rows = positions) and the B-operand is Q (cols = the G query heads of one
// (b, kv_head) GQA group), so C = K Q^T has rows = positions, cols = heads.
// The payoff over the straight S = Q K^T layout is the online softmax: the
// reduction runs over POSITIONS, which now live in each lane's own C regs
// (2 tiles x 4 regs) plus the 4 hi-subgroups — 4 cross-lane ops per 32-pos
// tile instead of 32, and the running (m, l) are per-head SCALARS held in
// the head's lane column instead of 4-reg arrays. P.V then runs as
// O^T = V^T P^T (A = V^T, B = P^T), whose C is (d rows, head cols).
// Heads beyond the group width are garbage COLUMNS that pollute only
// garbage output columns.
//
// BOTH operand streams read straight from the paged pool — no LDS staging:
//  * K pages are token-major (np, Hkv, P, D): an A-fragment of K (8
//    consecutive d at fixed position) is a contiguous 16 B load.
//  * V pages are d-major (np, Hkv, D, P): an A-fragment of V^T (8
//    consecutive positions at fixed d) is a contiguous 16 B load. The
//    transpose happened once at kv-write time.
// LDS is only used for the P^T C->B relayout (tiny) and the final 4-wave
// merge, so occupancy is VGPR-limited, not LDS-limited.
//
// Work decomposition: grid = (B*Hkv*nch, n_split) flash-decode splits, 4
// waves per workgroup; wave w owns KV tiles (32 positions) w, w+4, w+8, ...
// of its split, with a final 4-way LDS merge of (m, l, acc). V fragment
// loads are issued BEFORE the QK^T/softmax phase so their HBM latency is
// covered (measured: moving them later costs ~35%).

#include "common.h"

static constexpr int DKVBLK = 32;  // positions per KV tile

typedef __attribute__((ext_vector_type(4))) short short4v;  // 4 x bf16 = 8 B

// Reduce across the four 16-lane subgroups: lanes {l, l+16, l+32, l+48}.
DEVINL float quad16_reduce_max(float x) {
  x = fmaxf(x, __shfl_xor(x, 16));
  return fmaxf(x, __shfl_xor(x, 32));
}
DEVINL float quad16_reduce_sum(float x) {
  x += __shfl_xor(x, 16);
  return x + __shfl_xor(x, 32);
}

// nch: GQA group chunks per kv head (ceil(G/16)) — MQA groups wider than the
// 16-col MFMA q-tile (falcon-7b G=71) split into chunks sharing the kv head.
// alibi: per-query-head slopes (bloom family), or null.
// The S^T form compiles to 118 VGPRs / 0 spills -> 4 waves/SIMD resident
// (the S-form needed 208 VGPR + 32 AGPR and sat at 2); launch_bounds keeps
// the floor at 2 WGs/CU. Host-side n_split sizes the grid for ~1024
// workgroups = 4 resident WGs on all 256 CUs (profiles/r01_st_attention.md).
// NT: KV tiles (32 positions each) processed per wave iteration. NT=2
// doubles the K/V bytes in flight per wave — the PMC diagnosis at 4
// waves/SIMD was latency-bound (SQ_WAIT ~70%, profiles/r01_st_attention.md)
// and more outstanding loads is the lever; it also halves the softmax
// shuffle + LDS-relayout overhead per position. Costs ~40 VGPRs (V
// fragments + score regs for the second tile); host falls back to NT=1 for
// short splits.
// PIPE=1 (NT must be 1): register double-buffered software pipeline — the
// next tile's K AND V loads are issued before the current tile's MFMA/
// softmax consume their buffers, so a full iteration of compute covers each
// load's latency. The r02 PMC diagnosis (WAIT_ANY 82%, ACTIVE 12%, no DRAM
// credit stalls, and n_split>0 makes things WORSE) shows the kernel is
// per-wave-latency-bound, not occupancy- or bandwidth-bound: the serial
// K-wait -> softmax -> PV chain leaves ~2.5 us/iter of HBM idle per wave.
// Costs ~96 VGPRs of double buffers — free at this grid (B*Hkv WGs = 1
// WG/CU: occupancy is grid-bound, not VGPR-bound).
template <int D, int MAXG, int NT, int PIPE = 0>
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
  constexpr int NPT = DKVBLK / 16;  // position tiles per KV tile
  constexpr int PROW_B = NT * DKVBLK * 2 + 16;  // P^T LDS row pitch

  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / (Hkv * nch);
  const int rem = bh % (Hkv * nch);
  const int kvh = rem / nch;
  const int g0 = (rem % nch) * MAXG;   // this chunk's first group col
  const int Gl = min(G - g0, MAXG);    // live head cols in this chunk
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

  // per-wave LDS: P^T tile (C->B relayout, head-major rows) + merge scratch
  __shared__ __attribute__((aligned(16))) unsigned char lds[
      4 * 16 * PROW_B + 3 * (MAXG * (D + 2) * 4)];
  unsigned char* p_lds = lds + wave * 16 * PROW_B;
  float* merge_lds = (float*)(lds + 4 * 16 * PROW_B);

  // Q fragments: B[k=d][col=head]; col = li (cols >= Gl harmless garbage)
  const int qh = g0 + ((li < Gl) ? li : Gl - 1);
  bf16x8 qfrag[NKK];
#pragma unroll
  for (int kk = 0; kk < NKK; ++kk)
    qfrag[kk] = as_bf16x8(*reinterpret_cast<const short8*>(
        q + (long)b * q_sb + (long)(kvh * G + qh) * q_sh + hi * 8 + 32 * kk));
  // per-head alibi slope (head col = li)
  const int ag = g0 + li;
  const float aslope =
      (alibi != nullptr && ag < G) ? alibi[kvh * G + ag] * LOG2E : 0.f;

  // online softmax state: PER-HEAD scalars (head = this lane's col li)
  float m2 = NEG_BIG, l = 0.f;
  f32x4 acc_o[NDT];  // O^T: rows = d (hi*4+reg + 16n), cols = head (li)
#pragma unroll
  for (int n = 0; n < NDT; ++n) acc_o[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const long head_slab_k = (long)kvh * P * D;  // same bytes, both layouts
  const int tile0 = (max(c0, lo) - c0) / DKVBLK;  // window skip, tile-aligned

  // ---------------------------------------------------------------------
  // PIPE=1: statically-unrolled register double buffer (two tiles per loop
  // iteration). Dynamic buffer indexing (buf[cur]) demotes the arrays to
  // scratch memory on amdgcn — the buffers must be distinct named arrays
  // with compile-time indices, and sched_barrier pins each prefetch above
  // the compute phase it overlaps (the scheduler otherwise sinks loads to
  // their uses and the pipeline degenerates to the serial form).
  constexpr int TSTRIDE = 4 * NT * DKVBLK;
  const int tbs = c0 + (tile0 + wave * NT) * DKVBLK;

  if constexpr (PIPE) {
    static_assert(NT == 1, "PIPE implies NT==1");
    bf16x8 vb0[NDT], vb1[NDT];
    bf16x8 kb0[NPT][NKK], kb1[NPT][NKK];
    auto load_tile = [&](bf16x8 (&vb)[NDT], bf16x8 (&kb)[NPT][NKK], int tb) {
      const int vpos = tb + hi * 8;
      int vpage = -1;
      if (vpos < c1) vpage = page_table[b * maxp + vpos / P];
      const long vrow = ((long)vpage * Hkv + kvh) * D;
#pragma unroll
      for (int n = 0; n < NDT; ++n) {
        vb[n] = as_bf16x8(short8{});
        if (vpage >= 0)
          vb[n] = as_bf16x8(__builtin_nontemporal_load(
              reinterpret_cast<const short8*>(
                  v_pages + (vrow + li + 16 * n) * P + (vpos % P))));
      }
#pragma unroll
      for (int n = 0; n < NPT; ++n) {
        const int pos = tb + li + 16 * n;
        const int cpos = min(pos, c1 - 1);
        const int page = page_table[b * maxp + cpos / P];
        const unsigned short* krow = k_pages + ((long)page * Hkv) * P * D +
                                     head_slab_k + (long)(cpos % P) * D;
#pragma unroll
        for (int kk = 0; kk < NKK; ++kk)
          kb[n][kk] = as_bf16x8(__builtin_nontemporal_load(
              reinterpret_cast<const short8*>(krow + hi * 8 + 32 * kk)));
      }
    };
    auto compute_tile = [&](const bf16x8 (&vb)[NDT],
                            const bf16x8 (&kb)[NPT][NKK], int tb) {
      f32x4 st[NPT];
#pragma unroll
      for (int n = 0; n < NPT; ++n) {
        st[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < NKK; ++kk)
          st[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kb[n][kk], qfrag[kk], st[n], 0, 0, 0);
      }
      float p[NPT][4];
      float rm = NEG_BIG;
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int kpos = tb + 16 * n + hi * 4 + reg;
          float sv = st[n][reg] * sc2 + aslope * (float)kpos;
          const bool dead = (kpos >= c1) | (kpos < lo);
          sv = dead ? NEG_BIG : sv;
          p[n][reg] = sv;
          rm = fmaxf(rm, sv);
        }
      rm = quad16_reduce_max(rm);
      const float mn = fmaxf(m2, rm);
      const float corr = fast_exp2(m2 - mn);
      float psum = 0.f;
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          p[n][reg] = fast_exp2(p[n][reg] - mn);
          psum += p[n][reg];
        }
      psum = quad16_reduce_sum(psum);
      l = l * corr + psum;
      m2 = mn;
      if (__builtin_amdgcn_ballot_w64(corr < 0.9999f)) {
#pragma unroll
        for (int n = 0; n < NDT; ++n)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) acc_o[n][reg] *= corr;
      }
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          *(unsigned short*)(p_lds + li * PROW_B +
                             (16 * n + hi * 4 + reg) * 2) = f2bf(p[n][reg]);
      bf16x8 pfrag = as_bf16x8(
          *reinterpret_cast<const short8*>(p_lds + li * PROW_B + hi * 16));
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vb[n], pfrag,
                                                           acc_o[n], 0, 0, 0);
    };
    if (tbs < c1) load_tile(vb0, kb0, tbs);
    for (int tb = tbs; tb < c1; tb += 2 * TSTRIDE) {
      if (tb + TSTRIDE < c1) load_tile(vb1, kb1, tb + TSTRIDE);
      __builtin_amdgcn_sched_barrier(0);
      compute_tile(vb0, kb0, tb);
      if (tb + TSTRIDE >= c1) break;
      if (tb + 2 * TSTRIDE < c1) load_tile(vb0, kb0, tb + 2 * TSTRIDE);
      __builtin_amdgcn_sched_barrier(0);
      compute_tile(vb1, kb1, tb + TSTRIDE);
    }
  } else {
  // wave w owns NT adjacent tiles starting at tile w*NT, stride 4*NT tiles
  for (int tb = tbs; tb < c1; tb += TSTRIDE) {
    // ---- V^T fragments for ALL NT tiles: direct A-layout loads, issued
    // FIRST so the QK^T phase covers their latency. A[row=d][k=pos]: lane
    // (li -> d row, hi -> position octet); 8 consecutive positions at fixed
    // d are contiguous in the d-major pool. tb is 32-aligned and P | 32, so
    // each octet sits in one page.
    bf16x8 vfrag[NT][NDT];
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      const int vpos = tb + t * DKVBLK + hi * 8;
      int vpage = -1;
      if (vpos < c1) vpage = page_table[b * maxp + vpos / P];
      const long vrow = ((long)vpage * Hkv + kvh) * D;
#pragma unroll
      for (int n = 0; n < NDT; ++n) {
        vfrag[t][n] = as_bf16x8(short8{});
        if (vpage >= 0)
          vfrag[t][n] = as_bf16x8(__builtin_nontemporal_load(
              reinterpret_cast<const short8*>(
                  v_pages + (vrow + li + 16 * n) * P + (vpos % P))));
      }
    }

    // ---- S^T = K Q^T : K A-fragments straight from the paged pool ----
    // A[row=pos][k=d]: lane li -> position row, hi -> d octet.
    f32x4 st[NT][NPT];
#pragma unroll
    for (int t = 0; t < NT; ++t)
#pragma unroll
      for (int n = 0; n < NPT; ++n) {
        const int pos = tb + t * DKVBLK + li + 16 * n;
        const int cpos = min(pos, c1 - 1);
        const int page = page_table[b * maxp + cpos / P];
        const unsigned short* krow = k_pages + ((long)page * Hkv) * P * D +
                                     head_slab_k + (long)(cpos % P) * D;
        st[t][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < NKK; ++kk) {
          // KV is read once per decode step: nontemporal keeps L2 for the
          // GEMM weight streams that follow in the same step
          bf16x8 kfrag = as_bf16x8(__builtin_nontemporal_load(
              reinterpret_cast<const short8*>(krow + hi * 8 + 32 * kk)));
          st[t][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kfrag, qfrag[kk], st[t][n], 0, 0, 0);
        }
      }

    // ---- mask + online softmax over positions (all in-lane + 2 shuffles).
    // st[t][n][reg] = score at tb + t*32 + 16n + hi*4 + reg for head li.
    float p[NT][NPT][4];
    float rm = NEG_BIG;
#pragma unroll
    for (int t = 0; t < NT; ++t)
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int kpos = tb + t * DKVBLK + 16 * n + hi * 4 + reg;
          float sv = st[t][n][reg] * sc2 + aslope * (float)kpos;
          const bool dead = (kpos >= c1) | (kpos < lo);
          sv = dead ? NEG_BIG : sv;
          p[t][n][reg] = sv;
          rm = fmaxf(rm, sv);
        }
    rm = quad16_reduce_max(rm);
    const float mn = fmaxf(m2, rm);
    const float corr = fast_exp2(m2 - mn);
    float psum = 0.f;
#pragma unroll
    for (int t = 0; t < NT; ++t)
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          p[t][n][reg] = fast_exp2(p[t][n][reg] - mn);
          psum += p[t][n][reg];
        }
    psum = quad16_reduce_sum(psum);
    l = l * corr + psum;
    m2 = mn;
    // steady-state decode rarely raises the max after the first tiles:
    // skip the NDT-wide rescale when no lane needs it (wave-uniform branch)
    if (__builtin_amdgcn_ballot_w64(corr < 0.9999f)) {
#pragma unroll
      for (int n = 0; n < NDT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) acc_o[n][reg] *= corr;
    }

    // ---- P^T through LDS into B-fragment layout (head-major rows, so the
    // B-frag load of 8 consecutive positions at head li is one 16 B read)
#pragma unroll
    for (int t = 0; t < NT; ++t)
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          *(unsigned short*)(p_lds + li * PROW_B +
                             (t * DKVBLK + 16 * n + hi * 4 + reg) * 2) =
              f2bf(p[t][n][reg]);

    // ---- acc^T += V^T P^T (A-fragments were loaded at tile start) ----
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      bf16x8 pfrag = as_bf16x8(*reinterpret_cast<const short8*>(
          p_lds + li * PROW_B + t * DKVBLK * 2 + hi * 16));
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[t][n], pfrag,
                                                           acc_o[n], 0, 0, 0);
    }
  }
  }

  // ---- merge the 4 waves ----
  // C col li holds chunk-local head g = li (valid for g < Gl); C row
  // hi*4 + reg of tile n is dimension d = 16n + hi*4 + reg. (m, l) are
  // replicated across the 4 hi subgroups — hi == 0 publishes them.
  auto wave_slot = [&](int slot) { return merge_lds + slot * MAXG * (D + 2); };
  __syncthreads();
  if (wave > 0) {
    float* dst = wave_slot(wave - 1);
    if (li < Gl) {
#pragma unroll
      for (int n = 0; n < NDT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          dst[li * (D + 2) + 16 * n + hi * 4 + reg] = acc_o[n][reg];
      if (hi == 0) {
        dst[li * (D + 2) + D] = m2;
        dst[li * (D + 2) + D + 1] = l;
      }
    }
  }
  __syncthreads();
  if (wave != 0) return;
  for (int slot = 0; slot < 3; ++slot) {
    const float* src = wave_slot(slot);
    if (li < Gl) {
      const float mo = src[li * (D + 2) + D];
      const float lo2 = src[li * (D + 2) + D + 1];
      const float mn = fmaxf(m2, mo);
      const float c1f = fast_exp2(m2 - mn);
      const float c2f = fast_exp2(mo - mn);
      l = l * c1f + lo2 * c2f;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          acc_o[n][reg] = acc_o[n][reg] * c1f +
                          src[li * (D + 2) + 16 * n + hi * 4 + reg] * c2f;
      m2 = mn;
    }
  }

  if (n_split == 1) {
    if (li < Gl) {
      const float inv = (l > 0.f) ? 1.f / l : 0.f;
      unsigned short* orow =
          out + (long)b * out_sb + (long)(kvh * G + g0 + li) * out_sh;
#pragma unroll
      for (int n = 0; n < NDT; ++n) {
        // 4 consecutive d per lane -> one 8 B store
        short4v o4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) o4[reg] = f2bf(acc_o[n][reg] * inv);
        *reinterpret_cast<short4v*>(orow + 16 * n + hi * 4) = o4;
      }
    }
  } else {
    const long pbase = ((long)bh * n_split + split);
    if (li < Gl) {
      float* prow = part_acc + (pbase * MAXG + li) * D;
#pragma unroll
      for (int n = 0; n < NDT; ++n)
        *reinterpret_cast<f32x4*>(prow + 16 * n + hi * 4) = acc_o[n];
      if (hi == 0) {
        part_ml[(pbase * MAXG + li) * 2 + 0] = m2;
        part_ml[(pbase * MAXG + li) * 2 + 1] = l;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Wide-head decode attention (D = 512: gemma-4 global layers).
//
// Same S^T formulation, but the head dim is split ACROSS the 4 waves instead
// of splitting positions: wave w owns d-slice [128w, 128w+128). All waves
// walk the SAME 32-position KV tile per iteration:
//   1. wave w computes partial scores over its d-slice (4 MFMAs per 16-pos
//      subtile) and publishes its C regs to LDS;
//   2. after a barrier every wave sums the four partials IN THE SAME ORDER
//      (bitwise-identical scores -> identical m/l/corr in all waves, so the
//      per-wave normalization of the d-partitioned output is consistent);
//   3. wave 0 writes P^T to LDS; every wave PV-accumulates its own 8 d-tiles
//      (32 acc VGPRs instead of the 128 a monolithic D=512 tile would need).
// No output merge: d-slices are disjoint. KV bytes read once, like D<=256.
// ---------------------------------------------------------------------------

template <int MAXG>
__global__ __launch_bounds__(256, 2) void attn_decode_wide_kernel(
    const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k_pages,  // (np, Hkv, P, 512)
    const unsigned short* __restrict__ v_pages,  // (np, Hkv, 512, P) d-major
    const int* __restrict__ page_table, const int* __restrict__ ctx_lens,
    const float* __restrict__ alibi, unsigned short* __restrict__ out,
    float* __restrict__ part_ml, float* __restrict__ part_acc,
    int B, int Hkv, int G, int nch, int P, int maxp, int n_split, int window,
    float scale, long q_sb, long q_sh, long out_sb, long out_sh) {
  constexpr int D = 512;
  constexpr int NKK_W = 4;          // QK k-slices per wave (128 d)
  constexpr int NDT_W = 8;          // PV d-tiles per wave (128 d)
  constexpr int NPT = DKVBLK / 16;  // position subtiles
  constexpr int PROW_B = DKVBLK * 2 + 16;

  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / (Hkv * nch);
  const int rem = bh % (Hkv * nch);
  const int kvh = rem / nch;
  const int g0 = (rem % nch) * MAXG;
  const int Gl = min(G - g0, MAXG);
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
  const int d0 = wave * 128;        // this wave's d-slice
  const float sc2 = scale * LOG2E;

  // LDS: per-wave partial-score slots (4 x 64 lanes x 8 floats) + one shared
  // P^T tile (16 head rows x 32 positions, bf16)
  __shared__ __attribute__((aligned(16))) float sc_lds[4][WAVE][NPT * 4];
  __shared__ __attribute__((aligned(16))) unsigned char p_lds[16 * PROW_B];

  const int qh = g0 + ((li < Gl) ? li : Gl - 1);
  bf16x8 qfrag[NKK_W];
#pragma unroll
  for (int kk = 0; kk < NKK_W; ++kk)
    qfrag[kk] = as_bf16x8(*reinterpret_cast<const short8*>(
        q + (long)b * q_sb + (long)(kvh * G + qh) * q_sh + d0 + hi * 8 +
        32 * kk));
  const int ag = g0 + li;
  const float aslope =
      (alibi != nullptr && ag < G) ? alibi[kvh * G + ag] * LOG2E : 0.f;

  float m2 = NEG_BIG, l = 0.f;
  f32x4 acc_o[NDT_W];
#pragma unroll
  for (int n = 0; n < NDT_W; ++n) acc_o[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const long head_slab_k = (long)kvh * P * D;
  const int tile0 = (max(c0, lo) - c0) / DKVBLK;

  for (int tb = c0 + tile0 * DKVBLK; tb < c1; tb += DKVBLK) {
    // V^T fragments for this wave's d-slice (issued first, latency cover)
    const int vpos = tb + hi * 8;
    int vpage = -1;
    if (vpos < c1) vpage = page_table[b * maxp + vpos / P];
    const long vrow = ((long)vpage * Hkv + kvh) * D + d0;
    bf16x8 vfrag[NDT_W];
#pragma unroll
    for (int n = 0; n < NDT_W; ++n) {
      vfrag[n] = as_bf16x8(short8{});
      if (vpage >= 0)
        vfrag[n] = as_bf16x8(__builtin_nontemporal_load(
            reinterpret_cast<const short8*>(
                v_pages + (vrow + li + 16 * n) * P + (vpos % P))));
    }

    // partial S^T over this wave's 128 d
    f32x4 st[NPT];
#pragma unroll
    for (int n = 0; n < NPT; ++n) {
      const int pos = tb + li + 16 * n;
      const int cpos = min(pos, c1 - 1);
      const int page = page_table[b * maxp + cpos / P];
      const unsigned short* krow = k_pages + ((long)page * Hkv) * P * D +
                                   head_slab_k + (long)(cpos % P) * D + d0;
      st[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < NKK_W; ++kk) {
        bf16x8 kfrag = as_bf16x8(__builtin_nontemporal_load(
            reinterpret_cast<const short8*>(krow + hi * 8 + 32 * kk)));
        st[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag, qfrag[kk],
                                                        st[n], 0, 0, 0);
      }
    }
    __syncthreads();  // previous iteration's P^T reads are done
#pragma unroll
    for (int n = 0; n < NPT; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        sc_lds[wave][lane][n * 4 + reg] = st[n][reg];
    __syncthreads();
    // full scores: fixed summation order -> bitwise identical in all waves
    float p[NPT][4];
    float rm = NEG_BIG;
#pragma unroll
    for (int n = 0; n < NPT; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float sv = sc_lds[0][lane][n * 4 + reg];
#pragma unroll
        for (int w = 1; w < 4; ++w) sv += sc_lds[w][lane][n * 4 + reg];
        const int kpos = tb + 16 * n + hi * 4 + reg;
        sv = sv * sc2 + aslope * (float)kpos;
        const bool dead = (kpos >= c1) | (kpos < lo);
        sv = dead ? NEG_BIG : sv;
        p[n][reg] = sv;
        rm = fmaxf(rm, sv);
      }
    rm = quad16_reduce_max(rm);
    const float mn = fmaxf(m2, rm);
    const float corr = fast_exp2(m2 - mn);
    float psum = 0.f;
#pragma unroll
    for (int n = 0; n < NPT; ++n)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        p[n][reg] = fast_exp2(p[n][reg] - mn);
        psum += p[n][reg];
      }
    psum = quad16_reduce_sum(psum);
    l = l * corr + psum;
    m2 = mn;
    if (__builtin_amdgcn_ballot_w64(corr < 0.9999f)) {
#pragma unroll
      for (int n = 0; n < NDT_W; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) acc_o[n][reg] *= corr;
    }

    // one shared P^T tile (identical p in every wave; wave 0 publishes)
    if (wave == 0) {
#pragma unroll
      for (int n = 0; n < NPT; ++n)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          *(unsigned short*)(p_lds + li * PROW_B +
                             (16 * n + hi * 4 + reg) * 2) = f2bf(p[n][reg]);
    }
    __syncthreads();
    bf16x8 pfrag = as_bf16x8(
        *reinterpret_cast<const short8*>(p_lds + li * PROW_B + hi * 16));
#pragma unroll
    for (int n = 0; n < NDT_W; ++n)
      acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[n], pfrag,
                                                         acc_o[n], 0, 0, 0);
  }

  // d-partitioned output: every wave writes its own 128-d slice; (m, l) are
  // identical across waves by construction
  if (n_split == 1) {
    if (li < Gl) {
      const float inv = (l > 0.f) ? 1.f / l : 0.f;
      unsigned short* orow =
          out + (long)b * out_sb + (long)(kvh * G + g0 + li) * out_sh + d0;
#pragma unroll
      for (int n = 0; n < NDT_W; ++n) {
        short4v o4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) o4[reg] = f2bf(acc_o[n][reg] * inv);
        *reinterpret_cast<short4v*>(orow + 16 * n + hi * 4) = o4;
      }
    }
  } else {
    const long pbase = ((long)bh * n_split + split);
    if (li < Gl) {
      float* prow = part_acc + (pbase * MAXG + li) * D + d0;
#pragma unroll
      for (int n = 0; n < NDT_W; ++n)
        *reinterpret_cast<f32x4*>(prow + 16 * n + hi * 4) = acc_o[n];
      if (hi == 0 && wave == 0) {
        part_ml[(pbase * MAXG + li) * 2 + 0] = m2;
        part_ml[(pbase * MAXG + li) * 2 + 1] = l;
      }
    }
  }
}
