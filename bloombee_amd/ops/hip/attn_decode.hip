// Single-token (decode) paged attention for gfx950 — flash-decoding style.
//
// Replaces the reference's mha_gen_llama decode branch
// (flexgen_utils/pytorch_backend.py:733-916): one fused kernel reads the
// paged KV cache in place (no gather, no (B,H,S,D) view materialization).
//
// Decode attention at Tq=1 is HBM-bound (guide Appendix B): the job is to
// stream each (batch, kv_head)'s K and V exactly once at full bandwidth.
//
// Decomposition: grid = (B * Hkv, n_split). Each 4-wave workgroup owns a
// contiguous page-aligned chunk of the context for one (b, kv_head) and all
// G = Hq/Hkv query heads of the group (GQA). n_split is chosen on the host so
// B*Hkv*n_split well exceeds the 256 CUs (8 XCDs need >>256 workgroups).
//
// Within a workgroup: 16 independent "lane groups" (4 waves x 4 groups of 16
// lanes). Lane group g processes positions c0 + g, c0 + g + 16, ... Each of
// its 16 lanes loads E = D/16 bf16 of the position's K row — a wave's 4
// groups touch 4 consecutive positions (>= 1 KB contiguous per instruction
// at D=128). Online softmax (m, l, acc[G][E] f32/lane) runs per lane group;
// groups merge via an LDS tree; splits merge in attn_decode_combine.
//
// Numerics: fp32 accumulation throughout, exp2-domain softmax with the scale
// folded in (p = 2^(s*scale*log2e - m2)), matching the reference's fp32
// softmax (pytorch_backend.py:919-934) within bf16 rounding.

#include "common.h"

typedef __attribute__((ext_vector_type(4))) short short4_v;

template <int E>
DEVINL void load_bf16_e(const unsigned short* p, float* out) {
  if constexpr (E == 4) {
    short4_v v = *reinterpret_cast<const short4_v*>(p);
#pragma unroll
    for (int j = 0; j < 4; ++j) out[j] = bf2f((unsigned short)v[j]);
  } else {
#pragma unroll
    for (int c = 0; c < E / 8; ++c) {
      short8 v = *reinterpret_cast<const short8*>(p + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) out[c * 8 + j] = bf2f((unsigned short)v[j]);
    }
  }
}

template <int E>
DEVINL void store_bf16_e(unsigned short* p, const float* in) {
  if constexpr (E == 4) {
    short4_v v;
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = (short)f2bf(in[j]);
    *reinterpret_cast<short4_v*>(p) = v;
  } else {
#pragma unroll
    for (int c = 0; c < E / 8; ++c) {
      short8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (short)f2bf(in[c * 8 + j]);
      *reinterpret_cast<short8*>(p + c * 8) = v;
    }
  }
}

template <int D, int MAXG>
__global__ __launch_bounds__(256) void attn_decode_kernel(
    const unsigned short* __restrict__ q,        // (B, Hq, D)
    const unsigned short* __restrict__ k_pages,  // (np, Hkv, P, D)
    const unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table,          // (B, maxp)
    const int* __restrict__ ctx_lens,            // (B,)
    unsigned short* __restrict__ out,            // (B, Hq, D)   [n_split==1]
    float* __restrict__ part_ml,                 // (B*Hkv*n_split, G, 2)
    float* __restrict__ part_acc,                // (B*Hkv*n_split, G, D)
    int B, int Hkv, int G, int P, int maxp, int n_split, int window,
    float scale, long q_sb, long q_sh, long out_sb, long out_sh) {
  // q_sb / out_sb: elements between consecutive batch rows of q / out —
  // lets the kernel read the q section of a fused QKV GEMM output directly.
  constexpr int E = D / 16;  // elements per lane
  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / Hkv, kvh = bh % Hkv;
  const int ctx = ctx_lens[b];

  // page-aligned chunk [c0, c1) of this split
  const int pages_total = (ctx + P - 1) / P;
  const int pages_per_split = (pages_total + n_split - 1) / n_split;
  int c0 = split * pages_per_split * P;
  int c1 = min(ctx, c0 + pages_per_split * P);
  int lo = 0;
  if (window > 0) lo = max(0, ctx - window);  // sliding-window families

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int lg = lane / 16;            // lane group within wave
  const int li = lane & 15;            // lane within group
  const int group_id = wave * 4 + lg;  // 0..15

  const float sc2 = scale * LOG2E;

  // q fragments: lane li holds q[g][li*E .. li*E+E)
  float qv[MAXG][E];
#pragma unroll
  for (int g = 0; g < MAXG; ++g) {
    if (g < G)
      load_bf16_e<E>(q + (long)b * q_sb + (kvh * G + g) * q_sh + li * E, qv[g]);
  }

  float m2[MAXG], l[MAXG], acc[MAXG][E];
#pragma unroll
  for (int g = 0; g < MAXG; ++g) {
    m2[g] = NEG_BIG;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < E; ++j) acc[g][j] = 0.f;
  }

  const long head_slab = (long)kvh * P * D;
  // Unroll the position walk 2-wide AND software-prefetch the next pair:
  // the per-position shfl-reduce + transcendental chain is latency-bound;
  // loads for iteration i+1 issue (raw bf16) before iteration i's compute so
  // ~the whole compute phase covers the HBM latency (guide T14 pattern).
  int pos = c0 + group_id;
  // sliding window: advance in full strides so the 16-group partition of
  // positions is preserved (no double-processing across groups)
  if (pos < lo) pos += ((lo - pos + 15) / 16) * 16;

  auto row_off = [&](int p) {
    const int page = page_table[b * maxp + p / P];
    return ((long)page * Hkv) * P * D + head_slab + (long)(p % P) * D + li * E;
  };
  constexpr int NR = E / 8 > 0 ? E / 8 : 1;  // short8 regs per row slice
  short8 kr_a[NR], vr_a[NR], kr_b[NR], vr_b[NR];
  short8 kn_a[NR], vn_a[NR], kn_b[NR], vn_b[NR];

  auto load_raw = [&](long off, short8* kr, short8* vr) {
#pragma unroll
    for (int c = 0; c < NR; ++c) {
      kr[c] = *reinterpret_cast<const short8*>(k_pages + off + c * 8);
      vr[c] = *reinterpret_cast<const short8*>(v_pages + off + c * 8);
    }
  };
  auto cvt = [&](const short8* r, float* out) {
#pragma unroll
    for (int c = 0; c < NR; ++c)
#pragma unroll
      for (int j = 0; j < 8; ++j) out[c * 8 + j] = bf2f((unsigned short)r[c][j]);
  };

  if constexpr (E >= 8) {
    if (pos + 16 < c1) {
      load_raw(row_off(pos), kr_a, vr_a);
      load_raw(row_off(pos + 16), kr_b, vr_b);
    }
    for (; pos + 16 < c1; pos += 32) {
      const bool have_next = pos + 48 < c1;
      if (have_next) {
        load_raw(row_off(pos + 32), kn_a, vn_a);
        load_raw(row_off(pos + 48), kn_b, vn_b);
      }
      float ka[E], va[E], kb[E], vb[E];
      cvt(kr_a, ka); cvt(kr_b, kb); cvt(vr_a, va); cvt(vr_b, vb);
#pragma unroll
      for (int g = 0; g < MAXG; ++g) {
        if (g >= G) break;
        float da = 0.f, db = 0.f;
#pragma unroll
        for (int j = 0; j < E; ++j) { da += ka[j] * qv[g][j]; db += kb[j] * qv[g][j]; }
#pragma unroll
        for (int m = 1; m < 16; m <<= 1) {
          da += __shfl_xor(da, m);
          db += __shfl_xor(db, m);
        }
        const float sa = da * sc2, sb = db * sc2;
        const float mx = fmaxf(sa, sb);
        if (mx <= m2[g]) {  // fast path: no rescale (T13 class)
          const float pa_ = fast_exp2(sa - m2[g]);
          const float pb_ = fast_exp2(sb - m2[g]);
          l[g] += pa_ + pb_;
#pragma unroll
          for (int j = 0; j < E; ++j) acc[g][j] += pa_ * va[j] + pb_ * vb[j];
        } else {
          const float corr = fast_exp2(m2[g] - mx);
          const float pa_ = fast_exp2(sa - mx);
          const float pb_ = fast_exp2(sb - mx);
          l[g] = l[g] * corr + pa_ + pb_;
#pragma unroll
          for (int j = 0; j < E; ++j)
            acc[g][j] = acc[g][j] * corr + pa_ * va[j] + pb_ * vb[j];
          m2[g] = mx;
        }
      }
      if (have_next) {
#pragma unroll
        for (int c = 0; c < NR; ++c) {
          kr_a[c] = kn_a[c]; vr_a[c] = vn_a[c];
          kr_b[c] = kn_b[c]; vr_b[c] = vn_b[c];
        }
      }
    }
  }
  for (; pos < c1; pos += 16) {  // tail (and sliding-window head skip)
    if (pos < lo) continue;
    const int page = page_table[b * maxp + pos / P];
    const long off =
        ((long)page * Hkv) * P * D + head_slab + (long)(pos % P) * D + li * E;
    float kv[E], vv[E];
    load_bf16_e<E>(k_pages + off, kv);
    load_bf16_e<E>(v_pages + off, vv);
#pragma unroll
    for (int g = 0; g < MAXG; ++g) {
      if (g >= G) break;
      float dot = 0.f;
#pragma unroll
      for (int j = 0; j < E; ++j) dot += kv[j] * qv[g][j];
      dot = group16_reduce_sum(dot);
      const float s = dot * sc2;
      if (s <= m2[g]) {
        const float p = fast_exp2(s - m2[g]);
        l[g] += p;
#pragma unroll
        for (int j = 0; j < E; ++j) acc[g][j] += p * vv[j];
      } else {
        const float corr = fast_exp2(m2[g] - s);
        l[g] = l[g] * corr + 1.f;
#pragma unroll
        for (int j = 0; j < E; ++j) acc[g][j] = acc[g][j] * corr + vv[j];
        m2[g] = s;
      }
    }
  }

  // ---- merge the 16 lane groups (tree over LDS) ----
  __shared__ float s_acc[8][MAXG][D];
  __shared__ float s_ml[8][MAXG][2];
#pragma unroll
  for (int stride = 8; stride >= 1; stride >>= 1) {
    __syncthreads();
    if (group_id >= stride && group_id < 2 * stride) {
      const int slot = group_id - stride;
#pragma unroll
      for (int g = 0; g < MAXG; ++g) {
        if (g >= G) break;
#pragma unroll
        for (int j = 0; j < E; ++j) s_acc[slot][g][li * E + j] = acc[g][j];
        if (li == 0) { s_ml[slot][g][0] = m2[g]; s_ml[slot][g][1] = l[g]; }
      }
    }
    __syncthreads();
    if (group_id < stride) {
#pragma unroll
      for (int g = 0; g < MAXG; ++g) {
        if (g >= G) break;
        const float mo = s_ml[group_id][g][0];
        const float lo2 = s_ml[group_id][g][1];
        const float mn = fmaxf(m2[g], mo);
        const float c1f = fast_exp2(m2[g] - mn);
        const float c2f = fast_exp2(mo - mn);
        l[g] = l[g] * c1f + lo2 * c2f;
#pragma unroll
        for (int j = 0; j < E; ++j)
          acc[g][j] = acc[g][j] * c1f + s_acc[group_id][g][li * E + j] * c2f;
        m2[g] = mn;
      }
    }
  }

  if (group_id != 0) return;
  if (n_split == 1) {
#pragma unroll
    for (int g = 0; g < MAXG; ++g) {
      if (g >= G) break;
      float o[E];
      const float inv = (l[g] > 0.f) ? 1.f / l[g] : 0.f;
#pragma unroll
      for (int j = 0; j < E; ++j) o[j] = acc[g][j] * inv;
      store_bf16_e<E>(out + (long)b * out_sb + (kvh * G + g) * out_sh + li * E, o);
    }
  } else {
    const long pbase = ((long)bh * n_split + split);
#pragma unroll
    for (int g = 0; g < MAXG; ++g) {
      if (g >= G) break;
#pragma unroll
      for (int j = 0; j < E; ++j)
        part_acc[(pbase * G + g) * D + li * E + j] = acc[g][j];
      if (li == 0) {
        part_ml[(pbase * G + g) * 2 + 0] = m2[g];
        part_ml[(pbase * G + g) * 2 + 1] = l[g];
      }
    }
  }
}

// Combine n_split partials. grid = (B*Hkv), block = G*16 (<= 128 threads).
template <int D>
__global__ void attn_decode_combine_kernel(
    const float* __restrict__ part_ml, const float* __restrict__ part_acc,
    unsigned short* __restrict__ out, int Hkv, int G, int n_split,
    long out_sb, long out_sh) {
  constexpr int E = D / 16;
  const int bh = blockIdx.x;
  const int g = threadIdx.x / 16;
  const int li = threadIdx.x & 15;
  if (g >= G) return;
  float m2 = NEG_BIG, l = 0.f, acc[E];
#pragma unroll
  for (int j = 0; j < E; ++j) acc[j] = 0.f;
  for (int s = 0; s < n_split; ++s) {
    const long pbase = (long)bh * n_split + s;
    const float mo = part_ml[(pbase * G + g) * 2 + 0];
    const float lo = part_ml[(pbase * G + g) * 2 + 1];
    if (lo == 0.f) continue;
    const float mn = fmaxf(m2, mo);
    const float c1f = fast_exp2(m2 - mn);
    const float c2f = fast_exp2(mo - mn);
    l = l * c1f + lo * c2f;
#pragma unroll
    for (int j = 0; j < E; ++j)
      acc[j] = acc[j] * c1f + part_acc[(pbase * G + g) * D + li * E + j] * c2f;
    m2 = mn;
  }
  float o[E];
  const float inv = (l > 0.f) ? 1.f / l : 0.f;
#pragma unroll
  for (int j = 0; j < E; ++j) o[j] = acc[j] * inv;
  const int b = bh / Hkv, kvh = bh % Hkv;
  store_bf16_e<E>(out + (long)b * out_sb + (kvh * G + g) * out_sh + li * E, o);
}

// Pure-streaming probe: same grid/walk/loads as attn_decode_kernel but no
// softmax/acc — measures the access pattern's bandwidth ceiling. NT variant
// uses nontemporal loads (KV is read-once-per-step; keep L2 for weights).
template <int D, bool NT>
__global__ __launch_bounds__(256) void kv_stream_probe_kernel(
    const unsigned short* __restrict__ k_pages,
    const unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table, const int* __restrict__ ctx_lens,
    float* __restrict__ out, int B, int Hkv, int P, int maxp, int n_split) {
  constexpr int E = D / 16;
  const int bh = blockIdx.x;
  const int split = blockIdx.y;
  const int b = bh / Hkv, kvh = bh % Hkv;
  const int ctx = ctx_lens[b];
  const int pages_total = (ctx + P - 1) / P;
  const int pages_per_split = (pages_total + n_split - 1) / n_split;
  const int c0 = split * pages_per_split * P;
  const int c1 = min(ctx, c0 + pages_per_split * P);
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int group_id = wave * 4 + lane / 16;
  const int li = lane & 15;
  const long head_slab = (long)kvh * P * D;
  float acc = 0.f;
  for (int pos = c0 + group_id; pos < c1; pos += 16) {
    const int page = page_table[b * maxp + pos / P];
    const long off =
        ((long)page * Hkv) * P * D + head_slab + (long)(pos % P) * D + li * E;
    const short8* kp8 = reinterpret_cast<const short8*>(k_pages + off);
    const short8* vp8 = reinterpret_cast<const short8*>(v_pages + off);
#pragma unroll
    for (int c = 0; c < E / 8; ++c) {
      short8 kv = NT ? __builtin_nontemporal_load(kp8 + c) : kp8[c];
      short8 vv = NT ? __builtin_nontemporal_load(vp8 + c) : vp8[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += bf2f((unsigned short)kv[j]) + bf2f((unsigned short)vv[j]);
    }
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) out[(blockIdx.y * gridDim.x + blockIdx.x) * 4 + wave] = acc;
}
