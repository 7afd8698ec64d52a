// Decode-attention support kernels for gfx950.
//
// attn_decode_combine_kernel — folds the per-split (m, l, acc) partials of
// the flash-decode MFMA kernel (attn_decode_mfma.hip) into the final output
// (softmax merge identical to the reference fp32 semantics,
// pytorch_backend.py:919-934).
// kv_stream_probe_kernel     — pure-streaming bandwidth probe of the KV walk.

#include "common.h"

typedef __attribute__((ext_vector_type(4))) short short4_v;
typedef __attribute__((ext_vector_type(2))) short short2_v;

template <int E>
DEVINL void load_bf16_e(const unsigned short* p, float* out) {
  if constexpr (E == 2) {
    short2_v v = *reinterpret_cast<const short2_v*>(p);
    out[0] = bf2f((unsigned short)v[0]);
    out[1] = bf2f((unsigned short)v[1]);
  } else if constexpr (E == 4) {
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
  if constexpr (E == 2) {
    short2_v v;
    v[0] = (short)f2bf(in[0]);
    v[1] = (short)f2bf(in[1]);
    *reinterpret_cast<short2_v*>(p) = v;
  } else if constexpr (E == 4) {
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


// Combine n_split partials. grid = (B*Hkv), block = G*16 (<= 128 threads).
template <int D>
__global__ void attn_decode_combine_kernel(
    const float* __restrict__ part_ml, const float* __restrict__ part_acc,
    unsigned short* __restrict__ out, int Hkv, int G, int nch, int maxg,
    int n_split, long out_sb, long out_sh) {
  constexpr int E = D / 16;
  const int bh = blockIdx.x;           // (b, kvh, chunk)
  const int b = bh / (Hkv * nch);
  const int rem = bh % (Hkv * nch);
  const int kvh = rem / nch;
  const int g0 = (rem % nch) * maxg;
  const int g = threadIdx.x / 16;
  const int li = threadIdx.x & 15;
  if (g0 + g >= G || g >= maxg) return;
  float m2 = NEG_BIG, l = 0.f, acc[E];
#pragma unroll
  for (int j = 0; j < E; ++j) acc[j] = 0.f;
  for (int s = 0; s < n_split; ++s) {
    const long pbase = (long)bh * n_split + s;
    const float mo = part_ml[(pbase * maxg + g) * 2 + 0];
    const float lo = part_ml[(pbase * maxg + g) * 2 + 1];
    if (lo == 0.f) continue;
    const float mn = fmaxf(m2, mo);
    const float c1f = fast_exp2(m2 - mn);
    const float c2f = fast_exp2(mo - mn);
    l = l * c1f + lo * c2f;
#pragma unroll
    for (int j = 0; j < E; ++j)
      acc[j] = acc[j] * c1f + part_acc[(pbase * maxg + g) * D + li * E + j] * c2f;
    m2 = mn;
  }
  float o[E];
  const float inv = (l > 0.f) ? 1.f / l : 0.f;
#pragma unroll
  for (int j = 0; j < E; ++j) o[j] = acc[j] * inv;
  store_bf16_e<E>(out + (long)b * out_sb + (kvh * G + g0 + g) * out_sh + li * E,
                  o);
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
