// Fused elementwise / normalization kernels for gfx950.
//
// rmsnorm_kernel       — fused (optional residual-add) + RMSNorm, fp32 accum
//                        (replaces reference flexgen_utils/pytorch_backend.py:111-120;
//                        fused per SURVEY.md: HBM3E-bound ops get one pass).
// layernorm_kernel     — LayerNorm for bloom/falcon-family blocks.
// rope_kernel          — rotary embedding with arbitrary position ids
//                        (ref pytorch_backend.py:59-110), host-precomputed
//                        cos/sin tables (guide Appendix B: no device trig).
// swiglu_kernel        — SiLU(gate) * up from a fused gate_up GEMM output
//                        (ref mlp_llama, pytorch_backend.py:1033-1048).
// gelu_mul / bias ops  — falcon/bloom variants.
//
// All bf16 I/O is vectorized 16 B/lane (guide G13: scalar bf16 ~2x slower).

#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: one workgroup per row. ITERS*BLOCK*8 >= H.
// If `res` != nullptr: h = x + res is written to out_h and normalized.
// ---------------------------------------------------------------------------

template <int BLOCK, int ITERS>
__global__ __launch_bounds__(BLOCK) void rmsnorm_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ res,
    const unsigned short* __restrict__ w, unsigned short* __restrict__ out_h,
    unsigned short* __restrict__ out_y, int H, float eps) {
  const long row = blockIdx.x;
  const unsigned short* xr = x + row * (long)H;
  const unsigned short* rr = res ? res + row * (long)H : nullptr;

  float v[ITERS][8];
  float ssq = 0.f;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int d = (it * BLOCK + threadIdx.x) * 8;
    if (d < H) {
      load_bf16x8(xr + d, v[it]);
      if (rr) {
        float r[8];
        load_bf16x8(rr + d, r);
#pragma unroll
        for (int j = 0; j < 8; ++j) v[it][j] = bf2f(f2bf(v[it][j] + r[j]));
        if (out_h) store_bf16x8(out_h + row * (long)H + d, v[it]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) ssq += v[it][j] * v[it][j];
    }
  }
  // wave reduce then cross-wave via LDS
  ssq = wave_reduce_sum(ssq);
  __shared__ float warp_ssq[BLOCK / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_ssq[threadIdx.x / WAVE] = ssq;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < BLOCK / WAVE; ++i) total += warp_ssq[i];
  const float inv = rsqrtf(total / (float)H + eps);

#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int d = (it * BLOCK + threadIdx.x) * 8;
    if (d < H) {
      float wv[8], o[8];
      load_bf16x8(w + d, wv);
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = v[it][j] * inv * wv[j];
      store_bf16x8(out_y + row * (long)H + d, o);
    }
  }
}

// ---------------------------------------------------------------------------
// LayerNorm (bloom/falcon). Same structure, mean+var in fp32.
// ---------------------------------------------------------------------------

template <int BLOCK, int ITERS>
__global__ __launch_bounds__(BLOCK) void layernorm_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ res,
    const unsigned short* __restrict__ w, const unsigned short* __restrict__ b,
    unsigned short* __restrict__ out_h, unsigned short* __restrict__ out_y,
    int H, float eps) {
  const long row = blockIdx.x;
  const unsigned short* xr = x + row * (long)H;
  const unsigned short* rr = res ? res + row * (long)H : nullptr;

  float v[ITERS][8];
  float s1 = 0.f, s2 = 0.f;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int d = (it * BLOCK + threadIdx.x) * 8;
    if (d < H) {
      load_bf16x8(xr + d, v[it]);
      if (rr) {
        float r[8];
        load_bf16x8(rr + d, r);
#pragma unroll
        for (int j = 0; j < 8; ++j) v[it][j] += r[j];
        if (out_h) store_bf16x8(out_h + row * (long)H + d, v[it]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) { s1 += v[it][j]; s2 += v[it][j] * v[it][j]; }
    }
  }
  s1 = wave_reduce_sum(s1);
  s2 = wave_reduce_sum(s2);
  __shared__ float ws1[BLOCK / WAVE], ws2[BLOCK / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    ws1[threadIdx.x / WAVE] = s1;
    ws2[threadIdx.x / WAVE] = s2;
  }
  __syncthreads();
  float t1 = 0.f, t2 = 0.f;
#pragma unroll
  for (int i = 0; i < BLOCK / WAVE; ++i) { t1 += ws1[i]; t2 += ws2[i]; }
  const float mu = t1 / (float)H;
  const float inv = rsqrtf(t2 / (float)H - mu * mu + eps);

#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int d = (it * BLOCK + threadIdx.x) * 8;
    if (d < H) {
      float wv[8], bv[8], o[8];
      load_bf16x8(w + d, wv);
      if (b) load_bf16x8(b + d, bv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        o[j] = (v[it][j] - mu) * inv * wv[j] + (b ? bv[j] : 0.f);
      }
      store_bf16x8(out_y + row * (long)H + d, o);
    }
  }
}

// ---------------------------------------------------------------------------
// RoPE (half-split convention), in place on q and k.
// q: (B, Hq, T, D)  k: (B, Hkv, T, D)  pos: (B, T) int32
// cos/sin: (max_pos, D/2) f32, precomputed on host.
// grid: (B*T, Hq + Hkv), block: 64.
// ---------------------------------------------------------------------------

__global__ void rope_kernel(
    unsigned short* __restrict__ q, unsigned short* __restrict__ k,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    const int* __restrict__ pos, int B, int Hq, int Hkv, int T, int D,
    int max_pos) {
  const int bt = blockIdx.x;
  const int b = bt / T, t = bt % T;
  const int h = blockIdx.y;
  unsigned short* base =
      (h < Hq) ? q + (((long)b * Hq + h) * T + t) * D
               : k + (((long)b * Hkv + (h - Hq)) * T + t) * D;
  // clamp: positions past the rope table (model max_position_embeddings)
  // must not read out of bounds — the python layer guards the supported
  // range, this is the device-safety net
  const int p = min(max(pos[b * T + t], 0), max_pos - 1);
  const float* c = cos_t + (long)p * (D / 2);
  const float* s = sin_t + (long)p * (D / 2);
  for (int i = threadIdx.x; i < D / 2; i += blockDim.x) {
    float x1 = bf2f(base[i]), x2 = bf2f(base[i + D / 2]);
    float cc = c[i], ss = s[i];
    base[i] = f2bf(x1 * cc - x2 * ss);
    base[i + D / 2] = f2bf(x2 * cc + x1 * ss);
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: out[n, i] = silu(gu[n, i]) * gu[n, I + i];  gu: (N, 2I)
// ---------------------------------------------------------------------------

__global__ void swiglu_kernel(const unsigned short* __restrict__ gu,
                              unsigned short* __restrict__ out, long N, long I) {
  const long nvec = N * (I / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < nvec;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / (I / 8);
    const long col = (idx % (I / 8)) * 8;
    float g[8], u[8], o[8];
    load_bf16x8(gu + row * 2 * I + col, g);
    load_bf16x8(gu + row * 2 * I + I + col, u);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float s = g[j] / (1.f + fast_expf(-g[j]));
      o[j] = s * u[j];
    }
    store_bf16x8(out + row * I + col, o);
  }
}

// gelu(tanh approx)(x) * 1.0 — falcon/bloom MLP activation, out-of-place.
__global__ void gelu_kernel(const unsigned short* __restrict__ x,
                            unsigned short* __restrict__ out, long n) {
  const long nvec = n / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < nvec;
       idx += (long)gridDim.x * blockDim.x) {
    float v[8], o[8];
    load_bf16x8(x + idx * 8, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = tanhf(0.7978845608028654f * (v[j] + 0.044715f * v[j] * v[j] * v[j]));
      o[j] = 0.5f * v[j] * (1.f + t);
    }
    store_bf16x8(out + idx * 8, o);
  }
}

// residual add: out = a + b (bf16), used where no norm fusion applies
__global__ void add_kernel(const unsigned short* __restrict__ a,
                           const unsigned short* __restrict__ b,
                           unsigned short* __restrict__ out, long n) {
  const long nvec = n / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < nvec;
       idx += (long)gridDim.x * blockDim.x) {
    float va[8], vb[8], o[8];
    load_bf16x8(a + idx * 8, va);
    load_bf16x8(b + idx * 8, vb);
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = va[j] + vb[j];
    store_bf16x8(out + idx * 8, o);
  }
}

// ---------------------------------------------------------------------------
// Fused RoPE + paged-KV write on the raw QKV GEMM output (B, T, (Hq+2Hkv)*D).
// q heads: rotary in place. k heads: rotary -> k_pages (never written back).
// v heads: straight copy -> v_pages. One launch replaces rope + kv_write and
// removes every transpose copy between the QKV GEMM and attention.
// grid: (B*T, Hq+2Hkv), block: 64.
// ---------------------------------------------------------------------------

__global__ void rope_kv_write_kernel(
    unsigned short* __restrict__ qkv, const float* __restrict__ cos_t,
    const float* __restrict__ sin_t, const int* __restrict__ pos,
    unsigned short* __restrict__ k_pages, unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table, const int* __restrict__ start_pos,
    int B, int Hq, int Hkv, int T, int D, int P, int maxp, int max_pos) {
  const int bt = blockIdx.x;
  const int b = bt / T, t = bt % T;
  const int h = blockIdx.y;
  const int X = Hq + 2 * Hkv;
  unsigned short* base = qkv + (((long)b * T + t) * X + h) * D;

  const int abspos = start_pos[b] + t;
  if (h >= Hq + Hkv) {  // v head: copy to pages (d-major V layout: transpose
    const int hv = h - Hq - Hkv;  // on write, scattered 2 B stores, once/token)
    const int page = page_table[b * maxp + abspos / P];
    const int slot = abspos % P;
    unsigned short* dst = v_pages + ((long)page * Hkv + hv) * D * P + slot;
    for (int i = threadIdx.x; i < D; i += blockDim.x)
      dst[(long)i * P] = base[i];
    return;
  }
  const int p = min(max(pos ? pos[b * T + t] : abspos, 0), max_pos - 1);
  const float* c = cos_t + (long)p * (D / 2);
  const float* s = sin_t + (long)p * (D / 2);
  unsigned short* dst = base;
  if (h >= Hq) {  // k head: write rotated values to pages only
    const int hk = h - Hq;
    const int page = page_table[b * maxp + abspos / P];
    dst = k_pages + (((long)page * Hkv + hk) * P + (abspos % P)) * D;
  }
  for (int i = threadIdx.x; i < D / 2; i += blockDim.x) {
    float x1 = bf2f(base[i]), x2 = bf2f(base[i + D / 2]);
    float cc = c[i], ss = s[i];
    dst[i] = f2bf(x1 * cc - x2 * ss);
    dst[i + D / 2] = f2bf(x2 * cc + x1 * ss);
  }
}
