// Common device helpers for bloombee_amd gfx950 kernels.
// CDNA4 only: wave64, MFMA, 160 KiB LDS/CU. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>

#define DEVINL __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) short short8;   // 8 x bf16 = 16 B
typedef __attribute__((ext_vector_type(4))) float f32x4;    // MFMA 16x16 accum
typedef __attribute__((ext_vector_type(2))) float f32x2;

static constexpr int WAVE = 64;  // CDNA wavefront (guide §1: not 32)
static constexpr float LOG2E = 1.4426950408889634f;
static constexpr float NEG_BIG = -1e30f;

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
DEVINL bf16x8 as_bf16x8(short8 s) { return __builtin_bit_cast(bf16x8, s); }

// bf16 <-> f32 as raw ushort bits (round-to-nearest-even on pack).
DEVINL float bf2f(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

DEVINL unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int x = v.i;
  unsigned int lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;  // RNE
  // NaN stays NaN
  if ((v.i & 0x7f800000u) == 0x7f800000u && (v.i & 0x007fffffu)) x = v.i | 0x00400000u;
  return (unsigned short)(x >> 16);
}

// Load 8 bf16 (16 B) and widen to f32.
DEVINL void load_bf16x8(const unsigned short* p, float* out) {
  short8 v = *reinterpret_cast<const short8*>(p);
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = bf2f((unsigned short)v[j]);
}

DEVINL void store_bf16x8(unsigned short* p, const float* in) {
  short8 v;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = (short)f2bf(in[j]);
  *reinterpret_cast<short8*>(p) = v;
}

// Butterfly reduce across a full wave (all 64 lanes end with the result).
DEVINL float wave_reduce_sum(float x) {
#pragma unroll
  for (int m = 1; m < WAVE; m <<= 1) x += __shfl_xor(x, m);
  return x;
}

DEVINL float wave_reduce_max(float x) {
#pragma unroll
  for (int m = 1; m < WAVE; m <<= 1) x = fmaxf(x, __shfl_xor(x, m));
  return x;
}

// Reduce across a 16-lane subgroup (lanes l..l+15 with aligned base).
DEVINL float group16_reduce_sum(float x) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) x += __shfl_xor(x, m);
  return x;
}

DEVINL float group16_reduce_max(float x) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) x = fmaxf(x, __shfl_xor(x, m));
  return x;
}

// v_exp_f32 — hardware exp2 (TRANS pipe). exp2(-1e30) flushes to 0 as needed.
DEVINL float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }

DEVINL float fast_expf(float x) { return __builtin_amdgcn_exp2f(x * LOG2E); }

DEVINL long ldiv_up(long a, long b) { return (a + b - 1) / b; }

#define HIP_OK(expr)                                                         \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",       \
                  __FILE__, ":", __LINE__);                                  \
    }                                                                        \
  } while (0)
