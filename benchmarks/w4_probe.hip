// Standalone A/B probe for the W4 decode GEMM (hipcc, no torch).
// Variants (M=1, llama-3-8b gate_up shape N=28672 K=4096 unless noted):
//   A: committed v2 — 4 B code loads per 32-k slice, per-slice scale loads
//   B: A + scale/zero hoisted to one 8 B load per 128 k
//   C: 16 B code loads (1/4 request rate) + predicated-shfl distribution
//   S: pure stream of the packed codes (upper bound for this byte count)
//   G: bf16 gemm_skinny v1 shape-for-shape (the thing to beat)
// Build: hipcc --offload-arch=gfx950 -O3 w4_probe.hip -o w4_probe
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define WAVE 64
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) _Float16 half8;
typedef __attribute__((ext_vector_type(2))) _Float16 half2v;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

__device__ __forceinline__ bf16x8 as_bf16x8(short8 s) {
  return __builtin_bit_cast(bf16x8, s);
}
__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v; v.f = f;
  unsigned int x = v.i; x += 0x7fffu + ((x >> 16) & 1u);
  return (unsigned short)(x >> 16);
}

__device__ __forceinline__ half8 dq8(unsigned int c, half2v sc2, half2v zp2) {
  const half2v magic = {(_Float16)1024.f, (_Float16)1024.f};
  half8 out;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const unsigned byte = (c >> (8 * p)) & 0xFFu;
    const unsigned h2 = 0x64006400u | (byte & 0xFu) | ((byte & 0xF0u) << 12);
    half2v v = __builtin_bit_cast(half2v, h2);
    v = (v - magic) * sc2 + zp2;
    out[2 * p] = v[0];
    out[2 * p + 1] = v[1];
  }
  return out;
}

// ---- variant A: 4B loads, per-slice scale ----
template <int MODE>  // 0=A, 1=B, 2=C, 3=C+double-buffer prefetch
__global__ __launch_bounds__(256) void w4k(
    const _Float16* __restrict__ A, const unsigned char* __restrict__ Wq,
    const __half* __restrict__ scale, const __half* __restrict__ zero,
    unsigned short* __restrict__ C, int M, int N, int K, int kchunk) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  const int n0 = blockIdx.x * 64 + wave * 16;
  if (n0 >= N) return;
  const unsigned char* wrow = Wq + (long)(n0 + li) * (K / 2);
  const __half* srow = scale + (long)(n0 + li) * (K / 64);
  const __half* zrow = zero + (long)(n0 + li) * (K / 64);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int arow = 0;  // M=1
  const int k0 = blockIdx.y * kchunk;
  const int k1 = (k0 + kchunk < K) ? k0 + kchunk : K;

  if (MODE == 4) {
    // depth-2 pipeline: two 16B code words + scale words in flight
    uint4v cw[3]; unsigned int sv[3], zv[3];
    auto ld = [&](int slot, int k) {
      cw[slot] = *reinterpret_cast<const uint4v*>(wrow + (k + hi * 32) / 2);
      sv[slot] = *reinterpret_cast<const unsigned int*>(srow + k / 64);
      zv[slot] = *reinterpret_cast<const unsigned int*>(zrow + k / 64);
    };
    ld(0, k0);
    if (k0 + 128 < k1) ld(1, k0 + 128);
    int w = 0;
    for (int k = k0; k < k1; k += 128) {
      if (k + 256 < k1) ld((w + 2) % 3, k + 256);
      half2v sh = __builtin_bit_cast(half2v, sv[w]);
      half2v zh = __builtin_bit_cast(half2v, zv[w]);
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const half2v sc2 = {sh[s / 2], sh[s / 2]};
        const half2v zp2 = {zh[s / 2], zh[s / 2]};
        unsigned int c2 = 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          unsigned int v = __shfl(cw[w][j], s * 16 + li);
          if (hi == j) c2 = v;
        }
        half8 bfrag = dq8(c2, sc2, zp2);
        half8 afrag = *reinterpret_cast<const half8*>(
            A + (long)arow * K + k + s * 32 + hi * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(afrag, bfrag, acc, 0, 0, 0);
      }
      w = (w + 1) % 3;
    }
    if (hi == 0) C[(long)blockIdx.y * N + n0 + li] = f2bf(acc[0]);
    return;
  }

  if (MODE == 3) {
    // double-buffered 16B code words + hoisted 8B scale words: ~3 HBM
    // latencies of codes in flight per wave
    uint4v cwA = *reinterpret_cast<const uint4v*>(wrow + (k0 + hi * 32) / 2);
    unsigned int svA = *reinterpret_cast<const unsigned int*>(srow + k0 / 64);
    unsigned int zvA = *reinterpret_cast<const unsigned int*>(zrow + k0 / 64);
    for (int k = k0; k < k1; k += 128) {
      uint4v cwB; unsigned int svB, zvB;
      if (k + 128 < k1) {
        cwB = *reinterpret_cast<const uint4v*>(wrow + (k + 128 + hi * 32) / 2);
        svB = *reinterpret_cast<const unsigned int*>(srow + (k + 128) / 64);
        zvB = *reinterpret_cast<const unsigned int*>(zrow + (k + 128) / 64);
      }
      half2v sh = __builtin_bit_cast(half2v, svA);
      half2v zh = __builtin_bit_cast(half2v, zvA);
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const half2v sc2 = {sh[s / 2], sh[s / 2]};
        const half2v zp2 = {zh[s / 2], zh[s / 2]};
        unsigned int cw = 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          unsigned int v = __shfl(cwA[j], s * 16 + li);
          if (hi == j) cw = v;
        }
        half8 bfrag = dq8(cw, sc2, zp2);
        half8 afrag = *reinterpret_cast<const half8*>(
            A + (long)arow * K + k + s * 32 + hi * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(afrag, bfrag, acc, 0, 0, 0);
      }
      cwA = cwB; svA = svB; zvA = zvB;
    }
    if (hi == 0) C[(long)blockIdx.y * N + n0 + li] = f2bf(acc[0]);
    return;
  }

  for (int k = k0; k < k1; k += 128) {
    _Float16 scs[2], zps[2];
    if (MODE >= 1) {
      // one 4B load each for the two groups' scales + zeros
      const unsigned int sv = *reinterpret_cast<const unsigned int*>(srow + k / 64);
      const unsigned int zv = *reinterpret_cast<const unsigned int*>(zrow + k / 64);
      half2v sh = __builtin_bit_cast(half2v, sv);
      half2v zh = __builtin_bit_cast(half2v, zv);
      scs[0] = sh[0]; scs[1] = sh[1]; zps[0] = zh[0]; zps[1] = zh[1];
    }
    uint4v cw4;
    if (MODE == 2)
      cw4 = *reinterpret_cast<const uint4v*>(wrow + (k + hi * 32) / 2);
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int ks = k + s * 32;
      _Float16 sch, zph;
      if (MODE >= 1) { sch = scs[s / 2]; zph = zps[s / 2]; }
      else { sch = (_Float16)srow[ks / 64]; zph = (_Float16)zrow[ks / 64]; }
      const half2v sc2 = {sch, sch};
      const half2v zp2 = {zph, zph};
      unsigned int cw;
      if (MODE == 2) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          unsigned int v = __shfl(cw4[j], s * 16 + li);
          if (hi == j) cw = v;
        }
      } else {
        cw = *reinterpret_cast<const unsigned int*>(wrow + (ks + hi * 8) / 2);
      }
      half8 bfrag = dq8(cw, sc2, zp2);
      half8 afrag = *reinterpret_cast<const half8*>(A + (long)arow * K + ks + hi * 8);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(afrag, bfrag, acc, 0, 0, 0);
    }
  }
  const int m = hi * 4;  // only reg 0 row matters for M=1 check
  if (m == 0) C[(long)0 * N + n0 + li] = f2bf(acc[0]);
}

// ---- GEMV formulation for M<=1: one row per wave pass, 64 lanes cover
// 2048 codes (1KB contiguous per wave-load — the stream kernel's pattern).
// A is loaded once into registers (K fp16 = 64 elems/lane over K=4096).
__global__ __launch_bounds__(256) void w4gemv(
    const _Float16* __restrict__ A, const unsigned char* __restrict__ Wq,
    const __half* __restrict__ scale, const __half* __restrict__ zero,
    unsigned short* __restrict__ C, int N, int K, int rows_per_wave) {
  typedef __attribute__((ext_vector_type(2))) _Float16 h2;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_base = (blockIdx.x * 4 + wave) * rows_per_wave;
  if (n_base >= N) return;
  const int nloads = K / 2048;  // 1KB wave-loads per row (K=4096 -> 2)
  // A: lane l holds elements [h*2048 + l*32, +32) for h in 0..nloads-1
  h2 areg[2][16];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int j = 0; j < 16; ++j)
      areg[h][j] = *reinterpret_cast<const h2*>(A + h * 2048 + lane * 32 + 2 * j);

  const half2v magic = {(_Float16)1024.f, (_Float16)1024.f};
  uint4v buf[2][2];  // [row parity][half]
  auto ldrow = [&](int slot, int n) {
    const unsigned char* wrow = Wq + (long)n * (K / 2);
#pragma unroll
    for (int h = 0; h < 2; ++h)
      buf[slot][h] = *reinterpret_cast<const uint4v*>(wrow + h * 1024 + lane * 16);
  };
  ldrow(0, n_base);
  for (int r = 0; r < rows_per_wave; ++r) {
    const int n = n_base + r;
    if (n >= N) break;
    if (r + 1 < rows_per_wave && n + 1 < N) ldrow((r + 1) & 1, n + 1);
    const __half* srow = scale + (long)n * (K / 64);
    const __half* zrow = zero + (long)n * (K / 64);
    float acc = 0.f;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      // this lane's 32 codes start at k = h*2048 + lane*32 -> group (k/64)
      const int g = (h * 2048 + lane * 32) / 64;
      const _Float16 sch = (_Float16)srow[g];
      const _Float16 zph = (_Float16)zrow[g];
      const half2v sc2 = {sch, sch};
      const half2v zp2 = {zph, zph};
      const uint4v cw = buf[r & 1][h];
#pragma unroll
      for (int d = 0; d < 4; ++d) {
        const unsigned int c = cw[d];
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          const unsigned byte = (c >> (8 * p)) & 0xFFu;
          const unsigned hh = 0x64006400u | (byte & 0xFu) | ((byte & 0xF0u) << 12);
          half2v v = __builtin_bit_cast(half2v, hh);
          v = (v - magic) * sc2 + zp2;
          h2 a = areg[h][d * 4 + p];
          // v_dot2_f32_f16: one instruction per pair instead of scalarized
          // f16->f32 converts + adds (the issue-stall the PMC showed)
          acc = __builtin_amdgcn_fdot2(
              __builtin_bit_cast(half2v, a), v, acc, false);
        }
      }
    }
#pragma unroll
    for (int m = 1; m < WAVE; m <<= 1) acc += __shfl_xor(acc, m);
    if (lane == 0) C[n] = f2bf(acc);
  }
}

// ---- pure stream of the packed array ----
__global__ __launch_bounds__(256) void streamk(
    const unsigned char* __restrict__ Wq, float* __restrict__ sink,
    long nbytes) {
  const long tid = blockIdx.x * 256 + threadIdx.x;
  const long stride = (long)gridDim.x * 256;
  float acc = 0.f;
  for (long i = tid * 16; i + 16 <= nbytes; i += stride * 16) {
    uint4v v = *reinterpret_cast<const uint4v*>(Wq + i);
    acc += (float)(v[0] ^ v[1] ^ v[2] ^ v[3]);
  }
  if (acc == 12345.678f) sink[0] = acc;
}

// ---- bf16 skinny v1 (reference timing) ----
__global__ __launch_bounds__(256) void bf16k(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ W,
    unsigned short* __restrict__ C, int M, int N, int K) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int li = lane & 15;
  const int hi = lane >> 4;
  const int n0 = blockIdx.x * 64 + wave * 16;
  if (n0 >= N) return;
  const unsigned short* wrow = W + (long)(n0 + li) * K;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k = 0; k < K; k += 128) {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int kk = k + u * 32 + hi * 8;
      bf16x8 bfrag = as_bf16x8(*reinterpret_cast<const short8*>(wrow + kk));
      bf16x8 afrag = as_bf16x8(*reinterpret_cast<const short8*>(A + kk));
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0, 0, 0);
    }
  }
  if (hi == 0) C[n0 + li] = f2bf(acc[0]);
}



int main(int argc, char** argv) {
  const int N = 28672, K = 4096, M = 1;
  unsigned char* Wq; __half *sc, *zp; _Float16* A; unsigned short* C;
  unsigned short *Wb, *Ab;
  float* sink;
  hipMalloc(&Wq, (long)N * K / 2);
  hipMalloc(&sc, (long)N * K / 64 * 2);
  hipMalloc(&zp, (long)N * K / 64 * 2);
  hipMalloc(&A, (long)M * K * 2);
  hipMalloc(&C, (long)16 * N * 2);
  hipMalloc(&Wb, (long)N * K * 2);
  hipMalloc(&Ab, (long)M * K * 2);
  hipMalloc(&sink, 4);
  hipMemset(Wq, 0x53, (long)N * K / 2);
  hipMemset(sc, 0x3c, (long)N * K / 64 * 2);
  hipMemset(zp, 0, (long)N * K / 64 * 2);
  hipMemset(A, 0x3c, (long)M * K * 2);
  hipMemset(Wb, 0x3f, (long)N * K * 2);
  hipMemset(Ab, 0x3f, (long)M * K * 2);

  auto time_it = [&](const char* name, auto launch, double bytes) {
    for (int i = 0; i < 10; ++i) launch();
    hipDeviceSynchronize();
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < 50; ++i) launch();
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms;
    hipEventElapsedTime(&ms, e0, e1);
    double us = ms * 1000.0 / 50;
    printf("%-28s %8.1f us  %6.2f TB/s\n", name, us, bytes / (us * 1e-6) / 1e12);
  };

  const double w4_bytes = (double)N * K / 2 + (double)N * K / 64 * 4;
  dim3 grid(N / 64);
  time_it("A: 4B codes, slice scales", [&] {
    hipLaunchKernelGGL((w4k<0>), grid, 256, 0, 0, A, Wq, sc, zp, C, M, N, K, K);
  }, w4_bytes);
  time_it("B: 4B codes, hoisted scales", [&] {
    hipLaunchKernelGGL((w4k<1>), grid, 256, 0, 0, A, Wq, sc, zp, C, M, N, K, K);
  }, w4_bytes);
  time_it("C: 16B codes + shfl", [&] {
    hipLaunchKernelGGL((w4k<2>), grid, 256, 0, 0, A, Wq, sc, zp, C, M, N, K, K);
  }, w4_bytes);
  for (int ks : {2, 4, 8, 16}) {
    char nm[64];
    dim3 g2(N / 64, ks);
    const int kchunk = ((K / 128 + ks - 1) / ks) * 128;
    snprintf(nm, 64, "D: depth1 ksplit=%d", ks);
    time_it(nm, [&] {
      hipLaunchKernelGGL((w4k<3>), g2, 256, 0, 0, A, Wq, sc, zp, C, M, N, K, kchunk);
    }, w4_bytes);
    snprintf(nm, 64, "E: depth2 ksplit=%d", ks);
    time_it(nm, [&] {
      hipLaunchKernelGGL((w4k<4>), g2, 256, 0, 0, A, Wq, sc, zp, C, M, N, K, kchunk);
    }, w4_bytes);
  }
  for (int rpw : {4, 8, 16}) {
    char nm[64];
    snprintf(nm, 64, "V: gemv rows/wave=%d", rpw);
    dim3 gv((N / rpw + 3) / 4);
    time_it(nm, [&] {
      hipLaunchKernelGGL(w4gemv, gv, 256, 0, 0, A, Wq, sc, zp, C, N, K, rpw);
    }, w4_bytes);
  }
  time_it("S: pure 16B stream of codes", [&] {
    hipLaunchKernelGGL(streamk, dim3(2048), 256, 0, 0, Wq, sink, (long)N * K / 2);
  }, (double)N * K / 2);
  time_it("G: bf16 skinny v1", [&] {
    hipLaunchKernelGGL(bf16k, grid, 256, 0, 0, Ab, Wb, C, M, N, K);
  }, (double)N * K * 2);
  return 0;
}
