// 4-bit group quantization for gfx950 (ref flexgen_utils/compression.py:94-210):
// per-group-of-64 min/max scaling, packed 2 values/byte. Used by the offload
// tier for compressed weight/KV storage and by the wire codec.

#include "common.h"

// x: (nrows, ncols) bf16, group along cols. packed: (nrows, ncols/2) u8;
// scale/zero: (nrows, ncols/GS) f16. One wave per group (GS=64: one elem/lane).
template <int GS>
__global__ void quant4_pack_kernel(const unsigned short* __restrict__ x,
                                   unsigned char* __restrict__ packed,
                                   __half* __restrict__ scale,
                                   __half* __restrict__ zero,
                                   long nrows, long ncols) {
  const long ngroups_row = ncols / GS;
  const long gid = blockIdx.x * (long)(blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (gid >= nrows * ngroups_row) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const long row = gid / ngroups_row;
  const long g0 = (gid % ngroups_row) * GS;

  float v = (lane < GS) ? bf2f(x[row * ncols + g0 + lane]) : 0.f;
  float mn = (lane < GS) ? v : 1e30f;
  float mx = (lane < GS) ? v : -1e30f;
#pragma unroll
  for (int m = 1; m < WAVE; m <<= 1) {
    mn = fminf(mn, __shfl_xor(mn, m));
    mx = fmaxf(mx, __shfl_xor(mx, m));
  }
  const float sc = fmaxf(mx - mn, 1e-8f) / 15.f;
  const unsigned qv = (unsigned)fminf(fmaxf(rintf((v - mn) / sc), 0.f), 15.f);
  // pack pairs: even lane provides low nibble, odd lane high nibble
  const unsigned other = __shfl_xor(qv, 1);
  if (lane < GS && (lane & 1) == 0) {
    packed[(row * ncols + g0 + lane) / 2] = (unsigned char)(qv | (other << 4));
  }
  if (lane == 0) {
    scale[gid] = __float2half(sc);
    zero[gid] = __float2half(mn);
  }
}

template <int GS>
__global__ void quant4_unpack_kernel(const unsigned char* __restrict__ packed,
                                     const __half* __restrict__ scale,
                                     const __half* __restrict__ zero,
                                     unsigned short* __restrict__ out,
                                     long nrows, long ncols) {
  const long ngroups_row = ncols / GS;
  const long total = nrows * ncols;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / ncols, col = i % ncols;
    const long gid = row * ngroups_row + col / GS;
    const unsigned char byte = packed[i / 2];
    const unsigned q = (i & 1) ? (byte >> 4) : (byte & 0xF);
    out[i] = f2bf((float)q * __half2float(scale[gid]) + __half2float(zero[gid]));
  }
}
