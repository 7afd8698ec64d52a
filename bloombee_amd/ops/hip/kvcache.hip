// Paged KV cache device primitives for gfx950.
//
// kv_write_kernel   — scatter new K/V rows into the paged pool, in place
//                     (ref server/paged_kv.py:137-204 `write`, and the slab
//                     in-place write of pytorch_backend.py:843-850 — here
//                     paged-first, MI-native layout).
// kv_gather_kernel  — gather a prefix back to dense (debug/failover path,
//                     ref paged_kv.py:265-316 `gather_prefix`).
//
// Layouts: K pages (n_pages, Hkv, P, D) token-major; V pages
// (n_pages, Hkv, D, P) d-major. V is transposed on WRITE (once per token,
// scattered 2 B stores are cheap) so the decode attention P.V MFMA
// B-fragments — 8 consecutive positions at a fixed d — are direct contiguous
// 16 B READS from HBM every step, with no LDS transpose staging.

#include "common.h"

__global__ void kv_write_kernel(
    const unsigned short* __restrict__ k_new, const unsigned short* __restrict__ v_new,
    unsigned short* __restrict__ k_pages, unsigned short* __restrict__ v_pages,
    const int* __restrict__ page_table, const int* __restrict__ start_pos,
    int B, int Hkv, int T, int D, int P, int maxp) {
  const int bt = blockIdx.x;
  const int b = bt / T, t = bt % T;
  const int pos = start_pos[b] + t;
  const int page = page_table[b * maxp + pos / P];
  const int slot = pos % P;
  const int nvec = Hkv * (D / 8);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = i / (D / 8);
    const int d = (i % (D / 8)) * 8;
    const long src = (((long)b * Hkv + h) * T + t) * D + d;
    const long kdst = (((long)page * Hkv + h) * P + slot) * D + d;
    *reinterpret_cast<short8*>(k_pages + kdst) =
        *reinterpret_cast<const short8*>(k_new + src);
    const short8 vv = *reinterpret_cast<const short8*>(v_new + src);
    const long vbase = (((long)page * Hkv + h) * D + d) * P + slot;
#pragma unroll
    for (int j = 0; j < 8; ++j) v_pages[vbase + (long)j * P] = (unsigned short)vv[j];
  }
}

// Gather ctx_len tokens of one sequence to dense (Hkv, ctx, D) k/v.
__global__ void kv_gather_kernel(
    const unsigned short* __restrict__ k_pages, const unsigned short* __restrict__ v_pages,
    unsigned short* __restrict__ k_out, unsigned short* __restrict__ v_out,
    const int* __restrict__ page_table, int batch_index, int ctx, int Hkv,
    int D, int P, int maxp) {
  const long nvec = (long)ctx * Hkv * (D / 8);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    const int d = (int)(i % (D / 8)) * 8;
    const int h = (int)((i / (D / 8)) % Hkv);
    const int pos = (int)(i / ((D / 8) * (long)Hkv));
    const int page = page_table[batch_index * maxp + pos / P];
    const long ksrc = (((long)page * Hkv + h) * P + (pos % P)) * D + d;
    const long dst = (((long)h * ctx) + pos) * D + d;
    *reinterpret_cast<short8*>(k_out + dst) =
        *reinterpret_cast<const short8*>(k_pages + ksrc);
    const long vbase = (((long)page * Hkv + h) * D + d) * P + (pos % P);
    short8 vv;
#pragma unroll
    for (int j = 0; j < 8; ++j) vv[j] = (short)v_pages[vbase + (long)j * P];
    *reinterpret_cast<short8*>(v_out + dst) = vv;
  }
}
