"""Kernel microbenchmarks on a real MI355X (run via gpurun).

Reports per-kernel time and effective HBM bandwidth / TFLOPs so kernel
changes can be A/B'd without the end-to-end bench's noise.

    python benchmarks/bench_kernels.py [attn_decode attn_prefill norms ...]
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bloombee_amd import ops
from bloombee_amd.ops import reference as ref

DEV = "cuda:0"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def bench_attn_decode():
    for B, Hq, Hkv, ctx, D in [(32, 32, 8, 2048, 128), (32, 32, 8, 8192, 128),
                               (8, 32, 8, 2048, 128), (64, 32, 8, 2048, 128)]:
        P = 16
        maxp = (ctx + P - 1) // P
        npages = B * maxp + 1
        kp = torch.randn(npages, Hkv, P, D, dtype=torch.bfloat16, device=DEV)
        vp = torch.randn(kp.shape[0], Hkv, D, P, dtype=torch.bfloat16, device=DEV)
        pt = torch.arange(B * maxp, dtype=torch.int32, device=DEV).reshape(B, maxp)
        q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device=DEV) * 0.1
        ctx_l = torch.full((B,), ctx, dtype=torch.int32, device=DEV)
        bytes_ = B * Hkv * ctx * 2 * D * 2
        for ns in (0, 2, 4, 8, 16):
            t = timeit(lambda: ops.attn_decode(q, kp, vp, pt, ctx_l, n_split=ns))
            print(f"attn_decode B{B} Hq{Hq}/{Hkv} ctx{ctx} D{D} ns={ns}: "
                  f"{t*1e6:8.1f} us  {bytes_/t/1e12:6.2f} TB/s")


def bench_attn_prefill():
    for B, Hq, Hkv, T, D in [(32, 32, 8, 512, 128), (8, 32, 8, 2048, 128)]:
        P = 16
        maxp = (T + P - 1) // P
        npages = B * maxp + 1
        kp = torch.randn(npages, Hkv, P, D, dtype=torch.bfloat16, device=DEV)
        vp = torch.randn(kp.shape[0], Hkv, D, P, dtype=torch.bfloat16, device=DEV)
        pt = torch.arange(B * maxp, dtype=torch.int32, device=DEV).reshape(B, maxp)
        q = torch.randn(B, Hq, T, D, dtype=torch.bfloat16, device=DEV) * 0.1
        qs = torch.zeros(B, dtype=torch.int32, device=DEV)
        t = timeit(lambda: ops.attn_prefill(q, kp, vp, pt, qs), iters=20)
        flops = B * Hq * D * 2 * 2 * (T * (T + 1) / 2)  # causal
        print(f"attn_prefill B{B} Hq{Hq}/{Hkv} T{T} D{D}: "
              f"{t*1e6:8.1f} us  {flops/t/1e12:7.1f} TF/s")


def bench_kv_stream():
    from bloombee_amd.ops import hip_ops
    B, Hkv, ctx, D, P = 32, 8, 2048, 128, 16
    maxp = ctx // P
    npages = B * maxp + 1
    kp = torch.randn(npages, Hkv, P, D, dtype=torch.bfloat16, device=DEV)
    vp = torch.randn(kp.shape[0], Hkv, D, P, dtype=torch.bfloat16, device=DEV)
    pt = torch.arange(B * maxp, dtype=torch.int32, device=DEV).reshape(B, maxp)
    ctx_l = torch.full((B,), ctx, dtype=torch.int32, device=DEV)
    bytes_ = B * Hkv * ctx * 2 * D * 2
    for ns in (2, 4, 8, 16):
        for nt in (False, True):
            t = timeit(lambda: hip_ops.kv_stream_probe(kp, vp, pt, ctx_l, ns, nt))
            print(f"kv_stream ns{ns} nt{int(nt)}: {t*1e6:7.1f} us  {bytes_/t/1e12:5.2f} TB/s")


def bench_norms():
    for N, H in [(32, 4096), (16384, 4096)]:
        x = torch.randn(N, H, dtype=torch.bfloat16, device=DEV)
        r = torch.randn_like(x)
        w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: ops.rms_norm(x, w))
        print(f"rms_norm ({N},{H}): {t*1e6:7.1f} us  {2*N*H*2/t/1e12:5.2f} TB/s")
        t = timeit(lambda: ops.rms_norm_residual(x, r, w))
        print(f"rms_norm_res ({N},{H}): {t*1e6:7.1f} us  {5*N*H*2/t/1e12:5.2f} TB/s")


def bench_swiglu():
    for N, I in [(32, 14336), (16384, 14336)]:
        gu = torch.randn(N, 2 * I, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: ops.swiglu(gu))
        print(f"swiglu ({N},{I}): {t*1e6:7.1f} us  {3*N*I*2/t/1e12:5.2f} TB/s")


def bench_gemm():
    """hipBLASLt skinny-M decode GEMMs baseline."""
    for M, N, K, tag in [(32, 6144, 4096, "qkv"), (32, 4096, 4096, "o"),
                         (32, 28672, 4096, "gate_up"), (32, 4096, 14336, "down"),
                         (32, 128256, 4096, "lm_head")]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: torch.nn.functional.linear(x, w))
        print(f"gemm[blaslt] {tag} {M}x{N}x{K}: {t*1e6:7.1f} us  "
              f"{N*K*2/t/1e12:5.2f} TB/s(wt)  {2*M*N*K/t/1e12:6.1f} TF/s")
        if N % 64 == 0:
            for ks in (0, 1, 2, 4, 8, 16):
                ts = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, ks))
                print(f"gemm[skinny ks={ks}] {tag}: {ts*1e6:7.1f} us  "
                      f"{N*K*2/ts/1e12:5.2f} TB/s(wt)")



def bench_gemm_norm():
    """Fused-rmsnorm GEMM A/B per llama decode shape: overhead of the
    per-WG prepass+scale vs the separate rms_norm launch it replaces."""
    for M, N, K, tag in [(32, 6144, 4096, "qkv"), (32, 4096, 4096, "o"),
                         (32, 28672, 4096, "gate_up"), (32, 4096, 14336, "down")]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        nw = torch.randn(K, dtype=torch.bfloat16, device=DEV)
        t0 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, 0))
        t1 = timeit(lambda: ops.hip_ops.gemm_skinny(x, w, None, None, 0, nw, 1e-5))
        tn = timeit(lambda: ops.hip_ops.rms_norm(x, nw, 1e-5))
        print(f"gemm_norm {tag} {M}x{N}x{K}: plain {t0*1e6:6.1f} us  "
              f"fused {t1*1e6:6.1f} us (+{(t1-t0)*1e6:5.1f})  "
              f"rms_norm alone {tn*1e6:5.1f} us")
    h = torch.randn(32, 4096, dtype=torch.bfloat16, device=DEV)
    a = torch.randn(32, 4096, dtype=torch.bfloat16, device=DEV)
    nw2 = torch.randn(4096, dtype=torch.bfloat16, device=DEV)
    tr = timeit(lambda: ops.hip_ops.rms_norm_residual(a, h, nw2, 1e-5))
    print(f"rms_norm_residual 32x4096: {tr*1e6:5.1f} us")



ALL = {
    "attn_decode": bench_attn_decode,
    "attn_prefill": bench_attn_prefill,
    "kv_stream": bench_kv_stream,
    "norms": bench_norms,
    "swiglu": bench_swiglu,
    "gemm": bench_gemm,
    "gemm_norm": bench_gemm_norm,
}

if __name__ == "__main__":
    if not torch.cuda.is_available():
        sys.exit("bench_kernels measures the gfx950 HIP kernels — "
                 "run on a GPU box (e.g. via gpurun)")
    which = sys.argv[1:] or list(ALL)
    for name in which:
        ALL[name]()
