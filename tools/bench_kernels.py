#!/usr/bin/env python3
"""Per-kernel microbenchmarks on the decode shapes (8B, TP=1).

Prints effective bandwidth of each GEMV call shape and timings of the
other decode kernels.  Run on the GPU box:
    python tools/bench_kernels.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lfmq.ops import hip_ext, ref  # noqa: E402

DEV = "cuda:0"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemv(m, n, k, g, bits=4, gateup=False):
    per = 32 // bits
    qw = torch.randint(-2**31, 2**31 - 1, (n, k // per), dtype=torch.int32, device=DEV)
    G = k // g
    sc = (torch.rand(n, G, device=DEV) * 0.01).to(torch.bfloat16)
    qz = torch.randint(-2**31, 2**31 - 1, (n, (G + per - 1) // per), dtype=torch.int32, device=DEV)
    x = torch.randn(m, k, device=DEV).to(torch.bfloat16)
    ext = hip_ext()
    for cols in (2, 4, 6):
        if gateup:
            if cols in (1, 5, 6):
                continue
            fn = lambda: ext.dq_gemv_gateup(x, qw, sc, qz, bits, g, -1.0, cols)
        else:
            fn = lambda: ext.dq_linear(x, qw, sc, qz, bits, g, None, None, -1.0, cols)
        t = timeit(fn)
        wbytes = n * k * bits / 8 + n * G * 2 + qz.numel() * 4
        print(f"gemv{'-gu' if gateup else '   '} m={m:3d} n={n:6d} k={k:6d} g={g} "
              f"w{bits} cols={cols}: {t*1e6:8.2f} us  {wbytes/t/1e12:6.2f} TB/s")


def main():
    assert torch.cuda.is_available()
    print("== dq_gemv decode shapes (Llama-3-8B) ==")
    for shape in [(1, 6144, 4096, 128), (1, 4096, 4096, 128),
                  (1, 4096, 14336, 128), (1, 128256, 4096, 128)]:
        bench_gemv(*shape)
    bench_gemv(1, 28672, 4096, 128, gateup=True)
    print("== batch sweep (qkv shape) ==")
    for m in (1, 2, 4, 8, 16):
        bench_gemv(m, 6144, 4096, 128)
    print("== w8 ==")
    bench_gemv(1, 4096, 14336, 128, bits=8)

    print("== dq_gemm prefill shapes ==")
    ext = hip_ext()
    for (m, n, k) in [(2048, 6144, 4096), (2048, 4096, 14336)]:
        per, g = 8, 128
        G = k // g
        qw = torch.randint(-2**31, 2**31 - 1, (n, k // per), dtype=torch.int32, device=DEV)
        sc = (torch.rand(n, G, device=DEV) * 0.01).to(torch.bfloat16)
        qz = torch.randint(-2**31, 2**31 - 1, (n, (G + 7) // 8), dtype=torch.int32, device=DEV)
        x = torch.randn(m, k, device=DEV).to(torch.bfloat16)
        t = timeit(lambda: ext.dq_linear(x, qw, sc, qz, 4, g, None, None, -1.0, 0), iters=20, warmup=5)
        tf = 2 * m * n * k / t / 1e12
        print(f"gemm m={m} n={n} k={k}: {t*1e6:8.1f} us  {tf:6.1f} TF/s")

    print("== attn_decode (b=1, hq=32, hkv=8, d=128) ==")
    ext = hip_ext()
    for T in (256, 640, 2048, 8192):
        tmax = T + 64
        q = torch.randn(1, 32, 1, 128, device=DEV).to(torch.bfloat16)
        kc = torch.randn(1, 8, tmax, 128, device=DEV).to(torch.bfloat16)
        vc = torch.randn_like(kc)
        t = timeit(lambda: ext.attn_decode(q, kc, vc, T, 0.0883, None))
        bytes_ = 2 * 8 * T * 128 * 2
        print(f"attn T={T:5d}: {t*1e6:8.2f} us  ({bytes_/t/1e12:5.2f} TB/s cache read)")

    print("== rmsnorm/argmax/rope-append (decode shapes) ==")
    x = torch.randn(1, 4096, device=DEV).to(torch.bfloat16)
    r = torch.randn_like(x)
    w = torch.randn(4096, device=DEV).to(torch.bfloat16)
    t = timeit(lambda: ext.rmsnorm_residual(x, r, w, 1e-5))
    print(f"rmsnorm_residual [1,4096]: {t*1e6:6.2f} us")
    logits = torch.randn(1, 128256, device=DEV).to(torch.bfloat16)
    t = timeit(lambda: ext.greedy_sample(logits))
    print(f"greedy_sample [1,128256]: {t*1e6:6.2f} us")
    qkv = torch.randn(1, 1, 6144, device=DEV).to(torch.bfloat16)
    kc = torch.zeros(1, 8, 1024, 128, device=DEV).to(torch.bfloat16)
    vc = torch.zeros_like(kc)
    pos = torch.tensor([5], dtype=torch.int32, device=DEV)
    inv = ref.rope_freqs(128).to(DEV)
    t = timeit(lambda: ext.qkv_rope_append(qkv, kc, vc, pos, inv, 32, 5, None))
    print(f"qkv_rope_append decode: {t*1e6:6.2f} us")


def bench_read():
    """Raw-read ceilings in the GEMV access pattern."""
    import lfmq_hip
    sink = torch.zeros(1, device=DEV)
    for gb, row_kb in [(4, 2), (4, 8)]:
        row_bytes = row_kb * 1024
        n_rows = gb * 2**30 // row_bytes
        src = torch.randint(-2**31, 2**31-1, (gb * 2**28,), dtype=torch.int32, device=DEV)
        for streams, nt, mode in [(4, 2, 0), (4, 2, 1), (4, 2, 2),
                                  (4, 0, 2), (8, 2, 0), (1, 2, 0)]:
            def fn():
                lfmq_hip.read_probe(src, sink, row_bytes, n_rows, nt, streams, mode)
            t = timeit(fn, iters=10, warmup=3)
            print(f"read row={row_kb}KB streams={streams} nt={nt} mode={mode}: "
                  f"{gb/t/1e3:6.2f} TB/s")


if __name__ == "__main__":
    main()
    bench_read()
