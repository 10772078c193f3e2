"""Time the experimental MFMA flash prefill (csrc/attn_prefill.hip)
against the rocBLAS matmul-chain path on llama-8B attention shapes.

    gpurun -- 'timeout 100 python tools/prefill_flash_bench.py'
"""
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lfmq.ops import functional as F

DEV = "cuda:0"


def timeit(fn, iters=30):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    b, hq, hkv, d = 1, 32, 8, 128
    scale = 1.0 / math.sqrt(d)
    for s in (256, 512, 2048):
        tmax = s + 64
        torch.manual_seed(0)
        q = torch.randn(b, s, hq, d, dtype=torch.bfloat16, device=DEV)
        kc = torch.randn(b, hkv, tmax, d, dtype=torch.bfloat16, device=DEV)
        vc = torch.randn(b, hkv, tmax, d, dtype=torch.bfloat16, device=DEV)

        def chain():
            qh = q.transpose(1, 2)
            return F.attn_prefill(qh, kc[:, :, :s], vc[:, :, :s],
                                  causal=True, scale=scale)

        def flash():
            return F.attn_prefill_fused(q, kc, vc, 0, scale)

        # parity spot-check before timing
        o_c = chain().transpose(1, 2).float()
        o_f = flash().float()
        err = float((o_c - o_f).abs().max() / (o_c.abs().max() + 1e-8))
        t_c = timeit(chain)
        t_f = timeit(flash)
        print(f"s={s:5d}  chain {t_c:7.3f} ms  flash {t_f:7.3f} ms  "
              f"speedup {t_c / t_f:5.2f}x  rel_err {err:.2e}", flush=True)


if __name__ == "__main__":
    main()
