"""Prefill (dq_gemm path) probe for rocprof capture."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from lfmq.bench.decode import build_quantized_llama
m, cfg, _ = build_quantized_llama("llama3-8b", "rtn", 4, 128, "cuda:0")
ids = torch.randint(0, cfg.vocab_size, (1, 2048), device="cuda:0")
for _ in range(3):
    cache = m.make_cache(1, 2112, device="cuda:0")
    with torch.no_grad():
        m(ids, cache)
torch.cuda.synchronize()
print("prefill probe done")
