"""Static per-layer KV cache sized for 288 GB of HBM3E per GPU.

One stacked [L, b, hkv, T_max, d] bf16 allocation per side; `k[i]`/`v[i]`
are contiguous per-layer views (static addresses so the whole decode step
can be captured into a hipGraph — SURVEY.md §3.3).  The stacked base
tensors (`k_all`/`v_all`) give the persistent decode engine a single
pointer + layer stride (lfmq/models/engine.py).
"""
from __future__ import annotations

import torch


class KVCache:
    def __init__(self, n_layers: int, batch: int, n_kv_heads: int,
                 max_seq: int, head_dim: int, device, dtype=torch.bfloat16):
        self.max_seq = max_seq
        self.seq_len = 0
        self.k_all = torch.zeros(n_layers, batch, n_kv_heads, max_seq,
                                 head_dim, device=device, dtype=dtype)
        self.v_all = torch.zeros(n_layers, batch, n_kv_heads, max_seq,
                                 head_dim, device=device, dtype=dtype)
        self.k = [self.k_all[i] for i in range(n_layers)]
        self.v = [self.v_all[i] for i in range(n_layers)]

    def reset(self) -> None:
        self.seq_len = 0

    def advance(self, n: int) -> None:
        self.seq_len += n
        if self.seq_len > self.max_seq:
            raise RuntimeError(f"KV cache overflow: {self.seq_len} > {self.max_seq}")
