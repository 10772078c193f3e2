"""Static per-layer KV cache sized for 288 GB of HBM3E per GPU.

Preallocated [b, hkv, T_max, d] bf16 tensors — static addresses so the
whole decode step can be captured into a hipGraph.
"""
from __future__ import annotations

import torch


class KVCache:
    def __init__(self, n_layers: int, batch: int, n_kv_heads: int,
                 max_seq: int, head_dim: int, device, dtype=torch.bfloat16):
        self.max_seq = max_seq
        self.seq_len = 0
        self.k = [torch.zeros(batch, n_kv_heads, max_seq, head_dim,
                              device=device, dtype=dtype) for _ in range(n_layers)]
        self.v = [torch.zeros(batch, n_kv_heads, max_seq, head_dim,
                              device=device, dtype=dtype) for _ in range(n_layers)]

    def reset(self) -> None:
        self.seq_len = 0

    def advance(self, n: int) -> None:
        self.seq_len += n
        if self.seq_len > self.max_seq:
            raise RuntimeError(f"KV cache overflow: {self.seq_len} > {self.max_seq}")
