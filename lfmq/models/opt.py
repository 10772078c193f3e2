"""OPT-125m — the plumbing model for BASELINE.json config #1:
per-tensor int8 RTN quantize + CPU matmul decode (no GPU involved).

LayerNorm / ReLU architecture, learned positional embeddings (offset 2,
as OPT does), pre-LN.  Random-init; runs entirely in local pytest.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, asdict
from typing import Optional

import torch
import torch.nn as nn

from .kvcache import KVCache


@dataclass
class OPTConfig:
    vocab_size: int = 50272
    hidden_size: int = 768
    ffn_dim: int = 3072
    n_layers: int = 12
    n_heads: int = 12
    max_pos: int = 2048
    arch: str = "opt"

    @classmethod
    def opt_125m(cls, **kw):
        return cls(**kw)

    @classmethod
    def tiny(cls, **kw):
        return cls(**{**dict(vocab_size=256, hidden_size=64, ffn_dim=128,
                             n_layers=2, n_heads=4, max_pos=128), **kw})

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "OPTConfig":
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


class OPTAttention(nn.Module):
    def __init__(self, cfg: OPTConfig, dtype):
        super().__init__()
        self.n_heads = cfg.n_heads
        self.head_dim = cfg.hidden_size // cfg.n_heads
        h = cfg.hidden_size
        self.q_proj = nn.Linear(h, h, dtype=dtype)
        self.k_proj = nn.Linear(h, h, dtype=dtype)
        self.v_proj = nn.Linear(h, h, dtype=dtype)
        self.out_proj = nn.Linear(h, h, dtype=dtype)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, x, cache_k, cache_v, seq_len_after):
        b, s, h = x.shape
        nh, d = self.n_heads, self.head_dim
        q = self.q_proj(x).reshape(b, s, nh, d).transpose(1, 2)
        k = self.k_proj(x).reshape(b, s, nh, d).transpose(1, 2)
        v = self.v_proj(x).reshape(b, s, nh, d).transpose(1, 2)
        pos0 = seq_len_after - s
        cache_k[:, :, pos0:seq_len_after] = k.to(cache_k.dtype)
        cache_v[:, :, pos0:seq_len_after] = v.to(cache_v.dtype)
        kc = cache_k[:, :, :seq_len_after].float()
        vc = cache_v[:, :, :seq_len_after].float()
        sc = torch.einsum("bhqd,bhtd->bhqt", q.float(), kc) * self.scale
        t_idx = torch.arange(seq_len_after, device=x.device)
        q_idx = pos0 + torch.arange(s, device=x.device)
        sc = sc.masked_fill(t_idx[None, :] > q_idx[:, None], float("-inf"))
        p = torch.softmax(sc, dim=-1)
        o = torch.einsum("bhqt,bhtd->bhqd", p, vc).to(x.dtype)
        o = o.transpose(1, 2).reshape(b, s, h)
        return self.out_proj(o)


class OPTLayer(nn.Module):
    def __init__(self, cfg: OPTConfig, dtype):
        super().__init__()
        h = cfg.hidden_size
        self.self_attn_layer_norm = nn.LayerNorm(h, dtype=dtype)
        self.self_attn = OPTAttention(cfg, dtype)
        self.final_layer_norm = nn.LayerNorm(h, dtype=dtype)
        self.fc1 = nn.Linear(h, cfg.ffn_dim, dtype=dtype)
        self.fc2 = nn.Linear(cfg.ffn_dim, h, dtype=dtype)

    def forward(self, x, cache_k, cache_v, seq_len_after):
        r = x
        x = self.self_attn_layer_norm(x)
        x = r + self.self_attn(x, cache_k, cache_v, seq_len_after)
        r = x
        x = self.final_layer_norm(x)
        x = r + self.fc2(torch.relu(self.fc1(x)))
        return x


class OPTForCausalLM(nn.Module):
    def __init__(self, cfg: OPTConfig, dtype=torch.float32):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.embed_positions = nn.Embedding(cfg.max_pos + 2, cfg.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(OPTLayer(cfg, dtype) for _ in range(cfg.n_layers))
        self.final_layer_norm = nn.LayerNorm(cfg.hidden_size, dtype=dtype)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False, dtype=dtype)

    @torch.no_grad()
    def init_random(self, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        for p in self.parameters():
            p.data.copy_(torch.randn(p.shape, generator=g, dtype=torch.float32)
                         .mul_(0.02).to(p.dtype))
        return self

    def make_cache(self, batch: int, max_seq: Optional[int] = None, device=None):
        dev = device or next(self.parameters()).device
        return KVCache(self.cfg.n_layers, batch, self.cfg.n_heads,
                       max_seq or self.cfg.max_pos,
                       self.cfg.hidden_size // self.cfg.n_heads, dev,
                       dtype=self.dtype)

    def forward(self, input_ids, cache: KVCache):
        b, s = input_ids.shape
        pos0 = cache.seq_len
        positions = torch.arange(pos0, pos0 + s, device=input_ids.device) + 2
        x = self.embed_tokens(input_ids) + self.embed_positions(positions)
        for i, layer in enumerate(self.layers):
            x = layer(x, cache.k[i], cache.v[i], pos0 + s)
        x = self.final_layer_norm(x)
        logits = self.lm_head(x)
        cache.advance(s)
        return logits

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 16, cache=None,
                 gen_cfg=None, generator=None):
        from .generation import autoregressive_generate
        return autoregressive_generate(self, input_ids, max_new_tokens,
                                       cache, gen_cfg, generator)
