"""Llama-3 family architecture, built natively for the MI355X decode path.

Random-init from a local config (no network => no pretrained weights,
BASELINE.md protocol).  MI355X-first choices:
  - fused qkv_proj and gate_up_proj Linears (one GEMV launch each at
    decode instead of 3/2 — launch overhead is the TP=8 risk,
    SURVEY.md §2.5)
  - static KV cache + pure-tensor decode step, hipGraph-capturable
  - attention decode / rmsnorm / rope / silu·mul via the gfx950 HIP
    kernels (lfmq.ops.functional)

TP sharding (applied by lfmq.dist.tp): qkv/gate_up/lm_head column-split,
o_proj/down_proj row-split with all-reduce.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field, asdict
from typing import Optional

import torch
import torch.nn as nn

from ..ops import functional as F
from .kvcache import KVCache


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    head_dim: int = 128
    rms_eps: float = 1e-5
    rope_theta: float = 500000.0
    max_seq: int = 8192
    tie_embeddings: bool = False
    arch: str = "llama"

    @classmethod
    def llama3_8b(cls, **kw):
        return cls(**{**dict(vocab_size=128256, hidden_size=4096,
                             intermediate_size=14336, n_layers=32, n_heads=32,
                             n_kv_heads=8), **kw})

    @classmethod
    def llama3_70b(cls, **kw):
        return cls(**{**dict(vocab_size=128256, hidden_size=8192,
                             intermediate_size=28672, n_layers=80, n_heads=64,
                             n_kv_heads=8), **kw})

    @classmethod
    def tiny(cls, **kw):
        """2-layer 8B-shaped-per-layer test model."""
        return cls(**{**dict(vocab_size=512, hidden_size=256,
                             intermediate_size=512, n_layers=2, n_heads=4,
                             n_kv_heads=2, head_dim=64, max_seq=512), **kw})

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "LlamaConfig":
        d = {k: v for k, v in d.items() if k in cls.__dataclass_fields__}
        return cls(**d)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float, dtype=torch.bfloat16):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim, dtype=dtype))

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        return F.rmsnorm(x, self.weight, self.eps, residual)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype):
        super().__init__()
        self.cfg = cfg
        self.n_heads = cfg.n_heads
        self.n_kv_heads = cfg.n_kv_heads
        self.head_dim = cfg.head_dim
        qkv_out = (cfg.n_heads + 2 * cfg.n_kv_heads) * cfg.head_dim
        self.qkv_proj = nn.Linear(cfg.hidden_size, qkv_out, bias=False, dtype=dtype)
        self.o_proj = nn.Linear(cfg.n_heads * cfg.head_dim, cfg.hidden_size,
                                bias=False, dtype=dtype)
        self.scale = 1.0 / math.sqrt(cfg.head_dim)

    def forward(self, x: torch.Tensor, cache_k: torch.Tensor,
                cache_v: torch.Tensor, positions: torch.Tensor,
                seq_len_after: int) -> torch.Tensor:
        b, s, _ = x.shape
        hq, hkv, d = self.n_heads, self.n_kv_heads, self.head_dim
        qkv = self.qkv_proj(x)
        pos0 = seq_len_after - s
        if x.is_cuda:
            # fused rope + KV append straight off the qkv buffer (one launch)
            q = F.qkv_rope_append(qkv, cache_k, cache_v, positions, pos0, hq,
                                  self.cfg.rope_theta)
            if s == 1:
                o = F.attn_decode(q.view(b, hq, 1, d), cache_k, cache_v,
                                  seq_len_after, self.scale)
                o = o.view(b, s, hq * d)
            elif F.USE_FLASH_PREFILL and d == 128:
                # one-kernel MFMA flash prefill; handles pos0 > 0 too
                # (LFMQ_FLASH_PREFILL=0 restores the matmul-chain path)
                o = F.attn_prefill_fused(q, cache_k, cache_v, pos0,
                                         self.scale).reshape(b, s, hq * d)
            else:
                qh = q.transpose(1, 2)
                kc = cache_k[:, :, pos0:seq_len_after]
                vc = cache_v[:, :, pos0:seq_len_after]
                if pos0 == 0:
                    o = F.attn_prefill(qh, kc, vc, causal=True, scale=self.scale)
                else:
                    o = self._chunk_attn(qh, cache_k, cache_v, pos0, seq_len_after)
                o = o.transpose(1, 2).reshape(b, s, hq * d)
            return self.o_proj(o)
        q, k, v = qkv.split([hq * d, hkv * d, hkv * d], dim=-1)
        q = q.reshape(b, s, hq, d)
        k = k.reshape(b, s, hkv, d)
        v = v.reshape(b, s, hkv, d)
        q, k = F.apply_rope(q, k, positions, self.cfg.rope_theta)
        F.kv_append(cache_k, cache_v, k, v, pos0)
        qh = q.transpose(1, 2)  # [b, hq, s, d]
        if s == 1:
            o = F.attn_decode(qh, cache_k, cache_v, seq_len_after, self.scale)
        elif pos0 == 0:
            o = F.attn_prefill(qh, k.transpose(1, 2), v.transpose(1, 2),
                               causal=True, scale=self.scale)
        else:  # chunked continuation: attend to full cache, causal inside chunk
            o = self._chunk_attn(qh, cache_k, cache_v, pos0, seq_len_after)
        o = o.transpose(1, 2).reshape(b, s, hq * d)
        return self.o_proj(o)

    def forward_decode_graph(self, x, cache_k, cache_v, state, max_seq):
        """hipGraph-capturable decode step (s == 1, GPU only)."""
        b, s, _ = x.shape
        o = self.decode_qkv_attn(x, cache_k, cache_v, state, max_seq)
        return self.o_proj(o)

    def decode_qkv_attn(self, x, cache_k, cache_v, state, max_seq,
                        rms_eps: float = -1.0):
        """qkv GEMV -> ONE fused rope+KV-append+attention launch; returns
        [b, s, hq*d] (pre o_proj).  rms_eps >= 0 additionally fuses the
        input RMSNorm into the qkv GEMV (norm weight pre-folded)."""
        b, s, _ = x.shape
        if rms_eps >= 0:
            qkv = self.qkv_proj.forward_rms(x, rms_eps)
        else:
            qkv = self.qkv_proj(x)
        o = F.attn_decode_fused(qkv.view(b, -1), cache_k, cache_v, state,
                                max_seq, self.scale, self.n_heads,
                                self.cfg.rope_theta)
        return o.view(b, s, self.n_heads * self.head_dim)

    def _chunk_attn(self, q, cache_k, cache_v, pos0, seq_len_after):
        b, hq, s, d = q.shape
        hkv = cache_k.shape[1]
        rep = hq // hkv
        kc = cache_k[:, :, :seq_len_after].float().repeat_interleave(rep, dim=1)
        vc = cache_v[:, :, :seq_len_after].float().repeat_interleave(rep, dim=1)
        sc = torch.einsum("bhqd,bhtd->bhqt", q.float(), kc) * self.scale
        t_idx = torch.arange(seq_len_after, device=q.device)
        q_idx = pos0 + torch.arange(s, device=q.device)
        mask = t_idx[None, :] > q_idx[:, None]
        sc = sc.masked_fill(mask, float("-inf"))
        p = torch.softmax(sc, dim=-1)
        return torch.einsum("bhqt,bhtd->bhqd", p, vc).to(q.dtype)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype):
        super().__init__()
        self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                                      bias=False, dtype=dtype)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size,
                                   bias=False, dtype=dtype)
        self.inter = cfg.intermediate_size

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from .qlinear import QLinear
        gp = self.gate_up_proj
        m = x.numel() // x.shape[-1]
        if (x.is_cuda and m <= 4 and type(gp).forward is QLinear.forward
                and isinstance(gp, QLinear) and not hasattr(gp, "perm")):
            # fused gate_up GEMV + silu·mul epilogue (no separate pass)
            h = F.dq_gemv_gateup(x, gp.qweight, gp.scales, gp.qzeros,
                                 gp.bits, gp.group_size)
            return self.down_proj(h)
        gu = gp(x)
        return self.down_proj(F.silu_mul_gateup(gu, self.inter))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.self_attn = LlamaAttention(cfg, dtype)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.mlp = LlamaMLP(cfg, dtype)

    def forward(self, x, cache_k, cache_v, positions, seq_len_after,
                residual=None):
        """Fused-residual form: returns (mlp_out, residual_after_attn) so
        the NEXT norm folds the pending add (one rmsnorm_residual kernel
        instead of add + rmsnorm).  residual=None => x is the raw input."""
        if residual is None:
            residual = x
            h = self.input_layernorm(x)
        else:
            h, residual = F.rmsnorm_residual(x, residual,
                                             self.input_layernorm.weight,
                                             self.input_layernorm.eps)
        h = self.self_attn(h, cache_k, cache_v, positions, seq_len_after)
        h, residual = F.rmsnorm_residual(h, residual,
                                         self.post_attention_layernorm.weight,
                                         self.post_attention_layernorm.eps)
        h = self.mlp(h)
        return h, residual

    def forward_decode_step(self, x, cache_k, cache_v, state, max_seq,
                            residual=None):
        if residual is None:
            residual = x
            h = self.input_layernorm(x)
        else:
            h, residual = F.rmsnorm_residual(x, residual,
                                             self.input_layernorm.weight,
                                             self.input_layernorm.eps)
        h = self.self_attn.forward_decode_graph(h, cache_k, cache_v, state,
                                                max_seq)
        h, residual = F.rmsnorm_residual(h, residual,
                                         self.post_attention_layernorm.weight,
                                         self.post_attention_layernorm.eps)
        h = self.mlp(h)
        return h, residual

    def forward_decode_fused(self, x, cache_k, cache_v, state, max_seq):
        """Folded-norm decode step: both RMSNorms ride inside the GEMVs
        and both residual adds ride in the o/down epilogues — 6 kernels
        per layer, no separate norm/add launches.  Requires
        _in_norm_folded and _post_norm_folded (lfmq.quant.fold)."""
        eps = self.input_layernorm.eps
        o = self.self_attn.decode_qkv_attn(x, cache_k, cache_v, state,
                                           max_seq, rms_eps=eps)
        x = self.self_attn.o_proj.forward_resid(o, x)
        mlp = self.mlp
        h = F.dq_gemv_gateup(x, mlp.gate_up_proj.qweight,
                             mlp.gate_up_proj.scales, mlp.gate_up_proj.qzeros,
                             mlp.gate_up_proj.bits, mlp.gate_up_proj.group_size,
                             rms_eps=self.post_attention_layernorm.eps)
        x = mlp.down_proj.forward_resid(h, x)
        return x


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(LlamaDecoderLayer(cfg, dtype)
                                    for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False, dtype=dtype)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed_tokens.weight

    @torch.no_grad()
    def init_random(self, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        for p in self.parameters():
            p.data.copy_(torch.randn(p.shape, generator=g, dtype=torch.float32)
                         .mul_(0.02).to(p.dtype))
        return self

    def make_cache(self, batch: int, max_seq: Optional[int] = None,
                   device=None) -> KVCache:
        dev = device or next(self.parameters()).device
        # TP: layers may hold a KV-head shard; size the cache from the module
        hkv = self.layers[0].self_attn.n_kv_heads
        return KVCache(self.cfg.n_layers, batch, hkv,
                       max_seq or self.cfg.max_seq, self.cfg.head_dim, dev,
                       dtype=self.dtype)

    def forward(self, input_ids: torch.Tensor, cache: KVCache,
                positions: Optional[torch.Tensor] = None) -> torch.Tensor:
        """input_ids [b, s] -> logits [b, s, vocab].  Advances the cache."""
        b, s = input_ids.shape
        pos0 = cache.seq_len
        if positions is None:
            positions = torch.arange(pos0, pos0 + s, device=input_ids.device)
        seq_after = pos0 + s
        x = self.embed_tokens(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, cache.k[i], cache.v[i], positions,
                                seq_after, residual)
        x = self.norm(x, residual=residual)
        logits = self.lm_head(x)
        cache.advance(s)
        return logits

    def _engine_for(self, cache, batch: int) -> bool:
        """Build/bind the persistent decode engine for this cache (lazy;
        one stream build per cache).  False => launched-kernel path.
        The env opt-in is re-read every step so LFMQ_ENGINE can be
        toggled within one process; only STRUCTURAL unsupport (arch /
        shape) latches."""
        import os
        if os.environ.get("LFMQ_ENGINE", "0") != "1":
            return False
        eng = getattr(self, "_engine", None)
        if eng is not None and eng.cache is cache:
            return True
        if getattr(self, "_engine_unsupported", False):
            return False
        from .engine import DecodeEngine, engine_supported
        if not engine_supported(self, batch):
            self._engine_unsupported = True
            return False
        self._engine = DecodeEngine(self, cache, cache.max_seq)
        return True

    def _decode_fused_ready(self) -> bool:
        from .qlinear import QLinear
        for layer in self.layers:
            if not (getattr(layer, "_in_norm_folded", False)
                    and getattr(layer, "_post_norm_folded", False)):
                return False
            if not isinstance(layer.self_attn.qkv_proj, QLinear):
                return False
            for proj in (layer.self_attn.qkv_proj, layer.self_attn.o_proj,
                         layer.mlp.gate_up_proj, layer.mlp.down_proj):
                if hasattr(proj, "perm"):
                    # act-order: inputs need the perm gather — QLinear
                    # .forward handles it; the fused 6-kernel chain bypasses
                    # QLinear.forward, so fall back to the step path
                    return False
        return True

    def forward_decode_step(self, input_ids, cache, state, max_seq):
        """One hipGraph-capturable decode step: [b,1] ids -> [b,1,vocab]."""
        # persistent weight-streaming engine: the whole layer stack in ONE
        # kernel launch (csrc/engine.hip); lm_head stays a separate launch.
        if input_ids.is_cuda and self._engine_for(cache, input_ids.shape[0]):
            x = self._engine.step(input_ids, state)
            return self.lm_head(x)
        x = self.embed_tokens(input_ids)
        if getattr(self, "_fused_decode", None) is None:
            self._fused_decode = self._decode_fused_ready()
        # the rms/silu-fused GEMV variants are m <= 4 kernels
        if self._fused_decode and input_ids.shape[0] <= 4:
            for i, layer in enumerate(self.layers):
                x = layer.forward_decode_fused(x, cache.k[i], cache.v[i],
                                               state, max_seq)
            x = self.norm(x)
        else:
            residual = None
            for i, layer in enumerate(self.layers):
                x, residual = layer.forward_decode_step(
                    x, cache.k[i], cache.v[i], state, max_seq, residual)
            x = self.norm(x, residual=residual)
        return self.lm_head(x)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 cache: Optional[KVCache] = None, gen_cfg=None,
                 generator=None) -> torch.Tensor:
        from .generation import autoregressive_generate
        return autoregressive_generate(self, input_ids, max_new_tokens,
                                       cache, gen_cfg, generator)
