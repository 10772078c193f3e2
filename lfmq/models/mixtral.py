"""Mixtral 8x7B MoE (BASELINE.json config #5).

Same attention/backbone as Llama (GQA, RoPE θ=1e6, RMSNorm) with the MLP
replaced by a top-2 router over 8 experts.  Experts use the fused
w13 (gate|up) + w2 (down) projections like the Llama MLP, so GPTQ/AWQ/RTN
quantize each expert Linear exactly like any other layer.

GPU decode path: tokens are routed, sorted by expert on-device, and run
through the grouped dequant-GEMM (csrc/grouped_dq_gemm.hip) — one launch
for all experts' w13 (with the silu·mul epilogue fused) and one for w2 —
instead of 8 per-expert launches.  With expert
parallelism (lfmq/dist/ep.py) tokens go through the all-to-all dispatch/
combine pair instead.
"""
from __future__ import annotations

from dataclasses import dataclass, asdict
from typing import Optional

import torch
import torch.nn as nn

from ..ops import functional as F
from .kvcache import KVCache
from .llama import LlamaAttention, LlamaConfig, RMSNorm
from .qlinear import QLinear


@dataclass
class MixtralConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 14336
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    head_dim: int = 128
    n_experts: int = 8
    top_k: int = 2
    rms_eps: float = 1e-5
    rope_theta: float = 1000000.0
    max_seq: int = 8192
    arch: str = "mixtral"

    @classmethod
    def mixtral_8x7b(cls, **kw):
        return cls(**kw)

    @classmethod
    def tiny(cls, **kw):
        return cls(**{**dict(vocab_size=512, hidden_size=128,
                             intermediate_size=256, n_layers=2, n_heads=4,
                             n_kv_heads=2, head_dim=32, n_experts=4, top_k=2,
                             max_seq=256), **kw})

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "MixtralConfig":
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


class MixtralExpert(nn.Module):
    def __init__(self, cfg: MixtralConfig, dtype):
        super().__init__()
        self.w13 = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                             bias=False, dtype=dtype)  # [gate | up]
        self.w2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size,
                            bias=False, dtype=dtype)
        self.inter = cfg.intermediate_size

    def forward(self, x):
        gu = self.w13(x)
        return self.w2(F.silu_mul_gateup(gu, self.inter))


class MixtralMoE(nn.Module):
    """Top-k router + experts.  Forward modes:
      - dense loop (CPU / prefill): per-expert masked forward
      - grouped (GPU decode, consolidated buffers): one grouped
        dequant-GEMM launch pair for all experts
      - EP (world > 1): token all-to-all to the rank owning the expert
        (installed by lfmq.dist.ep.apply_expert_parallel)
    """

    def __init__(self, cfg: MixtralConfig, dtype):
        super().__init__()
        self.cfg = cfg
        self.n_experts = cfg.n_experts
        self.top_k = cfg.top_k
        self.gate = nn.Linear(cfg.hidden_size, cfg.n_experts, bias=False,
                              dtype=dtype)  # router stays dense bf16
        self.experts = nn.ModuleList(MixtralExpert(cfg, dtype)
                                     for _ in range(cfg.n_experts))
        self._grouped = None  # set by consolidate()

    def route(self, x2):
        logits = self.gate(x2).float()
        weights, ids = torch.topk(logits, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1)
        return weights, ids

    @torch.no_grad()
    def consolidate(self):
        """Stack the quantized experts' packed buffers for the grouped
        kernel: qweight [E, n, kw], scales [E, n, G], qzeros [E, n, zw]."""
        if not isinstance(self.experts[0].w13, QLinear):
            return None
        g = {}
        for name in ("w13", "w2"):
            mods = [getattr(e, name) for e in self.experts]
            g[name] = {
                "qweight": torch.stack([m.qweight for m in mods]),
                "scales": torch.stack([m.scales for m in mods]),
                "qzeros": torch.stack([m.qzeros for m in mods]),
                "bits": mods[0].bits, "g": mods[0].group_size,
                "n": mods[0].out_features, "k": mods[0].in_features,
            }
        self._grouped = g
        return g

    def forward(self, x):
        b, s, h = x.shape
        x2 = x.reshape(-1, h)
        t = x2.shape[0]
        if (x.is_cuda and self._grouped is None
                and isinstance(self.experts[0].w13, QLinear)):
            self.consolidate()  # lazy one-time stacking for the grouped kernel
        if x.is_cuda and self._grouped is not None and t * self.top_k <= 64:
            # routing runs inside _forward_grouped (one moe_route kernel)
            y2 = self._forward_grouped(x2, None, None)
        else:
            weights, ids = self.route(x2)
            y2 = self._forward_loop(x2, weights, ids)
        return y2.reshape(b, s, h)

    def _forward_loop(self, x2, weights, ids):
        y = torch.zeros_like(x2)
        for e in range(self.n_experts):
            mask = (ids == e)
            tok_mask = mask.any(dim=-1)
            if not bool(tok_mask.any()):
                continue
            xe = x2[tok_mask]
            ye = self.experts[e](xe)
            w = (weights * mask.float()).sum(dim=-1)[tok_mask]
            y[tok_mask] += ye * w.unsqueeze(-1).to(ye.dtype)
        return y

    def _forward_grouped(self, x2, weights, ids):
        """GPU decode: ONE routing kernel (moe_route: topk+softmax+counting
        sort), grouped gateup GEMV, then the w2 grouped GEMV with the
        weighted token scatter FUSED into its epilogue.  5 launches per MoE
        layer vs ~16 with the round-1 aten chain; all
        device-side (graph-capturable).  weights/ids args are unused here —
        routing runs on the raw router logits."""
        del weights, ids
        from ..ops import hip_ext
        t = x2.shape[0]
        logits16 = self.gate(x2)  # dense bf16 router GEMV
        src_tok, slot_w, offsets, active = hip_ext().moe_route(
            logits16.contiguous(), self.top_k)

        gg = self._grouped
        h = hip_ext().grouped_dq_gateup(
            x2, src_tok, offsets, gg["w13"]["qweight"], gg["w13"]["scales"],
            gg["w13"]["qzeros"], gg["w13"]["bits"], gg["w13"]["g"], active)
        y = torch.zeros(t, x2.shape[1], dtype=torch.float32,
                        device=x2.device)
        hip_ext().grouped_dq_linear_scatter(
            h, offsets, src_tok, slot_w, y, gg["w2"]["qweight"],
            gg["w2"]["scales"], gg["w2"]["qzeros"], gg["w2"]["bits"],
            gg["w2"]["g"], active)
        return y.to(x2.dtype)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: MixtralConfig, dtype):
        super().__init__()
        lcfg = LlamaConfig(hidden_size=cfg.hidden_size, n_heads=cfg.n_heads,
                           n_kv_heads=cfg.n_kv_heads, head_dim=cfg.head_dim,
                           rope_theta=cfg.rope_theta, rms_eps=cfg.rms_eps)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.self_attn = LlamaAttention(lcfg, dtype)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.moe = MixtralMoE(cfg, dtype)

    def forward(self, x, cache_k, cache_v, positions, seq_len_after,
                residual=None):
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
        h = self.moe(h)
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
        h = self.moe(h)
        return h, residual


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: MixtralConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(MixtralDecoderLayer(cfg, dtype)
                                    for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False,
                                 dtype=dtype)

    @torch.no_grad()
    def init_random(self, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        for p in self.parameters():
            p.data.copy_(torch.randn(p.shape, generator=g, dtype=torch.float32)
                         .mul_(0.02).to(p.dtype))
        return self

    @torch.no_grad()
    def consolidate_moe(self):
        for layer in self.layers:
            layer.moe.consolidate()
        return self

    def make_cache(self, batch: int, max_seq: Optional[int] = None, device=None):
        dev = device or next(self.parameters()).device
        hkv = self.layers[0].self_attn.n_kv_heads
        return KVCache(self.cfg.n_layers, batch, hkv,
                       max_seq or self.cfg.max_seq, self.cfg.head_dim, dev,
                       dtype=self.dtype)

    def forward(self, input_ids, cache: KVCache, positions=None):
        b, s = input_ids.shape
        pos0 = cache.seq_len
        if positions is None:
            positions = torch.arange(pos0, pos0 + s, device=input_ids.device)
        x = self.embed_tokens(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, cache.k[i], cache.v[i], positions,
                                pos0 + s, residual)
        x = self.norm(x, residual=residual)
        logits = self.lm_head(x)
        cache.advance(s)
        return logits

    def forward_decode_step(self, input_ids, cache, state, max_seq):
        x = self.embed_tokens(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer.forward_decode_step(
                x, cache.k[i], cache.v[i], state, max_seq, residual)
        x = self.norm(x, residual=residual)
        return self.lm_head(x)

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 16, cache=None,
                 gen_cfg=None, generator=None):
        from .generation import autoregressive_generate
        return autoregressive_generate(self, input_ids, max_new_tokens,
                                       cache, gen_cfg, generator)
