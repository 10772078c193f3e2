"""Fold RMSNorm weights into the following projection's weights.

rmsnorm(x, w) @ Wᵀ == rmsnorm(x, 1) @ (W·diag(w))ᵀ — applied BEFORE
quantization, this leaves every decoder-layer norm with an all-ones
weight, which lets the decode GEMV fuse the whole normalization as a
scalar rsqrt on its accumulator (csrc/dq_gemv.hip RMS path) and removes
two kernels per layer from the decode step.

The transform is exact in fp32 and changes only bf16 rounding; the
packed checkpoint stays self-consistent (it stores the folded weights
and the all-ones norms).  Models whose norms are not all-ones (e.g.
after AWQ's scale fold wrote 1/s there) simply keep the separate-kernel
path — refresh_norm_fold_flags() detects this at load time.
"""
from __future__ import annotations

import torch
import torch.nn as nn


@torch.no_grad()
def fold_rms_norms(model: nn.Module) -> nn.Module:
    """Fold input/post-attention norm weights into qkv/gate_up (Llama)."""
    if getattr(model.cfg, "arch", None) != "llama":
        return model
    for layer in model.layers:
        for norm_name, lin in (("input_layernorm", layer.self_attn.qkv_proj),
                               ("post_attention_layernorm",
                                layer.mlp.gate_up_proj)):
            norm = getattr(layer, norm_name)
            if not isinstance(lin, nn.Linear):
                continue  # already quantized; cannot fold
            w = norm.weight.float()
            lin.weight.data = (lin.weight.float() * w).to(lin.weight.dtype)
            norm.weight.data = torch.ones_like(norm.weight.data)
    refresh_norm_fold_flags(model)
    return model


@torch.no_grad()
def refresh_norm_fold_flags(model: nn.Module) -> None:
    """Set per-layer flags telling the decode path which norms are
    fused-safe (weight ≡ 1).  Called after quantize() and load()."""
    if getattr(model.cfg, "arch", None) != "llama":
        return
    for layer in model.layers:
        layer._in_norm_folded = bool(
            (layer.input_layernorm.weight.float() == 1.0).all().item())
        layer._post_norm_folded = bool(
            (layer.post_attention_layernorm.weight.float() == 1.0).all().item())
