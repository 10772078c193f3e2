"""Shared quantizer helpers: module walking/replacement, block discovery."""
from __future__ import annotations

from typing import List, Tuple

import torch.nn as nn

from ..config import QuantConfig


def _skip(name: str, cfg: QuantConfig) -> bool:
    return any(name == s or name.endswith("." + s) or name.split(".")[-1] == s
               for s in cfg.skip_modules)


def iter_target_linears(model: nn.Module, cfg: QuantConfig,
                        prefix: str = "") -> List[Tuple[str, nn.Linear]]:
    out = []
    for name, mod in model.named_modules():
        if isinstance(mod, nn.Linear) and not _skip(name, cfg):
            out.append((name, mod))
    return out


def replace_module(model: nn.Module, name: str, new: nn.Module) -> None:
    parts = name.split(".")
    parent = model
    for p in parts[:-1]:
        parent = getattr(parent, p)
    setattr(parent, parts[-1], new)


def get_module(model: nn.Module, name: str) -> nn.Module:
    mod = model
    for p in name.split("."):
        mod = getattr(mod, p)
    return mod


def make_block_runner(model: nn.Module):
    """Arch adapter: fn(block, x [b,s,h]) -> y [b,s,h], prefill-shaped
    (fresh scratch KV for the batch; no persistent cache)."""
    import torch
    arch = getattr(model.cfg, "arch", "llama")
    if arch in ("llama", "mixtral"):
        hkv = model.layers[0].self_attn.n_kv_heads
        d = model.cfg.head_dim

        def run(block, x):
            b, s, _ = x.shape
            ck = torch.zeros(b, hkv, s, d, device=x.device, dtype=x.dtype)
            cv = torch.zeros_like(ck)
            pos = torch.arange(s, device=x.device)
            h, res = block(x, ck, cv, pos, s)  # fused-residual form
            return (h.float() + res.float()).to(h.dtype)
        return run
    if arch == "opt":
        nh = model.cfg.n_heads
        d = model.cfg.hidden_size // nh

        def run(block, x):
            b, s, _ = x.shape
            ck = torch.zeros(b, nh, s, d, device=x.device, dtype=x.dtype)
            cv = torch.zeros_like(ck)
            return block(x, ck, cv, s)
        return run
    raise ValueError(f"unknown arch {arch!r}")


def decoder_blocks(model: nn.Module) -> List[Tuple[str, nn.Module]]:
    """The sequential per-block calibration unit list (model.layers[i])."""
    if hasattr(model, "layers"):
        return [(f"layers.{i}", blk) for i, blk in enumerate(model.layers)]
    raise ValueError("model has no .layers — cannot do per-block calibration")
