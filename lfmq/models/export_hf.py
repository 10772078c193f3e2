"""Export an lfmq model to an HF-transformers-layout bf16 checkpoint —
the inverse of lfmq/models/import_hf.py, closing the interchange loop:
quantize here, hand the dequantized result back to any HF-layout tool.

Quantized modules are DEQUANTIZED (QLinear.dequant returns original
input order, so act-order perm disappears); norm-folded checkpoints
export as-is — all-ones norm weights plus folded projections are a
function-identical HF llama.  Name mapping mirrors import_hf:
  - everything except lm_head gains the "model." prefix
  - qkv_proj splits into self_attn.{q,k,v}_proj ([q|k|v] rows)
  - gate_up_proj splits into mlp.{gate,up}_proj ([gate|up] rows)
  - moe.gate -> block_sparse_moe.gate;
    moe.experts.E.w13 -> block_sparse_moe.experts.E.{w1,w3}; w2 -> w2
Tied-embedding models write one copy (no lm_head.weight), as HF does.

    qm = lfmq.load("ckpt/")
    from lfmq.models.export_hf import export_hf
    export_hf(qm, "hf_out/")        # -> hf_out/model.safetensors
"""
from __future__ import annotations

import os
from typing import Dict

import torch
import torch.nn as nn

from .qlinear import Int8PerTensorLinear, QLinear


def _weight_of(mod: nn.Module) -> torch.Tensor:
    if isinstance(mod, QLinear):
        return mod.dequant().to(torch.bfloat16)
    if isinstance(mod, Int8PerTensorLinear):
        return (mod.qweight.float() * mod.scale).to(torch.bfloat16)
    return mod.weight.detach().to(torch.bfloat16)


@torch.no_grad()
def export_hf_state(model: nn.Module) -> Dict[str, torch.Tensor]:
    arch = getattr(model.cfg, "arch", None)
    if arch not in ("llama", "mixtral"):
        raise NotImplementedError(f"HF export supports llama/mixtral, not {arch!r}")
    cfg = model.cfg
    d, hq, hkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
    inter = cfg.intermediate_size
    out: Dict[str, torch.Tensor] = {}

    def put(name: str, t: torch.Tensor) -> None:
        pre = "" if name.startswith("lm_head") else "model."
        out[pre + name] = t.contiguous()

    tied = (isinstance(getattr(model, "lm_head", None), nn.Linear)
            and model.lm_head.weight is model.embed_tokens.weight)
    for name, mod in model.named_modules():
        is_lin = isinstance(mod, (nn.Linear, QLinear, Int8PerTensorLinear))
        if not is_lin:
            if isinstance(mod, (nn.Embedding,)):
                put(name + ".weight", mod.weight.detach().to(torch.bfloat16))
            elif hasattr(mod, "weight") and isinstance(
                    getattr(mod, "weight"), torch.Tensor) and not list(
                    mod.children()):
                # leaf norm modules
                put(name + ".weight", mod.weight.detach().to(torch.bfloat16))
            continue
        if name == "lm_head" and tied:
            continue
        w = _weight_of(mod)
        bias = getattr(mod, "bias", None)
        if name.endswith("qkv_proj"):
            base = name[: -len("qkv_proj")]
            q, k, v = torch.split(w, [hq * d, hkv * d, hkv * d], dim=0)
            put(base + "q_proj.weight", q)
            put(base + "k_proj.weight", k)
            put(base + "v_proj.weight", v)
        elif name.endswith("gate_up_proj"):
            base = name[: -len("gate_up_proj")]
            g, u = torch.split(w, [inter, inter], dim=0)
            put(base + "gate_proj.weight", g)
            put(base + "up_proj.weight", u)
        elif ".experts." in name and name.endswith(".w13"):
            base = name[: -len("w13")].replace("moe.experts.",
                                               "block_sparse_moe.experts.")
            w1, w3 = torch.split(w, [inter, inter], dim=0)
            put(base + "w1.weight", w1)
            put(base + "w3.weight", w3)
        else:
            hf_name = name.replace("moe.gate", "block_sparse_moe.gate")
            hf_name = hf_name.replace("moe.experts.",
                                      "block_sparse_moe.experts.")
            put(hf_name + ".weight", w)
            if bias is not None:
                put(hf_name + ".bias", bias.detach().to(torch.bfloat16))
    return out


@torch.no_grad()
def export_hf(model: nn.Module, path: str) -> str:
    """Write model.safetensors in HF-transformers layout; returns the file."""
    from safetensors.torch import save_file
    os.makedirs(path, exist_ok=True)
    state = {k: v.cpu() for k, v in export_hf_state(model).items()}
    fn = os.path.join(path, "model.safetensors")
    save_file(state, fn)
    return fn
