"""Import pretrained weights from a local HF-transformers-layout
checkpoint (safetensors shards) into an lfmq model, for quantization.

Covers the naming deltas between HF-transformers llama/mixtral state
dicts and ours:
  - the "model." prefix is dropped (lm_head is unprefixed on both sides)
  - self_attn.{q,k,v}_proj fuse into self_attn.qkv_proj ([q|k|v] rows,
    the fused layout lfmq/dist/tp.py shards by)
  - mlp.{gate,up}_proj fuse into mlp.gate_up_proj ([gate|up] rows)
  - mixtral block_sparse_moe.gate -> moe.gate and
    block_sparse_moe.experts.E.{w1,w3} -> moe.experts.E.w13 ([w1|w3]),
    w2 -> w2
RoPE layout: HF-transformers llama checkpoints store q/k for the
rotate-half convention (pair i, i + d/2) — the same convention our
kernels implement, so no permutation is applied.  Tied-embedding
checkpoints (no lm_head.weight) fall back to embed_tokens.

    model = lfmq.build_model(LlamaConfig.llama3_8b())
    import_hf_state(model, "/path/to/hf_checkpoint_dir")
    qm = lfmq.quantize(model, qcfg)

CLI: `python -m lfmq.cli quantize --hf-dir /path/to/ckpt ...`.
"""
from __future__ import annotations

import glob
import os
import re
from typing import Dict, Union

import torch
import torch.nn as nn


def read_hf_dir(path: str) -> Dict[str, torch.Tensor]:
    """Load every *.safetensors shard in a directory (CPU tensors)."""
    from safetensors import safe_open
    files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no .safetensors shards under {path!r}")
    out: Dict[str, torch.Tensor] = {}
    for fn in files:
        with safe_open(fn, framework="pt", device="cpu") as f:
            for name in f.keys():
                out[name] = f.get_tensor(name)
    return out


def _strip(name: str) -> str:
    return name[len("model."):] if name.startswith("model.") else name


def convert_hf_state(hf: Dict[str, torch.Tensor], arch: str = "llama"
                     ) -> Dict[str, torch.Tensor]:
    """HF-transformers llama/mixtral state dict -> lfmq state dict."""
    plain: Dict[str, torch.Tensor] = {}
    qkv: Dict[str, Dict[str, torch.Tensor]] = {}
    gu: Dict[str, Dict[str, torch.Tensor]] = {}
    w13: Dict[str, Dict[str, torch.Tensor]] = {}

    for name, t in hf.items():
        n = _strip(name)
        m = re.match(r"(.*self_attn)\.([qkv])_proj\.weight$", n)
        if m:
            qkv.setdefault(m.group(1), {})[m.group(2)] = t
            continue
        m = re.match(r"(.*mlp)\.(gate|up)_proj\.weight$", n)
        if m:
            gu.setdefault(m.group(1), {})[m.group(2)] = t
            continue
        m = re.match(r"(.*)block_sparse_moe\.experts\.(\d+)\.(w1|w3)\.weight$", n)
        if m:
            w13.setdefault(f"{m.group(1)}moe.experts.{m.group(2)}", {})[
                m.group(3)] = t
            continue
        n = n.replace("block_sparse_moe.", "moe.")
        plain[n] = t

    out = dict(plain)
    for base, parts in qkv.items():
        missing = {"q", "k", "v"} - set(parts)
        if missing:
            raise KeyError(f"{base}: missing {sorted(missing)}_proj")
        out[f"{base}.qkv_proj.weight"] = torch.cat(
            [parts["q"], parts["k"], parts["v"]], dim=0)
    for base, parts in gu.items():
        missing = {"gate", "up"} - set(parts)
        if missing:
            raise KeyError(f"{base}: missing {sorted(missing)}_proj")
        out[f"{base}.gate_up_proj.weight"] = torch.cat(
            [parts["gate"], parts["up"]], dim=0)
    for base, parts in w13.items():
        missing = {"w1", "w3"} - set(parts)
        if missing:
            raise KeyError(f"{base}: missing {sorted(missing)}")
        out[f"{base}.w13.weight"] = torch.cat([parts["w1"], parts["w3"]],
                                              dim=0)
    if "lm_head.weight" not in out and "embed_tokens.weight" in out:
        out["lm_head.weight"] = out["embed_tokens.weight"]  # tied embeddings
    return out


@torch.no_grad()
def import_hf_state(model: nn.Module,
                    src: Union[str, Dict[str, torch.Tensor]],
                    strict: bool = True) -> nn.Module:
    """Load an HF-layout checkpoint (directory path or raw state dict)
    into a freshly built fp lfmq model (BEFORE quantization)."""
    hf = read_hf_dir(src) if isinstance(src, str) else src
    arch = getattr(model.cfg, "arch", "llama")
    state = convert_hf_state(hf, arch)
    dt = next(model.parameters()).dtype
    state = {k: v.to(dt) for k, v in state.items()}
    model.load_state_dict(state, strict=strict)
    return model
