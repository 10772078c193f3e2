"""Packed checkpoint: safetensors tensors + quant_config.json metadata.

Format (FORMAT_VERSION 1, layout doc in lfmq/ops/ref.py):
  <dir>/model.safetensors
      per quantized Linear  <path>.qweight  int32  [n, k*bits/32]
                            <path>.scales   bf16   [n, G]
                            <path>.qzeros   int32  [n, ceil(G*bits/32)]
                            <path>.bias     bf16   [n]        (if present)
      per fp module         <path>.weight   bf16   (norms, embeddings, ...)
      per-tensor-int8 mode  <path>.qweight  int8   [n, k]
                            <path>.scale    fp32   []
  <dir>/quant_config.json   QuantConfig dict + model config echo + the
                            per-tensor shape manifest (n, k per layer) used
                            by the TP-shard-aware loader.

Writes are atomic (tmp file + os.replace).  load_packed(tp_rank, tp_world)
reads ONLY each rank's shard slices via safetensors lazy slicing — no
full-tensor read, no weight broadcast.
"""
from __future__ import annotations

import json
import os
import tempfile
from typing import Callable, Dict, Optional

import torch
from safetensors import safe_open
from safetensors.torch import save_file

from .config import FORMAT_VERSION, QuantConfig

MODEL_FILE = "model.safetensors"
CONFIG_FILE = "quant_config.json"


def save_packed(state: Dict[str, torch.Tensor], quant_cfg: QuantConfig,
                model_cfg: dict, path: str,
                manifest: Optional[dict] = None) -> None:
    os.makedirs(path, exist_ok=True)
    meta = {
        "format_version": FORMAT_VERSION,
        "quant": quant_cfg.to_dict(),
        "model": model_cfg,
        "manifest": manifest or {},
    }
    state = {k: v.contiguous().cpu() for k, v in state.items()}

    fd, tmp = tempfile.mkstemp(dir=path, suffix=".tmp")
    os.close(fd)
    try:
        save_file(state, tmp, metadata={"lfmq": json.dumps(meta["quant"])})
        os.replace(tmp, os.path.join(path, MODEL_FILE))
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)

    fd, tmp = tempfile.mkstemp(dir=path, suffix=".tmp")
    with os.fdopen(fd, "w") as f:
        json.dump(meta, f, indent=2, sort_keys=True)
    os.replace(tmp, os.path.join(path, CONFIG_FILE))


def load_meta(path: str) -> dict:
    with open(os.path.join(path, CONFIG_FILE)) as f:
        meta = json.load(f)
    ver = meta.get("format_version")
    if ver != FORMAT_VERSION:
        raise ValueError(f"checkpoint format_version {ver} != supported {FORMAT_VERSION}")
    return meta


def load_packed(
    path: str,
    shard_fn: Optional[Callable[[str, torch.Size], Optional[tuple]]] = None,
    device: str = "cpu",
) -> Dict[str, torch.Tensor]:
    """Read tensors, optionally slicing each one for this TP rank.

    shard_fn(name, shape) -> None (take whole tensor) or a tuple of slices
    to apply (safetensors lazy slicing: only the selected bytes are read).
    """
    out: Dict[str, torch.Tensor] = {}
    file = os.path.join(path, MODEL_FILE)
    with safe_open(file, framework="pt", device="cpu") as f:
        for name in f.keys():
            sl = f.get_slice(name)
            shape = torch.Size(sl.get_shape())
            spec = shard_fn(name, shape) if shard_fn is not None else None
            t = sl[spec] if spec is not None else f.get_tensor(name)
            out[name] = t.to(device) if device != "cpu" else t
    return out
