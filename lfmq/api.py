"""Public API: quantize() / save() / load() (BASELINE.json north star).

    from lfmq import quantize, save, load, QuantConfig
    qm = quantize(model, QuantConfig(scheme="gptq"))     # in-place swap to QLinear
    save(qm, "ckpt/")                                     # packed safetensors
    qm = load("ckpt/", device="cuda", tp_rank=r, tp_world=w)
"""
from __future__ import annotations

from typing import Optional, Union

import torch
import torch.nn as nn

from .checkpoint import load_meta, load_packed, save_packed
from .config import QuantConfig
from .models.llama import LlamaConfig, LlamaForCausalLM
from .models.opt import OPTConfig, OPTForCausalLM
from .models.qlinear import Int8PerTensorLinear, QLinear
from .quant.common import replace_module


def build_model(model_cfg: Union[dict, LlamaConfig, OPTConfig], dtype=None,
                seed: Optional[int] = None) -> nn.Module:
    """Build a model (random-init if seed given) from a local config."""
    if isinstance(model_cfg, dict):
        arch = model_cfg.get("arch", "llama")
        if arch == "llama":
            model_cfg = LlamaConfig.from_dict(model_cfg)
        elif arch == "opt":
            model_cfg = OPTConfig.from_dict(model_cfg)
        elif arch == "mixtral":
            from .models.mixtral import MixtralConfig
            model_cfg = MixtralConfig.from_dict(model_cfg)
        else:
            raise ValueError(f"unknown arch {arch!r}")
    arch = model_cfg.arch
    if arch == "llama":
        m = LlamaForCausalLM(model_cfg, dtype=dtype or torch.bfloat16)
    elif arch == "opt":
        m = OPTForCausalLM(model_cfg, dtype=dtype or torch.float32)
    elif arch == "mixtral":
        from .models.mixtral import MixtralForCausalLM
        m = MixtralForCausalLM(model_cfg, dtype=dtype or torch.bfloat16)
    else:
        raise ValueError(f"unknown arch {arch!r}")
    if seed is not None:
        m.init_random(seed)
    return m


def quantize(model: nn.Module, cfg: QuantConfig, calib=None,
             device: Optional[torch.device] = None) -> nn.Module:
    """Quantize every target nn.Linear in-place; returns the model."""
    if cfg.fold_rms and getattr(model.cfg, "arch", None) == "llama":
        from .quant.fold import fold_rms_norms
        fold_rms_norms(model)
    if cfg.scheme == "rtn":
        from .quant.rtn import RTNQuantizer
        return RTNQuantizer(cfg).run(model)
    if cfg.scheme == "gptq":
        from .quant.gptq import GPTQQuantizer
        return GPTQQuantizer(cfg).run(model, calib, device)
    if cfg.scheme == "awq":
        from .quant.awq import AWQQuantizer
        return AWQQuantizer(cfg).run(model, calib, device)
    raise ValueError(cfg.scheme)


def _manifest(model: nn.Module) -> dict:
    man = {}
    for name, mod in model.named_modules():
        if isinstance(mod, QLinear):
            man[name] = {"kind": "qlinear", "in": mod.in_features,
                         "out": mod.out_features, "bits": mod.bits,
                         "g": mod.group_size, "bias": mod.bias is not None,
                         "perm": hasattr(mod, "perm")}
        elif isinstance(mod, Int8PerTensorLinear):
            man[name] = {"kind": "int8pt", "in": mod.in_features,
                         "out": mod.out_features, "bias": mod.bias is not None}
    return man


def save(model: nn.Module, path: str, quant_cfg: QuantConfig) -> None:
    state = {k: v for k, v in model.state_dict().items()}
    # tied embeddings alias lm_head.weight to embed_tokens.weight;
    # safetensors refuses shared storage, so persist ONE copy and let
    # load() re-alias (tie_embeddings is in the model-config echo).
    lm = getattr(model, "lm_head", None)
    emb = getattr(model, "embed_tokens", None)
    if (isinstance(lm, nn.Linear) and emb is not None
            and "lm_head.weight" in state and lm.weight is emb.weight):
        del state["lm_head.weight"]
    save_packed(state, quant_cfg, model.cfg.to_dict(), path,
                manifest=_manifest(model))


def _build_quantized_skeleton(meta: dict, dtype=None) -> nn.Module:
    qcfg = QuantConfig.from_dict(meta["quant"])
    model = build_model(meta["model"], dtype=dtype)
    for name, spec in meta["manifest"].items():
        if spec["kind"] == "qlinear":
            q = QLinear(spec["in"], spec["out"], spec["bits"], spec["g"],
                        bias=spec["bias"])
            if spec.get("perm") or spec.get("g_idx"):
                q.register_buffer("perm", torch.zeros(spec["in"],
                                                      dtype=torch.int64))
        else:
            q = Int8PerTensorLinear(spec["in"], spec["out"], bias=spec["bias"])
        replace_module(model, name, q)
    return model


def load(path: str, device: str = "cpu", tp_rank: int = 0, tp_world: int = 1,
         dtype=None) -> nn.Module:
    """Load a packed checkpoint.  tp_world>1: each rank reads only its own
    shard slices (no broadcast) and gets a TP-sharded module tree."""
    meta = load_meta(path)
    if tp_world > 1:
        from .dist.tp import load_tp
        return load_tp(path, meta, device, tp_rank, tp_world, dtype=dtype)
    model = _build_quantized_skeleton(meta, dtype=dtype)
    # tensors go straight to the target device and are ASSIGNED into the
    # module tree (no CPU materialization of a second copy): a 70B W4
    # checkpoint (~35 GB packed) loads with one host pass, not two.
    state = load_packed(path, device=str(device))
    if ("lm_head.weight" not in state and "embed_tokens.weight" in state
            and isinstance(getattr(model, "lm_head", None), nn.Linear)):
        # tied checkpoint stores one copy (save() dropped the alias)
        state["lm_head.weight"] = state["embed_tokens.weight"]
    if dtype is not None:
        # assign=True would otherwise let the checkpoint dtype override an
        # explicitly requested one: cast float tensors here.
        state = {k: (v.to(dtype) if v.is_floating_point() else v)
                 for k, v in state.items()}
    model.load_state_dict(state, strict=True, assign=True)
    model = model.to(device)  # anything not in the checkpoint (none today)
    # assign=True unties tied embeddings (two independent tensors after the
    # load): re-tie so memory and semantics match build time.
    if getattr(model.cfg, "tie_embeddings", False) and \
            isinstance(getattr(model, "lm_head", None), nn.Linear):
        model.lm_head.weight = model.embed_tokens.weight
    if getattr(model.cfg, "arch", None) == "llama":
        from .quant.fold import refresh_norm_fold_flags
        refresh_norm_fold_flags(model)
    return model
