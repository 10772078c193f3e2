"""AWQ: activation-aware weight quantization.

Per scale-group (linears sharing one input), grid-search α ∈ [0,1] for
per-channel scales s = a_absmean^α (normalized), fold s into the producing
op (RMSNorm/LayerNorm weight, or the previous Linear's output rows), then
group-wise RTN int4/int8 on the scaled weights (BASELINE.json:5 inventory:
"group-wise scale/zero-point search").

Scale groups (fused-projection Llama arch):
  input_layernorm        -> qkv_proj
  post_attention_layernorm -> gate_up_proj
  gate_up_proj[up half]  -> down_proj
o_proj is quantized without a folded scale: with GQA (hq != hkv) the
v-channel -> o-input mapping is many-to-one, so a per-o-input scale is not
expressible through v rows (the standard llm-awq simplification).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn as nn

from ..config import QuantConfig
from ..models.qlinear import QLinear
from ..ops import ref
from .calib import CalibStream, InputCatcher
from .common import decoder_blocks, get_module, iter_target_linears, make_block_runner, replace_module
from .gptq import _capture_block0_inputs


def _pseudo_quant(w: torch.Tensor, bits: int, group_size: int, sym: bool) -> torch.Tensor:
    codes, s, z = ref.quant_groupwise(w, bits, group_size, sym)
    return ref.dequant_groupwise(codes, s, z,
                                 w.shape[1] if group_size == -1 else group_size)


@torch.no_grad()
def search_scale(
    x: torch.Tensor, weights: List[torch.Tensor], cfg: QuantConfig,
    n_grid: Optional[int] = None,
) -> torch.Tensor:
    """x [t, k] sample; weights: fp32 [n_i, k] sharing x.  Returns s [k]."""
    n_grid = n_grid or cfg.awq_grid
    xa = x.float().abs().mean(dim=0)           # per-channel activation scale
    xa = torch.clamp(xa, min=1e-8)
    best_err, best_s = None, torch.ones_like(xa)
    for gi in range(n_grid):
        alpha = gi / max(n_grid - 1, 1)
        s = xa.pow(alpha)
        s = s / (s.max() * s.min()).sqrt()     # normalize the dynamic range
        s = torch.clamp(s, 1e-4, 1e4)
        err = 0.0
        for w in weights:
            wq = _pseudo_quant(w * s, cfg.bits, cfg.group_size, cfg.sym) / s
            # ‖(W − Q(Ws)/s) x‖² with per-channel E[x²] weighting
            d = (w - wq)
            err += float(((d * d) * (x.float().pow(2).mean(0))).sum())
        if best_err is None or err < best_err:
            best_err, best_s = err, s
    return best_s


@torch.no_grad()
def search_clip(x: torch.Tensor, w: torch.Tensor, cfg: QuantConfig,
                grid=(1.0, 0.9, 0.8, 0.7, 0.6)) -> torch.Tensor:
    """Per-output-channel weight-clip search (AWQ's optional second
    stage): shrink each row's quantization range by the ratio that
    minimizes the activation-weighted reconstruction error.  Returns the
    clip ratio per row [n]."""
    xw = x.float().pow(2).mean(0)                    # per-in-channel energy
    n = w.shape[0]
    best_err = torch.full((n,), float("inf"), device=w.device)
    best_r = torch.ones(n, device=w.device)
    for r in grid:
        wc = torch.clamp(w, min=w.amin(dim=1, keepdim=True) * r,
                         max=w.amax(dim=1, keepdim=True) * r)
        wq = _pseudo_quant(wc, cfg.bits, cfg.group_size, cfg.sym)
        err = ((w - wq).pow(2) * xw).sum(dim=1)
        better = err < best_err
        best_err = torch.where(better, err, best_err)
        best_r = torch.where(better, torch.full_like(best_r, r), best_r)
    return best_r


@torch.no_grad()
def apply_clip(lin: nn.Linear, ratios: torch.Tensor) -> None:
    w = lin.weight.data.float()
    lo = w.amin(dim=1, keepdim=True) * ratios.unsqueeze(1)
    hi = w.amax(dim=1, keepdim=True) * ratios.unsqueeze(1)
    lin.weight.data = torch.clamp(w, min=lo, max=hi).to(lin.weight.dtype)


def _scale_plan(model) -> List[dict]:
    """Per-block scale groups: {prev: (kind, name, slice), linears: [...]}"""
    arch = getattr(model.cfg, "arch", "llama")
    if arch == "llama":
        inter = model.cfg.intermediate_size
        return [
            {"prev": ("norm", "input_layernorm", None),
             "linears": ["self_attn.qkv_proj"]},
            {"prev": ("norm", "post_attention_layernorm", None),
             "linears": ["mlp.gate_up_proj"]},
            {"prev": ("linear_rows", "mlp.gate_up_proj", (inter, 2 * inter)),
             "linears": ["mlp.down_proj"]},
        ]
    if arch == "opt":
        h = model.cfg.hidden_size
        return [
            {"prev": ("norm", "self_attn_layer_norm", None),
             "linears": ["self_attn.q_proj", "self_attn.k_proj", "self_attn.v_proj"]},
            {"prev": ("linear_rows", "self_attn.v_proj", (0, h)),
             "linears": ["self_attn.out_proj"]},
            {"prev": ("norm", "final_layer_norm", None),
             "linears": ["fc1"]},
            {"prev": ("linear_rows", "fc1", (0, None)),
             "linears": ["fc2"]},
        ]
    if arch == "mixtral":
        return [{"prev": ("norm", "input_layernorm", None),
                 "linears": ["self_attn.qkv_proj"]},
                {"prev": ("norm", "post_attention_layernorm", None),
                 "linears": [f"moe.experts.{e}.w13"
                             for e in range(model.cfg.n_experts)]}]
    raise ValueError(f"unknown arch {arch!r}")


@torch.no_grad()
def apply_scale(block: nn.Module, prev: tuple, linears: List[nn.Linear],
                s: torch.Tensor) -> None:
    """Fold s: linears' weight cols *= s; producing op's output /= s."""
    for lin in linears:
        lin.weight.data = (lin.weight.data.float() * s).to(lin.weight.dtype)
    kind, name, sl = prev
    mod = get_module(block, name)
    if kind == "norm":
        mod.weight.data = (mod.weight.data.float() / s).to(mod.weight.dtype)
        if getattr(mod, "bias", None) is not None:
            mod.bias.data = (mod.bias.data.float() / s).to(mod.bias.dtype)
    elif kind == "linear_rows":
        lo, hi = sl
        hi = hi if hi is not None else mod.weight.shape[0]
        mod.weight.data[lo:hi] = (mod.weight.data[lo:hi].float()
                                  / s.unsqueeze(1)).to(mod.weight.dtype)
        if getattr(mod, "bias", None) is not None:
            mod.bias.data[lo:hi] = (mod.bias.data[lo:hi].float() / s).to(mod.bias.dtype)
    else:
        raise ValueError(kind)


class AWQQuantizer:
    def __init__(self, cfg: QuantConfig):
        self.cfg = cfg

    @torch.no_grad()
    def run(self, model: nn.Module, calib: Optional[CalibStream] = None,
            device: Optional[torch.device] = None,
            max_sample_tokens: int = 2048) -> nn.Module:
        cfg = self.cfg
        device = device or next(model.parameters()).device
        if calib is None:
            calib = CalibStream(model.cfg.vocab_size, cfg.calib_seq_len,
                                cfg.calib_n_seqs, cfg.calib_seed)
        run_block = make_block_runner(model)
        xs = _capture_block0_inputs(model, calib, device)
        target_names = {n for n, _ in iter_target_linears(model, cfg)}
        plan = _scale_plan(model)

        blocks = list(decoder_blocks(model))
        from ..obs import get_logger
        log = get_logger()
        import time as _time
        t_run = _time.perf_counter()
        for bi, (bname, block) in enumerate(blocks):
            log.info("awq block %d/%d (%s) [%.1fs elapsed]", bi + 1,
                     len(blocks), bname, _time.perf_counter() - t_run)
            block.to(device)
            linears = {n: m for n, m in block.named_modules()
                       if isinstance(m, nn.Linear)
                       and f"{bname}.{n}" in target_names}
            # capture a bounded input sample per linear
            samples: Dict[str, List[torch.Tensor]] = {n: [] for n in linears}
            counts = {n: 0 for n in linears}

            def consume(name, x):
                if counts[name] < max_sample_tokens:
                    take = min(x.shape[0], max_sample_tokens - counts[name])
                    samples[name].append(x[:take].float().cpu())
                    counts[name] += take

            with InputCatcher(linears, consume=consume):
                for x in xs:
                    run_block(block, x.to(device))

            for group in plan:
                names = [n for n in group["linears"] if n in linears]
                if not names:
                    continue
                xcat = torch.cat(samples[names[0]], dim=0).to(device)
                ws = [linears[n].weight.float() for n in names]
                s = search_scale(xcat, ws, cfg)
                apply_scale(block, group["prev"], [linears[n] for n in names], s)

            for lname, lin in linears.items():
                if cfg.awq_clip and samples.get(lname):
                    xcat = torch.cat(samples[lname], dim=0).to(device)
                    ratios = search_clip(xcat, lin.weight.float(), cfg)
                    apply_clip(lin, ratios.to(lin.weight.device))
                q = QLinear.from_rtn(lin, cfg).to(device)
                replace_module(block, lname, q)

            xs = [run_block(block, x.to(device)).to(xs[0].device) for x in xs]
        return model
