"""GPTQ calibration: per-channel Hessian accumulate, damped Cholesky
inverse, column-blocked quantize-then-error-propagate loop.

Math (the north star's inventory, BASELINE.json:5):
  H   = Σ 2·XᵀX                      (ops.hessian_accum — HIP K1 on GPU)
  H_d = H + λ·mean(diag H)·I          (λ = damp_percent)
  U   = chol(H_d⁻¹, upper)            (torch.linalg → rocSOLVER on GPU)
  for each column j (in blocks):      q_j = quant(w_j)
      err = (w_j − dq_j) / U_jj ;  w_{j+1..} −= err ⊗ U_{j, j+1..}
Scales/zeros are recomputed at every group boundary from the
error-compensated weights.  act_order=True quantizes columns in
descending diag(H) order; the packed result keeps that (group-contiguous)
order and records `perm` — the forward gathers x[perm], so every fused
GPU kernel serves act-order checkpoints unchanged.

Calibration is sequential per decoder block: only one block's Hessians
(≈1 GB fp32 for 8B, ≈4 GB for 70B) are live at a time, and the next
block's inputs are produced by the already-quantized block, so error
propagates realistically.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from ..config import QuantConfig
from ..models.qlinear import QLinear
from ..ops import ref
from .calib import CalibStream, InputCatcher
from .common import decoder_blocks, iter_target_linears, replace_module


def _hessian_accum(h: torch.Tensor, x: torch.Tensor) -> None:
    if x.is_cuda:
        from ..ops import hip_ext
        hip_ext().hessian_accum(h, x)
    else:
        ref.hessian_accum(h, x)


@torch.no_grad()
def gptq_quantize_layer(
    w: torch.Tensor, h: torch.Tensor, cfg: QuantConfig
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """w fp32 [n, k], h fp32 [k, k] -> (codes int32 [n,k], scales fp32 [n,G],
    zeros int32 [n,G], perm int64 [k] or None).

    Without act_order, codes are in original column order and perm is
    None.  With act_order, codes stay in QUANT (permuted) order — groups
    contiguous, scales indexed by position//g — and perm maps quant
    column j to original input perm[j] (the forward gathers x[perm]).
    """
    n, k = w.shape
    g = k if cfg.group_size == -1 else cfg.group_size
    G = k // g
    w = w.float().clone()
    h = h.float().clone()
    dev = w.device

    dead = torch.diag(h) == 0
    if dead.any():
        h[dead, dead] = 1.0
        w[:, dead] = 0.0

    perm = None
    if cfg.act_order:
        perm = torch.argsort(torch.diag(h), descending=True)
        w = w[:, perm]
        h = h[perm][:, perm]

    damp = cfg.damp_percent * torch.diag(h).mean()
    h += torch.eye(k, device=dev) * damp

    # U = chol(H^-1, upper) via the standard two-step (robust on rocSOLVER):
    L = torch.linalg.cholesky(h)
    hinv = torch.cholesky_inverse(L)
    U = torch.linalg.cholesky(hinv, upper=True)

    codes = torch.zeros(n, k, dtype=torch.int32, device=dev)
    scales = torch.zeros(n, G, dtype=torch.float32, device=dev)
    zeros = torch.zeros(n, G, dtype=torch.int32, device=dev)
    qmax = (1 << cfg.bits) - 1

    bs = cfg.block_size
    scale_cur = torch.ones(n, device=dev)
    zero_cur = torch.zeros(n, dtype=torch.int32, device=dev)
    for i1 in range(0, k, bs):
        i2 = min(i1 + bs, k)
        wb = w[:, i1:i2].clone()
        err = torch.zeros_like(wb)
        ub = U[i1:i2, i1:i2]
        for j in range(i2 - i1):
            col = i1 + j
            if col % g == 0:
                # (re)fit scale/zero on the error-compensated group slice
                grp = w[:, col:col + g]
                _, s_, z_ = ref.quant_groupwise(grp, cfg.bits, g, cfg.sym)
                gi = col // g
                scales[:, gi] = s_[:, 0]
                zeros[:, gi] = z_[:, 0]
                scale_cur = s_[:, 0].to(dev)
                zero_cur = z_[:, 0].to(dev)
            wj = wb[:, j]
            q = torch.clamp(torch.round(wj / scale_cur) + zero_cur, 0, qmax)
            codes[:, col] = q.to(torch.int32)
            dq = (q - zero_cur) * scale_cur
            e = (wj - dq) / ub[j, j]
            if j + 1 < i2 - i1:
                wb[:, j + 1:] -= torch.outer(e, ub[j, j + 1:])
            err[:, j] = e
        w[:, i1:i2] = wb
        if i2 < k:
            w[:, i2:] -= err @ U[i1:i2, i2:]

    aperm = None
    if perm is not None:
        # codes stay in QUANT (permuted) order: groups are CONTIGUOUS, so
        # every fused GPU kernel works unchanged; the forward gathers
        # x[perm] instead (column j of W corresponds to input perm[j]).
        aperm = perm.to(torch.int64)
    return codes, scales, zeros, aperm


class GPTQQuantizer:
    def __init__(self, cfg: QuantConfig):
        self.cfg = cfg

    @torch.no_grad()
    def run(self, model: nn.Module, calib: Optional[CalibStream] = None,
            device: Optional[torch.device] = None,
            resume_dir: Optional[str] = None) -> nn.Module:
        """resume_dir: per-block completion checkpoints —
        a 70B calibration is long; `quantize --resume` skips blocks whose
        quantized state was already written (atomic tmp+rename)."""
        import os
        import tempfile
        cfg = self.cfg
        device = device or next(model.parameters()).device
        if calib is None:
            vocab = model.cfg.vocab_size
            calib = CalibStream(vocab, cfg.calib_seq_len, cfg.calib_n_seqs,
                                cfg.calib_seed)

        from .common import make_block_runner
        run_block = make_block_runner(model)
        # Hessian sampling sanity (profiles/gptq_fidelity_r02.md): with
        # fewer calibration tokens than a layer's in_features the Hessian
        # is rank-deficient and error propagation against its noisy
        # inverse makes GPTQ WORSE than plain RTN (measured 2x logit KL).
        n_tok = cfg.calib_n_seqs * cfg.calib_seq_len
        max_k = max((m.in_features for _, m in
                     iter_target_linears(model, cfg)), default=0)
        if n_tok < max_k:
            from ..obs import get_logger
            get_logger().warning(
                "GPTQ calibration has %d tokens but the widest target "
                "layer has in_features=%d: the Hessian will be rank-"
                "deficient and fidelity can drop BELOW plain RTN "
                "(profiles/gptq_fidelity_r02.md). Use calib_n_seqs * "
                "calib_seq_len >= %d.", n_tok, max_k, max_k)
        xs = _capture_block0_inputs(model, calib, device)

        if resume_dir:
            os.makedirs(os.path.join(resume_dir, "blocks"), exist_ok=True)

        target_names = {n for n, _ in iter_target_linears(model, cfg)}
        blocks = list(decoder_blocks(model))
        from ..obs import get_logger
        log = get_logger()
        import time as _time
        t_run = _time.perf_counter()
        for bi, (bname, block) in enumerate(blocks):
            log.info("gptq block %d/%d (%s) [%.1fs elapsed]", bi + 1,
                     len(blocks), bname, _time.perf_counter() - t_run)
            block.to(device)
            if resume_dir:
                bpath = os.path.join(resume_dir, "blocks", f"{bi:04d}.pt")
                if os.path.exists(bpath):
                    saved = torch.load(bpath, weights_only=False)
                    for lname, meta in saved["manifest"].items():
                        lin = block.get_submodule(lname)
                        q = QLinear(meta["in"], meta["out"], meta["bits"],
                                    meta["g"], bias=meta["bias"]).to(device)
                        if meta.get("perm"):
                            # act_order checkpoints carry a perm buffer;
                            # register it so load_state_dict finds the key
                            # (resume of act_order=True failed).
                            q.register_buffer("perm", torch.zeros(
                                meta["in"], dtype=torch.int64, device=device))
                        replace_module(block, lname, q)
                    block.load_state_dict(saved["state"])
                    xs = [run_block(block, x.to(device)).to(xs[0].device)
                          for x in xs]
                    continue
            linears = {n: m for n, m in block.named_modules()
                       if isinstance(m, nn.Linear) and f"{bname}.{n}" in
                       {t for t in target_names if t.startswith(bname + ".")}}
            # accumulate H per linear over all calib batches
            hess: Dict[str, torch.Tensor] = {
                n: torch.zeros(m.in_features, m.in_features,
                               dtype=torch.float32, device=device)
                for n, m in linears.items()}

            def consume(name, x):
                _hessian_accum(hess[name], x.to(device))

            with InputCatcher(linears, consume=consume):
                for x in xs:
                    run_block(block, x.to(device))

            for lname, lin in linears.items():
                codes, scales, zeros, perm = gptq_quantize_layer(
                    lin.weight.float(), hess[lname], cfg)
                q = QLinear(lin.in_features, lin.out_features, cfg.bits,
                            cfg.group_size if cfg.group_size != -1 else lin.in_features,
                            bias=lin.bias is not None)
                q.load_quantized(codes.cpu(), scales.cpu(), zeros.cpu(),
                                 lin.bias.detach() if lin.bias is not None else None)
                if perm is not None:
                    q.register_buffer("perm", perm.cpu())
                q.to(device)
                replace_module(block, lname, q)
                del hess[lname]

            if resume_dir:
                man = {}
                for lname, mod in block.named_modules():
                    if isinstance(mod, QLinear):
                        man[lname] = {"in": mod.in_features,
                                      "out": mod.out_features,
                                      "bits": mod.bits, "g": mod.group_size,
                                      "bias": mod.bias is not None,
                                      "perm": hasattr(mod, "perm")}
                fd, tmp = tempfile.mkstemp(dir=os.path.join(resume_dir, "blocks"))
                os.close(fd)
                torch.save({"state": {k: v.cpu() for k, v in
                                      block.state_dict().items()},
                            "manifest": man}, tmp)
                os.replace(tmp, os.path.join(resume_dir, "blocks",
                                             f"{bi:04d}.pt"))

            # next block's inputs come from the QUANTIZED block
            xs = [run_block(block, x.to(device)).to(xs[0].device) for x in xs]
        return model


@torch.no_grad()
def _capture_block0_inputs(model, calib: CalibStream, device) -> List[torch.Tensor]:
    """Run the embedding front-end only, catching block-0 inputs."""
    blocks = decoder_blocks(model)
    first = blocks[0][1]

    class _Stop(Exception):
        pass

    caught: List[torch.Tensor] = []

    def hook(m, args, kwargs):
        caught.append(args[0].detach().cpu())
        raise _Stop

    h = first.register_forward_pre_hook(hook, with_kwargs=True)
    try:
        for ids in calib.batches():
            try:
                cache = model.make_cache(ids.shape[0], max_seq=ids.shape[1],
                                         device=device)
                model(ids.to(device), cache)
            except _Stop:
                pass
    finally:
        h.remove()
    return caught
