"""Delta-perplexity harness: teacher-forced NLL on an
identical fixed-seed synthetic stream through the bf16 model and the
quantized model; delta_ppl = ppl_quant − ppl_bf16.  Random-init weights
make ppl noisy, so logit KL and per-layer weight MSE are reported
alongside as fidelity checks."""
from __future__ import annotations

import math

import torch

from ..models.qlinear import QLinear


@torch.no_grad()
def _nll_and_logits(model, ids, device):
    ids = ids.to(device)
    cache = model.make_cache(ids.shape[0], max_seq=ids.shape[1], device=device)
    logits = model(ids, cache).float()
    lsm = torch.log_softmax(logits[:, :-1], dim=-1)
    tgt = ids[:, 1:]
    nll = -lsm.gather(-1, tgt.unsqueeze(-1)).squeeze(-1)
    return nll.mean(), logits


@torch.no_grad()
def bench_ppl(fp_model, q_model, vocab_size: int, seq_len: int = 2048,
              n_seqs: int = 4, seed: int = 777, device="cpu") -> dict:
    g = torch.Generator().manual_seed(seed)
    nll_fp, nll_q, kl_sum, nb = 0.0, 0.0, 0.0, 0
    for _ in range(n_seqs):
        ids = torch.randint(0, vocab_size, (1, seq_len), generator=g)
        a, lf = _nll_and_logits(fp_model, ids, device)
        b, lq = _nll_and_logits(q_model, ids, device)
        nll_fp += float(a)
        nll_q += float(b)
        pf = torch.log_softmax(lf, -1)
        pq = torch.log_softmax(lq, -1)
        kl_sum += float((pf.exp() * (pf - pq)).sum(-1).mean())
        nb += 1
    ppl_fp = math.exp(nll_fp / nb)
    ppl_q = math.exp(nll_q / nb)
    return {
        "ppl_fp16": ppl_fp,
        "ppl_quant": ppl_q,
        "delta_ppl": ppl_q - ppl_fp,
        "logit_kl": kl_sum / nb,
        "seq_len": seq_len,
        "n_seqs": n_seqs,
    }


@torch.no_grad()
def weight_mse(fp_model, q_model) -> dict:
    """Per-layer relative weight MSE between fp Linears and QLinears."""
    fp = dict(fp_model.named_modules())
    out = {}
    for name, mod in q_model.named_modules():
        if isinstance(mod, QLinear) and name in fp and hasattr(fp[name], "weight"):
            w = fp[name].weight.float().cpu()
            dq = mod.dequant()
            out[name] = float(((w - dq) ** 2).mean() / (w ** 2).mean())
    return out
