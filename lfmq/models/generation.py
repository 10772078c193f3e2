"""Shared eager autoregressive generation for all model families.

Greedy by default (the BASELINE decode metric is greedy); a
GenerationConfig adds temperature sampling and eos early-stop for
serving (lfmq/obs/serve.py).  The hipGraph decode path
(lfmq/models/graph.py) stays greedy-only — sampling there runs through
the on-device argmax kernel.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..config import GenerationConfig
from ..ops import functional as F


def _next_token(logits_last: torch.Tensor, gc: Optional[GenerationConfig],
                generator: Optional[torch.Generator]) -> torch.Tensor:
    """logits_last [b, vocab] -> token [b, 1]."""
    if gc is None or gc.greedy:
        return F.greedy_sample(logits_last).unsqueeze(1)
    probs = torch.softmax(logits_last.float()
                          / max(gc.temperature, 1e-5), dim=-1)
    return torch.multinomial(probs, 1, generator=generator)


@torch.no_grad()
def autoregressive_generate(model, input_ids: torch.Tensor,
                            max_new_tokens: int = 32, cache=None,
                            gen_cfg: Optional[GenerationConfig] = None,
                            generator: Optional[torch.Generator] = None
                            ) -> torch.Tensor:
    """Prefill + token-by-token decode through model.forward.

    gen_cfg.max_new_tokens overrides max_new_tokens when set;
    gen_cfg.eos_token_id stops early once EVERY row has emitted eos
    (the returned tensor ends at that step — rows that finished earlier
    keep decoding until the slowest row is done, as is standard for
    batched eager generation)."""
    if gen_cfg is not None and gen_cfg.max_new_tokens:
        max_new_tokens = gen_cfg.max_new_tokens
    b, s = input_ids.shape
    if cache is None:
        cache = model.make_cache(b, max_seq=s + max_new_tokens + 8)
    eos = gen_cfg.eos_token_id if gen_cfg is not None else None
    done = torch.zeros(b, dtype=torch.bool, device=input_ids.device)

    logits = model.forward(input_ids, cache)
    out = [input_ids]
    tok = _next_token(logits[:, -1], gen_cfg, generator)
    for _ in range(max_new_tokens - 1):
        out.append(tok)
        if eos is not None:
            done |= (tok.squeeze(1) == eos)
            if bool(done.all()):
                return torch.cat(out, dim=1)
        logits = model.forward(tok, cache)
        tok = _next_token(logits[:, -1], gen_cfg, generator)
    out.append(tok)
    return torch.cat(out, dim=1)
