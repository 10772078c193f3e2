#!/usr/bin/env python3
"""Where does run-to-run nondeterminism enter? Compare:
(a) launched path logits, run twice
(b) engine xout (pre-lm_head hidden), run twice
(c) engine logits, run twice
(d) fixed xout -> lm_head twice (isolates hipBLASLt GEMV)
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import lfmq
from lfmq.models.llama import LlamaConfig


def build():
    cfg = LlamaConfig(vocab_size=1024, hidden_size=4096,
                      intermediate_size=14336, n_layers=2, n_heads=32,
                      n_kv_heads=8, head_dim=128, max_seq=1024)
    m = lfmq.build_model(cfg).to_empty(device="cuda")
    torch.manual_seed(7)
    for p in m.parameters():
        p.data.normal_(0.0, 0.02)
    return lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4,
                                             group_size=128)), cfg


@torch.no_grad()
def run(q, cfg, engine, collect_xout=False):
    q._engine = None
    q._engine_unsupported = False
    os.environ["LFMQ_ENGINE"] = "1" if engine else "0"
    cache = q.make_cache(1, max_seq=1024, device="cuda")
    ids = torch.randint(0, 1024, (1, 300),
                        generator=torch.Generator().manual_seed(5)).cuda()
    q(ids, cache)
    state = torch.tensor([300, 301, 0], dtype=torch.int32, device="cuda")
    outs, xouts = [], []
    g = torch.Generator().manual_seed(11)
    for i in range(6):
        tok = torch.randint(0, 1024, (1, 1), generator=g).cuda()
        lg = q.forward_decode_step(tok, cache, state, 1024)
        outs.append(lg[:, -1].float().clone())
        if collect_xout and q._engine is not None:
            xouts.append(q._engine.xout.float().clone())
        state += torch.tensor([1, 1, 1], dtype=torch.int32, device="cuda")
        cache.seq_len += 1
    os.environ.pop("LFMQ_ENGINE", None)
    return torch.cat(outs), (torch.cat(xouts) if xouts else None)


def main():
    q, cfg = build()
    a, _ = run(q, cfg, False)
    b, _ = run(q, cfg, False)
    print(f"launched logits det: {float((a-b).abs().max()):.6f}")
    a, xa = run(q, cfg, True, True)
    b, xb = run(q, cfg, True, True)
    print(f"engine logits det:  {float((a-b).abs().max()):.6f}")
    print(f"engine xout det:    {float((xa-xb).abs().max()):.6f}")
    x = torch.randn(1, 1, cfg.hidden_size, dtype=torch.bfloat16,
                    device="cuda")
    l1 = q.lm_head(x).float()
    l2 = q.lm_head(x).float()
    print(f"lm_head fixed-input det: {float((l1-l2).abs().max()):.6f}")


if __name__ == "__main__":
    main()
