#!/usr/bin/env python3
"""Run the engine with corruption detectors (debug 8|16) and report the
detector codes + determinism over repeated runs."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["LFMQ_ENGINE_DEBUG"] = "24"  # 8 | 16
import lfmq
from lfmq.models.llama import LlamaConfig


def main():
    cfg = LlamaConfig(vocab_size=1024, hidden_size=4096,
                      intermediate_size=14336, n_layers=2, n_heads=32,
                      n_kv_heads=8, head_dim=128, max_seq=1024)
    m = lfmq.build_model(cfg).to_empty(device="cuda")
    torch.manual_seed(7)
    for p in m.parameters():
        p.data.normal_(0.0, 0.02)
    q = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4,
                                          group_size=128))

    @torch.no_grad()
    def run():
        q._engine = None
        q._engine_unsupported = False
        os.environ["LFMQ_ENGINE"] = "1"
        cache = q.make_cache(1, max_seq=1024, device="cuda")
        ids = torch.randint(0, 1024, (1, 300),
                            generator=torch.Generator().manual_seed(5)).cuda()
        q(ids, cache)
        state = torch.tensor([300, 301, 0], dtype=torch.int32, device="cuda")
        xo = []
        g = torch.Generator().manual_seed(11)
        for i in range(6):
            tok = torch.randint(0, 1024, (1, 1), generator=g).cuda()
            q.forward_decode_step(tok, cache, state, 1024)
            xo.append(q._engine.xout.float().clone())
            state += torch.tensor([1, 1, 1], dtype=torch.int32, device="cuda")
            cache.seq_len += 1
        t = q._engine.tmo.cpu().tolist()
        print(f"  tmo: giveup={t[0]:#x} dblread={t[5]:#x} restage={t[6]:#x} xblocks={t[7]:#06x}")
        q._engine.tmo.zero_()
        return torch.cat(xo)

    ref = run()
    for rep in range(4):
        got = run()
        d = (got - ref).abs().max(dim=-1).values.flatten()
        print(f"rep {rep}: per-step xout max|diff| = "
              f"{[f'{float(v):.5f}' for v in d]}")


if __name__ == "__main__":
    main()
