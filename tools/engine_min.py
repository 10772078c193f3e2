#!/usr/bin/env python3
"""Minimal engine repro: 1 layer, 2 steps; prints tmo codes."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["LFMQ_ENGINE_PROF"] = "1"
import lfmq
from lfmq.models.llama import LlamaConfig

cfg = LlamaConfig(vocab_size=1024, hidden_size=4096, intermediate_size=14336,
                  n_layers=1, n_heads=32, n_kv_heads=8, head_dim=128,
                  max_seq=512)
m = lfmq.build_model(cfg).to_empty(device="cuda")
torch.manual_seed(7)
for p in m.parameters():
    p.data.normal_(0.0, 0.02)
q = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4, group_size=128))
cache = q.make_cache(1, max_seq=512, device="cuda")
ids = torch.randint(0, 1024, (1, 32)).cuda()
q(ids, cache)
state = torch.tensor([32, 33, 0], dtype=torch.int32, device="cuda")
os.environ["LFMQ_ENGINE"] = "1"
for i in range(2):
    tok = torch.tensor([[5]], device="cuda")
    lg = q.forward_decode_step(tok, cache, state, 512)
    torch.cuda.synchronize()
    t = q._engine.tmo.cpu().tolist()
    print(f"step {i}: tmo={[hex(v) for v in t]} logit_max={float(lg.float().abs().max()):.4f}")
    pr = q._engine.prof[:8].cpu().tolist()
    print(f"   phase stamps nonzero: {[j for j, v in enumerate(pr) if v]}")
    state += torch.tensor([1, 1, 1], dtype=torch.int32, device="cuda")
    cache.seq_len += 1
print("DONE")
