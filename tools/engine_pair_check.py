#!/usr/bin/env python3
"""Validate the record-pair ILP engine path in one short GPU pass:
parity vs the launched fused path (teacher-forced), bit-determinism,
and per-phase prof deltas.  Designed to fit one small gpurun call."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["LFMQ_ENGINE_PROF"] = "1"
import torch

import lfmq
from lfmq.models.llama import LlamaConfig

cfg = LlamaConfig(vocab_size=2048, hidden_size=4096, intermediate_size=14336,
                  n_layers=2, n_heads=32, n_kv_heads=8, head_dim=128,
                  max_seq=512)
m = lfmq.build_model(cfg).to_empty(device="cuda")
torch.manual_seed(11)
for p in m.parameters():
    p.data.normal_(0.0, 0.02)
q = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4, group_size=128))

torch.manual_seed(3)
prompt = torch.randint(0, 2048, (1, 32)).cuda()
steps = torch.randint(0, 2048, (12,)).tolist()


def run(engine: bool):
    os.environ["LFMQ_ENGINE"] = "1" if engine else "0"
    cache = q.make_cache(1, max_seq=512, device="cuda")
    q(prompt, cache)
    state = torch.tensor([32, 33, 0], dtype=torch.int32, device="cuda")
    outs = []
    for tid in steps:
        tok = torch.tensor([[tid]], device="cuda")
        lg = q.forward_decode_step(tok, cache, state, 512)
        torch.cuda.synchronize()
        outs.append(lg.float().clone())
        state += torch.tensor([1, 1, 1], dtype=torch.int32, device="cuda")
        cache.seq_len += 1
    return outs


ref = run(False)
eng1 = run(True)
# tmo[1..4] are spin-statistic ACCUMULATORS (nonzero in any healthy run);
# failure indicators are tmo[0] (first give-up code), tmo[5] (corruption
# detector) and tmo[6] (double-writer / stale sweep).
tmo = q._engine.tmo.cpu().tolist()
assert tmo[0] == 0 and tmo[5] == 0 and tmo[6] == 0, \
    f"engine fault: {[hex(v) for v in tmo]}"
eng2 = run(True)

worst = 0.0
for a, b in zip(ref, eng1):
    d = (a - b).abs().max().item() / (a.abs().max().item() + 1e-9)
    worst = max(worst, d)
det = all(torch.equal(a, b) for a, b in zip(eng1, eng2))
print(f"parity worst rel (12 teacher-forced steps): {worst:.2e}")
print(f"bit-deterministic across runs: {det}")

# timing: median per-step latency, engine vs launched, same 2-layer model
def clock(engine: bool, n=60):
    os.environ["LFMQ_ENGINE"] = "1" if engine else "0"
    cache = q.make_cache(1, max_seq=512, device="cuda")
    q(prompt, cache)
    state = torch.tensor([32, 33, 0], dtype=torch.int32, device="cuda")
    tok = torch.tensor([[7]], device="cuda")
    for _ in range(10):
        q.forward_decode_step(tok, cache, state, 512)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        q.forward_decode_step(tok, cache, state, 512)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


ms_eng = clock(True)
ms_ref = clock(False)
print(f"2-layer step: engine {ms_eng:.3f} ms vs launched {ms_ref:.3f} ms")
prof = q._engine.prof.cpu().tolist()
nz = [(i, v) for i, v in enumerate(prof[:16]) if v]
print(f"phase stamps (cycles @2GHz): {nz}")
ok = worst < 2e-3 and det
print("PAIR_CHECK", "PASS" if ok else "FAIL")
sys.exit(0 if ok else 1)
