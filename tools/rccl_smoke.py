#!/usr/bin/env python3
"""World-1 RCCL smoke on a real MI355X.

Every distributed test in the suite runs under gloo; the RCCL backend
("nccl" on ROCm) differs only by backend string, but graph capture of
collectives and device-tensor all_to_all were never executed against RCCL
itself.  This exercises, in one process (rank 0 of world 1):
  1. RCCL process-group init
  2. bf16 all_reduce / all_gather / all_to_all_single (with splits) on GPU
  3. hipGraph capture + replay of a step containing an all_reduce
  4. the EP static dispatch path (capacity-padded all_to_all) on device
Prints one status line per stage; exits nonzero on any failure.
"""
import os
import sys

import torch
import torch.distributed as dist


def main():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29617")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    assert torch.cuda.is_available(), "needs a GPU"
    dev = torch.device("cuda", 0)
    torch.cuda.set_device(dev)

    dist.init_process_group("nccl", rank=0, world_size=1)
    print(f"[rccl_smoke] init ok: backend={dist.get_backend()} "
          f"world={dist.get_world_size()}")

    # --- collectives ---
    t = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    ref = t.clone()
    dist.all_reduce(t)
    assert torch.equal(t, ref), "world-1 all_reduce should be identity"
    parts = [torch.empty_like(t)]
    dist.all_gather(parts, t)
    assert torch.equal(parts[0], t)
    out = torch.empty_like(t)
    dist.all_to_all_single(out, t, [t.numel()], [t.numel()])
    assert torch.equal(out, t)
    print("[rccl_smoke] collectives ok (all_reduce/all_gather/all_to_all)")

    # --- graph capture containing a collective ---
    x = torch.randn(1, 4096, dtype=torch.bfloat16, device=dev)
    w = torch.randn(4096, 4096, dtype=torch.bfloat16, device=dev)
    static_y = torch.zeros_like(x)
    g = torch.cuda.CUDAGraph()
    # warmup on a side stream BEFORE capture: hipBLASLt's first-call
    # workspace setup is capture-illegal (same reason GraphDecoder warms up)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            y = x @ w
            dist.all_reduce(y)
            static_y.copy_(y)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    capture_ok = True
    try:
        with torch.cuda.graph(g):
            y = x @ w
            dist.all_reduce(y)
            static_y.copy_(y)
        g.replay()
        torch.cuda.synchronize()
        want = x @ w
        rel = (static_y.float() - want.float()).abs().max() / \
            want.float().abs().max()
        assert float(rel) < 1e-2, f"graph replay mismatch {float(rel)}"
    except Exception as e:
        capture_ok = False
        print(f"[rccl_smoke] GRAPH CAPTURE FAILED: {e}")
    print(f"[rccl_smoke] graph capture of all_reduce: "
          f"{'ok' if capture_ok else 'FAILED'}")

    # --- EP static dispatch path on device ---
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import lfmq
    from lfmq.models.mixtral import MixtralConfig
    from lfmq.dist.ep import ExpertParallelMoE
    from lfmq.models.mixtral import MixtralMoE
    cfg = MixtralConfig.tiny()
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=5),
                       lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32))
    qm = qm.to(dev)
    moe = None
    for m in qm.modules():
        if isinstance(m, MixtralMoE):
            moe = m
            break
    ep = ExpertParallelMoE(moe, 0, 1).to(dev)
    xin = torch.randn(1, 3, cfg.hidden_size, dtype=next(qm.parameters()).dtype,
                      device=dev)
    y_ref = moe(xin)
    y_ep = ep(xin)
    rel = (y_ep.float() - y_ref.float()).abs().max() / \
        (y_ref.float().abs().max() + 1e-9)
    assert float(rel) < 0.03, f"EP static path mismatch {float(rel)}"
    print("[rccl_smoke] EP static dispatch (device all_to_all) ok")

    dist.destroy_process_group()
    print(f"[rccl_smoke] PASS graph_capture={capture_ok}")
    return 0 if capture_ok else 1


if __name__ == "__main__":
    sys.exit(main())
