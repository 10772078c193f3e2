#!/usr/bin/env python3
"""Engine-vs-launched error structure: is the parity gap accumulation
noise (bf16 re-rounding of o/down outputs) or a localized bug?
Prints max/mean rel per configuration; small & flat at 1 layer => noise."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import lfmq
from lfmq.models.llama import LlamaConfig


def build(n_layers, seed=7):
    cfg = LlamaConfig(vocab_size=1024, hidden_size=4096,
                      intermediate_size=14336, n_layers=n_layers, n_heads=32,
                      n_kv_heads=8, head_dim=128, max_seq=1024)
    m = lfmq.build_model(cfg).to_empty(device="cuda")
    torch.manual_seed(seed)
    for p in m.parameters():
        p.data.normal_(0.0, 0.02)
    return lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4,
                                             group_size=128)), cfg


@torch.no_grad()
def steps(model, cfg, ids, nsteps, variant, seed=11):
    """variant: 'engine' | 'fused' (launched, norm-fused GEMVs) |
    'nonfused' (launched, separate norm kernels — the numerical-noise
    yardstick between two valid launched implementations)."""
    use_engine = variant == "engine"
    model._engine = None
    model._engine_unsupported = False
    model._fused_decode = None if variant != "nonfused" else False
    os.environ["LFMQ_ENGINE"] = "1" if use_engine else "0"
    g = torch.Generator().manual_seed(seed)
    forced = torch.randint(0, cfg.vocab_size, (nsteps,), generator=g)
    cache = model.make_cache(1, max_seq=1024, device="cuda")
    model(ids, cache)
    state = torch.tensor([ids.shape[1], ids.shape[1] + 1, 0],
                         dtype=torch.int32, device="cuda")
    outs = []
    for i in range(nsteps):
        tok = forced[i].reshape(1, 1).to("cuda")
        lg = model.forward_decode_step(tok, cache, state, 1024)
        outs.append(lg[:, -1].float().clone())
        state += torch.tensor([1, 1, 1], dtype=torch.int32, device="cuda")
        cache.seq_len += 1
    prof = None
    if use_engine:
        assert model._engine is not None
        model._engine.check_health()
        t = model._engine.tmo.cpu().tolist()
        print(f"   spins(last step): consumer_fill={t[1]} loader_space={t[2]} "
              f"sweep_passes={t[3]} sync={t[4]}")
        model._engine.tmo.zero_()
        if model._engine.prof is not None:
            prof = model._engine.prof.clone().cpu()
    os.environ.pop("LFMQ_ENGINE", None)
    model._fused_decode = None
    return torch.cat(outs), prof


def rel(a, b):
    return float((a - b).abs().max() / b.abs().max()), \
        float((a - b).abs().mean() / b.abs().mean())


PHASES = ["layer_top", "qkv_rec", "qkv_gath", "attn", "attn_gath",
          "o+gath+inv", "gu_rec", "dn_rec"]


def main():
    torch.manual_seed(0)
    os.environ["LFMQ_ENGINE_PROF"] = "1"
    for nl in (2, 8):
        q, cfg = build(nl)
        for prompt, ns in ((64, 8), (300, 4)):
            ids = torch.randint(0, cfg.vocab_size, (1, prompt), device="cuda")
            ref, _ = steps(q, cfg, ids, ns, "fused")
            ref2, _ = steps(q, cfg, ids, ns, "nonfused")
            got, prof = steps(q, cfg, ids, ns, "engine")
            got2, _ = steps(q, cfg, ids, ns, "engine")
            det = float((got - got2).abs().max())
            print(f"   engine determinism: max|run1-run2|={det:.6f}")
            eng_abs = [float((got[i] - ref[i]).abs().max()) for i in range(ns)]
            yard = [float((ref2[i] - ref[i]).abs().max()) for i in range(ns)]
            scale = [float(ref[i].abs().max()) for i in range(ns)]
            print(f"layers={nl} prompt={prompt} steps={ns}:")
            print(f"   eng_abs={[f'{v:.4f}' for v in eng_abs]}")
            print(f"   yardstick(fused-vs-nonfused)={[f'{v:.4f}' for v in yard]}")
            print(f"   ref_step_scale={[f'{v:.3f}' for v in scale]}")
            if prof is not None:
                p = prof[:nl * 8].reshape(nl, 8).double()
                # deltas within layer (stamps 0..7) + layer-to-layer
                intra = (p[:, 1:] - p[:, :-1])
                l2l = p[1:, 0] - p[:-1, 0] if nl > 1 else None
                med = intra.median(dim=0).values.tolist()
                names = [f"{PHASES[i+1]}:{med[i]/2.0e3:.1f}us"
                         for i in range(7)]
                tot = float((p[-1, 7] - p[0, 0]) / 2.0e3)
                print(f"   phase medians (assuming 2GHz): {names}; "
                      f"stamp-span/step={tot:.1f}us")
                lts = prof[nl * 8:].double()
                lts = lts[lts > 0]
                if lts.numel() > 2:
                    d = (lts[1:] - lts[:-1]) / 2.0e3  # us between slot issues
                    dd = d.sort().values
                    print(f"   loader: {lts.numel()} slots, slot-issue deltas "
                          f"med={float(dd[len(dd)//2]):.2f}us "
                          f"p90={float(dd[int(len(dd)*0.9)]):.2f}us "
                          f"max={float(dd[-1]):.2f}us "
                          f"span={float((lts[-1]-lts[0])/2.0e3):.1f}us")
        del q
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
