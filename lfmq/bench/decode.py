"""Decode tok/s harness (the BASELINE.json headline metric).

Greedy decode on synthetic data / random-init weights (no network => no
pretrained checkpoints).  Timing protocol: untimed prefill +
warmup decode steps, then barrier + synchronize, time EXACTLY K decode
steps, synchronize + barrier, MAX over ranks.  Nothing is skipped inside
the timed region: every step runs all layers + lm_head + argmax and feeds
the sampled token back.
"""
from __future__ import annotations

import time
from typing import Optional

import torch

from .. import dist as lfmq_dist
from ..api import build_model, quantize
from ..config import QuantConfig
from ..models.llama import LlamaConfig


@torch.no_grad()
def _fast_random_init(model, seed: int):
    torch.manual_seed(seed)
    for p in model.parameters():
        p.data.normal_(0.0, 0.02)
    return model


def _bench_config(model_name: str, n_layers: Optional[int] = None):
    if model_name == "llama3-8b":
        cfg = LlamaConfig.llama3_8b()
    elif model_name == "llama3-70b":
        cfg = LlamaConfig.llama3_70b()
    elif model_name == "mixtral-8x7b":
        from ..models.mixtral import MixtralConfig
        cfg = MixtralConfig.mixtral_8x7b()
    elif model_name == "tiny-debug":
        # contract-test shape only (tests/test_bench_contract.py);
        # never a reportable benchmark config
        cfg = LlamaConfig(vocab_size=512, hidden_size=256,
                          intermediate_size=512, n_layers=2, n_heads=4,
                          n_kv_heads=2, head_dim=64, max_seq=256)
    elif model_name == "tiny-debug-tp8":
        # contract-test shape that shards at TP 4 and 8 (8 q/kv heads,
        # row-shard k dims divisible by 8 ranks x group 32) — exercises
        # the N=4/8 rank math of the driver's scale run on CPU/gloo
        cfg = LlamaConfig(vocab_size=512, hidden_size=512,
                          intermediate_size=1024, n_layers=2, n_heads=8,
                          n_kv_heads=8, head_dim=64, max_seq=256)
    else:
        raise ValueError(model_name)
    if n_layers:
        cfg.n_layers = n_layers
    return cfg


@torch.no_grad()
def build_bf16_llama(model_name: str, device, seed: int = 0,
                     n_layers: Optional[int] = None):
    """The un-quantized reference with the SAME init as the bench model
    (same cfg/seed/init path as build_quantized_llama before quantize) —
    the fp baseline for the contract's delta-ppl/logit-KL fidelity field."""
    cfg = _bench_config(model_name, n_layers)
    model = build_model(cfg)
    model = model.to_empty(device=str(device)) if str(device) != "cpu" else model
    _fast_random_init(model, seed)
    return model, cfg


@torch.no_grad()
def build_quantized_llama(model_name: str, scheme: str, bits: int, g: int,
                          device, tp_rank: int = 0, tp_world: int = 1,
                          seed: int = 0, n_layers: Optional[int] = None,
                          quant_lm_head="bf16"):
    """Random-init on device, quantize (RTN weights for bench speed: the
    decode kernel path and memory traffic are scheme-independent), shard.
    quant_lm_head: "bf16" (rocBLAS head), "w4" or "w8" (HIP GEMV head;
    legacy bool True == "w4")."""
    cfg = _bench_config(model_name, n_layers)
    model = build_model(cfg)
    model = model.to_empty(device=str(device)) if str(device) != "cpu" else model
    _fast_random_init(model, seed)
    if quant_lm_head is True:
        quant_lm_head = "w4"
    elif quant_lm_head is False or quant_lm_head is None:
        quant_lm_head = "bf16"
    head_same_cfg = quant_lm_head == "w4" and bits == 4
    qcfg = QuantConfig(scheme=scheme, bits=bits, group_size=g,
                       skip_modules=() if head_same_cfg else ("lm_head",))
    model = quantize(model, qcfg)
    if quant_lm_head in ("w4", "w8") and not head_same_cfg:
        import torch.nn as nn

        from ..models.qlinear import QLinear
        from ..quant.common import replace_module
        hb = 4 if quant_lm_head == "w4" else 8
        q = QLinear.from_rtn(model.lm_head,
                             QuantConfig(scheme="rtn", bits=hb, group_size=g))
        replace_module(model, "lm_head", q)
    if tp_world > 1:
        if cfg.arch == "mixtral":
            from ..dist.ep import apply_expert_parallel
            model = apply_expert_parallel(model, tp_rank, tp_world)
        else:
            from ..dist.tp import shard_llama
            model = shard_llama(model, tp_rank, tp_world)
    if cfg.arch == "mixtral" and hasattr(model, "consolidate_moe"):
        model.consolidate_moe()
    return model, cfg, qcfg


@torch.no_grad()
def bench_decode(model, cfg, batch: int, prompt: int, steps: int, warmup: int,
                 device, use_graph: bool = True) -> dict:
    ids = torch.randint(0, cfg.vocab_size, (batch, prompt), device=device)
    use_cuda = torch.cuda.is_available() and str(device) != "cpu"
    graphed = False
    if use_cuda and use_graph:
        try:
            from ..models.graph import GraphDecoder
            dec = GraphDecoder(model, batch,
                               max_seq=prompt + warmup + steps + 8,
                               max_new_tokens=warmup + steps + 4)
            dec.prefill(ids)
            dec.decode(max(warmup, 1))   # includes one-time capture
            graphed = True
        except Exception as e:  # pragma: no cover - graph capture fallback
            import sys
            print(f"[bench] hipGraph capture failed ({e}); eager decode",
                  file=sys.stderr)
            try:
                torch.cuda.synchronize()  # leave no half-captured stream state
            except Exception:
                pass
        if lfmq_dist.get_world_size() > 1:
            # All ranks must agree on the mode: a graphed rank replays its
            # captured collectives while an eager rank issues fresh ones —
            # a split decision deadlocks the TP all-reduces.  MIN-reduce
            # the outcome and drop to eager everywhere if any rank failed.
            import sys

            import torch.distributed as dist
            ok = torch.tensor([1.0 if graphed else 0.0], dtype=torch.float32)
            ok = ok.to(device)
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if graphed and float(ok.cpu()[0]) < 1.0:
                print("[bench] peer rank failed graph capture; all ranks "
                      "fall back to eager", file=sys.stderr)
                graphed = False
                del dec
                torch.cuda.synchronize()
    if not graphed:
        cache = model.make_cache(batch, max_seq=prompt + warmup + steps + 8,
                                 device=device)
        logits = model(ids, cache)          # prefill (untimed)
        tok = logits[:, -1:].argmax(dim=-1)
        for _ in range(warmup):
            logits = model(tok, cache)
            tok = logits[:, -1:].argmax(dim=-1)

    lfmq_dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if graphed:
        dec.decode(steps)
    else:
        for _ in range(steps):
            logits = model(tok, cache)
            tok = logits[:, -1:].argmax(dim=-1)
    if use_cuda:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    lfmq_dist.barrier()

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if lfmq_dist.get_world_size() > 1:
        import torch.distributed as dist
        e = elapsed.to(device) if use_cuda else elapsed
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = e.cpu()
    el = float(elapsed[0])
    return {
        "elapsed_s": el,
        "ms_per_step": el / steps * 1000.0,
        "tok_per_s": batch * steps / el,
        "steps": steps,
        "warmup": warmup,
        "batch": batch,
        "prompt": prompt,
        "hip_graph": graphed,
    }
