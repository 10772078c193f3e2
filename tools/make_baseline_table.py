#!/usr/bin/env python3
"""Produce the BASELINE.md results rows on a GPU box.

Quantizes Llama-3-8B with GPTQ and AWQ (reduced calibration to fit the
gpurun budget), measures delta-ppl vs the bf16 model and decode tok/s,
and writes gpurun_out/baseline_rows.json.

    python tools/make_baseline_table.py [--schemes gptq,awq] [--layers 0]
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import lfmq  # noqa: E402
from lfmq.bench.decode import bench_decode, _fast_random_init  # noqa: E402
from lfmq.bench.ppl import bench_ppl  # noqa: E402
from lfmq.models.llama import LlamaConfig  # noqa: E402
from lfmq.quant.calib import CalibStream  # noqa: E402

DEV = "cuda:0"


def run_scheme(scheme: str, layers: int, steps: int) -> dict:
    cfg = LlamaConfig.llama3_8b()
    if layers:
        cfg.n_layers = layers
    t0 = time.perf_counter()
    fp = lfmq.build_model(cfg)
    fp = fp.to_empty(device=DEV)
    _fast_random_init(fp, 0)
    qcfg = lfmq.QuantConfig(scheme=scheme, bits=4, group_size=128,
                            calib_n_seqs=4, calib_seq_len=256, awq_grid=10)
    calib = CalibStream(cfg.vocab_size, 256, 4, qcfg.calib_seed)
    qm = lfmq.build_model(cfg)
    qm = qm.to_empty(device=DEV)
    _fast_random_init(qm, 0)
    qm = lfmq.quantize(qm, qcfg, calib=calib, device=torch.device(DEV))
    t_quant = time.perf_counter() - t0

    ppl = bench_ppl(fp, qm, cfg.vocab_size, seq_len=2048, n_seqs=2,
                    seed=777, device=DEV)
    del fp
    torch.cuda.empty_cache()
    dec = bench_decode(qm, cfg, batch=1, prompt=128, steps=steps, warmup=16,
                       device=DEV)
    return {"scheme": scheme, "quant_seconds": t_quant, **ppl, **dec}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--schemes", default="gptq,awq")
    ap.add_argument("--layers", type=int, default=0)
    ap.add_argument("--steps", type=int, default=256)
    args = ap.parse_args()
    os.makedirs("gpurun_out", exist_ok=True)
    rows = []
    for scheme in args.schemes.split(","):
        try:
            r = run_scheme(scheme, args.layers, args.steps)
            print(json.dumps(r))
            rows.append(r)
        except Exception as e:
            print(f"{scheme} FAILED: {type(e).__name__}: {e}", flush=True)
        torch.cuda.empty_cache()
    with open("gpurun_out/baseline_rows.json", "w") as f:
        json.dump(rows, f, indent=2)


if __name__ == "__main__":
    main()
