"""lfmq CLI: quantize / eval / bench / serve.

    python -m lfmq.cli quantize --model llama3-8b --scheme gptq --out ckpt/
    python -m lfmq.cli eval --ckpt ckpt/
    python -m lfmq.cli bench --ckpt ckpt/ --steps 128
    python -m lfmq.cli serve --ckpt ckpt/ --port 8000
"""
from __future__ import annotations

import json
import time
from typing import Optional

import torch
import typer

app = typer.Typer(add_completion=False, pretty_exceptions_enable=False)

MODEL_PRESETS = {
    "llama3-8b": ("llama", "llama3_8b"),
    "llama3-70b": ("llama", "llama3_70b"),
    "mixtral-8x7b": ("mixtral", "mixtral_8x7b"),
    "opt-125m": ("opt", "opt_125m"),
    "tiny-llama": ("llama", "tiny"),
    "tiny-mixtral": ("mixtral", "tiny"),
    "tiny-opt": ("opt", "tiny"),
}


def _build_preset(name: str, seed: int, layers: int = 0):
    import lfmq
    from lfmq.models.llama import LlamaConfig
    from lfmq.models.mixtral import MixtralConfig
    from lfmq.models.opt import OPTConfig
    arch, ctor = MODEL_PRESETS[name]
    cfg_cls = {"llama": LlamaConfig, "mixtral": MixtralConfig,
               "opt": OPTConfig}[arch]
    cfg = getattr(cfg_cls, ctor)()
    if layers:
        cfg.n_layers = layers
    return lfmq.build_model(cfg, seed=seed), cfg


def _device() -> str:
    return "cuda:0" if torch.cuda.is_available() else "cpu"


@app.command()
def quantize(model: str = typer.Option("llama3-8b"),
             scheme: str = typer.Option("gptq"),
             bits: int = typer.Option(4),
             group_size: int = typer.Option(128),
             out: str = typer.Option(...),
             seed: int = typer.Option(0),
             layers: int = typer.Option(0, help="override layer count"),
             calib_seqs: int = typer.Option(16),
             calib_len: int = typer.Option(512),
             resume: bool = typer.Option(False, help="resume a partial run"),
             act_order: bool = typer.Option(False),
             sym: bool = typer.Option(False, help="symmetric quantization "
                                      "(no zero-point)"),
             awq_clip: bool = typer.Option(False, help="AWQ per-row weight-"
                                           "clip search (scheme=awq only)"),
             hf_dir: Optional[str] = typer.Option(
                 None, help="local HF-transformers checkpoint dir "
                 "(safetensors shards) to load into the preset before "
                 "quantizing; default is random-init")):
    """Build a model (random-init, or --hf-dir pretrained weights),
    calibrate, quantize and write the packed checkpoint."""
    import lfmq
    from lfmq.obs import phase_timer

    dev = _device()
    qcfg = lfmq.QuantConfig(scheme=scheme, bits=bits, group_size=group_size,
                            calib_n_seqs=calib_seqs, calib_seq_len=calib_len,
                            act_order=act_order, sym=sym, awq_clip=awq_clip)
    with phase_timer("build model"):
        m, cfg = _build_preset(model, seed, layers)
        if hf_dir is not None:
            from lfmq.models.import_hf import import_hf_state
            import_hf_state(m, hf_dir)
        m = m.to(dev)
    with phase_timer(f"{scheme} quantize"):
        if scheme == "gptq" and resume:
            from lfmq.quant.gptq import GPTQQuantizer
            m = GPTQQuantizer(qcfg).run(m, device=torch.device(dev),
                                        resume_dir=out)
        else:
            m = lfmq.quantize(m, qcfg, device=torch.device(dev))
    with phase_timer("save packed checkpoint"):
        lfmq.save(m, out, qcfg)
    typer.echo(f"saved packed checkpoint to {out}")


@app.command()
def info(ckpt: str = typer.Option(...)):
    """Inspect a packed checkpoint: quant config, model shape, size."""
    import os

    from lfmq.checkpoint import MODEL_FILE, load_meta
    meta = load_meta(ckpt)
    q = meta["quant"]
    mc = meta["model"]
    man = meta.get("manifest", {})
    size = os.path.getsize(os.path.join(ckpt, MODEL_FILE))
    bits_total = sum(s["in"] * s["out"] for s in man.values())
    typer.echo(json.dumps({
        "format_version": meta["format_version"],
        "scheme": q["scheme"], "bits": q["bits"],
        "group_size": q["group_size"], "sym": q["sym"],
        "arch": mc.get("arch"), "n_layers": mc.get("n_layers"),
        "hidden_size": mc.get("hidden_size"),
        "vocab_size": mc.get("vocab_size"),
        "quantized_linears": len(man),
        "quantized_params": bits_total,
        "file_bytes": size,
        "file_gib": round(size / (1 << 30), 3),
    }, indent=2))


@app.command()
def eval(ckpt: str = typer.Option(...),
         seq_len: int = typer.Option(2048),
         n_seqs: int = typer.Option(4),
         seed: int = typer.Option(777),
         seeds: int = typer.Option(1, help="evaluation-stream seeds to "
                                   "average over.  Single-seed delta-ppl "
                                   "is noise at random-init resolution "
                                   "(std 3-8 measured, profiles/"
                                   "gptq_fidelity_r02.md); >=3 reports "
                                   "mean/std per fidelity metric"),
         model_seed: int = typer.Option(0, help="seed the checkpointed "
                                        "model was built with (quantize "
                                        "--seed); the bf16 reference must "
                                        "match it or delta-ppl is noise")):
    """Delta-perplexity vs the bf16 model (BASELINE.md protocol):
    teacher-forced NLL on an identical fixed-seed synthetic stream,
    plus logit KL and per-layer weight MSE (random-init caveat)."""
    import lfmq
    from lfmq.bench.ppl import bench_ppl, weight_mse
    from lfmq.checkpoint import load_meta

    dev = _device()
    meta = load_meta(ckpt)
    qm = lfmq.load(ckpt, device=dev)
    fp = lfmq.build_model(meta["model"], seed=model_seed)
    if meta["quant"].get("fold_rms", False):
        # the checkpoint stores norm-FOLDED projections; fold the bf16
        # reference identically (function-preserving) so weight_mse
        # compares like with like — unfolded-vs-folded reads ~0.5 rel
        # MSE regardless of quantization quality
        from lfmq.quant.fold import fold_rms_norms
        fold_rms_norms(fp)
    fp = fp.to(dev)
    runs = [bench_ppl(fp, qm, qm.cfg.vocab_size, seq_len=seq_len,
                      n_seqs=n_seqs, seed=seed + i, device=dev)
            for i in range(max(seeds, 1))]
    r = dict(runs[0])
    if len(runs) > 1:
        import statistics
        for key in ("delta_ppl", "logit_kl", "ppl_fp16", "ppl_quant"):
            vals = [x[key] for x in runs if x.get(key) is not None]
            if vals:
                r[key] = statistics.fmean(vals)
                r[key + "_std"] = statistics.pstdev(vals)
        r["n_eval_seeds"] = len(runs)
    mse = weight_mse(fp.cpu(), qm.cpu())
    r["weight_mse_mean"] = sum(mse.values()) / max(len(mse), 1)
    typer.echo(json.dumps(r, indent=2))


@app.command()
def export_hf(ckpt: str = typer.Option(...),
              out: str = typer.Option(...)):
    """Dequantize a packed checkpoint back to an HF-transformers-layout
    bf16 safetensors dir (the inverse of quantize --hf-dir)."""
    import lfmq
    from lfmq.models.export_hf import export_hf as _export
    qm = lfmq.load(ckpt)
    fn = _export(qm, out)
    typer.echo(f"wrote {fn}")


@app.command()
def bench(ckpt: Optional[str] = typer.Option(None),
          model: str = typer.Option("llama3-8b"),
          scheme: str = typer.Option("rtn"),
          batch: int = typer.Option(1),
          prompt: int = typer.Option(128),
          steps: int = typer.Option(512),
          warmup: int = typer.Option(16),
          lm_head: str = typer.Option("w8", help="bf16 | w8 | w4 "
                                      "(profiles/lm_head_decision_r02.md)"),
          no_graph: bool = typer.Option(False)):
    """Decode tok/s (the BASELINE.json headline metric)."""
    import lfmq
    from lfmq.bench.decode import bench_decode, build_quantized_llama

    dev = _device()
    if ckpt:
        qm = lfmq.load(ckpt, device=dev)
        cfg = qm.cfg
    else:
        qm, cfg, _ = build_quantized_llama(model, scheme, 4, 128, dev,
                                           quant_lm_head=lm_head)
    r = bench_decode(qm, cfg, batch, prompt, steps, warmup, dev,
                     use_graph=not no_graph)
    typer.echo(json.dumps(r, indent=2))


@app.command()
def serve(ckpt: str = typer.Option(...),
          host: str = typer.Option("127.0.0.1"),
          port: int = typer.Option(8000)):
    """Minimal generation server (fastapi/uvicorn) with prometheus
    counters."""
    import lfmq
    from lfmq.obs.serve import make_app

    dev = _device()
    qm = lfmq.load(ckpt, device=dev)
    import uvicorn
    uvicorn.run(make_app(qm), host=host, port=port, log_level="info")


if __name__ == "__main__":
    app()
