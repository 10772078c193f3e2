"""CPU CLI coverage: quantize -> eval round trip on the tiny preset
(the GPU variant lives in test_gpu_e2e.py::test_cli_quantize_eval_gpu)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, **kw):
    return subprocess.run([sys.executable, "-m", "lfmq.cli"] + args,
                          capture_output=True, text=True, cwd=REPO,
                          timeout=300, **kw)


def test_cli_quantize_eval_cpu(tmp_path):
    ck = str(tmp_path / "ck")
    r = _run(["quantize", "--model", "tiny-llama", "--scheme", "rtn",
              "--group-size", "32", "--out", ck, "--seed", "3"])
    assert r.returncode == 0, r.stderr[-1500:]
    assert os.path.exists(os.path.join(ck, "model.safetensors"))
    r = _run(["eval", "--ckpt", ck, "--seq-len", "32", "--n-seqs", "1",
              "--model-seed", "3"])
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout[r.stdout.index("{"):])
    assert "delta_ppl" in out and "logit_kl" in out
    # quantization noise only: the matching-seed bf16 reference keeps the
    # divergence small; a wrong model_seed would give KL ~ ln(vocab)
    assert out["logit_kl"] < 1.0, out
    # weight_mse must compare against a norm-FOLDED reference: the
    # unfolded-vs-folded mistake reads ~0.5 rel MSE, true W4 error ~1e-2
    assert out["weight_mse_mean"] < 0.05, out


def test_cli_info(tmp_path):
    ck = str(tmp_path / "ck")
    r = _run(["quantize", "--model", "tiny-llama", "--scheme", "rtn",
              "--group-size", "32", "--out", ck])
    assert r.returncode == 0, r.stderr[-1500:]
    r = _run(["info", "--ckpt", ck])
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout[r.stdout.index("{"):])
    assert out["scheme"] == "rtn" and out["bits"] == 4
    assert out["quantized_linears"] > 0 and out["file_bytes"] > 0


def test_cli_eval_multiseed(tmp_path):
    """--seeds N averages the fidelity metrics over N evaluation streams
    and reports the spread (single-seed delta-ppl is noise at random-init
    resolution — profiles/gptq_fidelity_r02.md)."""
    ck = str(tmp_path / "ck")
    r = _run(["quantize", "--model", "tiny-llama", "--scheme", "rtn",
              "--group-size", "32", "--out", ck, "--seed", "3"])
    assert r.returncode == 0, r.stderr[-1500:]
    r = _run(["eval", "--ckpt", ck, "--seq-len", "32", "--n-seqs", "1",
              "--model-seed", "3", "--seeds", "3"])
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout[r.stdout.index("{"):])
    assert out["n_eval_seeds"] == 3
    for key in ("delta_ppl", "logit_kl"):
        assert key in out and key + "_std" in out
        assert out[key + "_std"] >= 0.0


def test_cli_export_hf_roundtrip(tmp_path):
    """quantize -> export-hf -> import back into a fresh fp model: the
    dequantized HF-layout checkpoint must reproduce the quantized
    model's logits (bf16 re-rounding tolerance)."""
    import torch

    import lfmq
    from lfmq.models.import_hf import import_hf_state

    ck = str(tmp_path / "ck")
    r = _run(["quantize", "--model", "tiny-llama", "--scheme", "rtn",
              "--group-size", "32", "--out", ck, "--seed", "7"])
    assert r.returncode == 0, r.stderr[-1500:]
    hf = str(tmp_path / "hf")
    r = _run(["export-hf", "--ckpt", ck, "--out", hf])
    assert r.returncode == 0, r.stderr[-1500:]
    assert os.path.exists(os.path.join(hf, "model.safetensors"))

    qm = lfmq.load(ck)
    from lfmq.models.llama import LlamaConfig
    fp = lfmq.build_model(LlamaConfig.tiny(), seed=0)
    import_hf_state(fp, hf)
    ids = torch.arange(10).reshape(1, 10)
    a = qm(ids, qm.make_cache(1, 16)).float()
    b = fp(ids, fp.make_cache(1, 16)).float()
    rel = (a - b).abs().max() / a.abs().max()
    assert rel < 0.02, float(rel)


def test_export_hf_mixtral_and_opt():
    """Mixtral round-trips through the expert-name mapping; OPT raises a
    clear NotImplementedError (HF OPT layout is a different tree)."""
    import pytest
    import torch

    import lfmq
    from lfmq.models.export_hf import export_hf_state
    from lfmq.models.import_hf import convert_hf_state, import_hf_state
    from lfmq.models.mixtral import MixtralConfig
    from lfmq.models.opt import OPTConfig

    cfg = MixtralConfig.tiny()
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=4),
                       lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32))
    state = export_hf_state(qm)
    assert any("block_sparse_moe.experts.0.w1.weight" in k for k in state)
    fp = lfmq.build_model(cfg, seed=0)
    import_hf_state(fp, state)
    ids = torch.arange(8).reshape(1, 8)
    a = qm(ids, qm.make_cache(1, 16)).float()
    b = fp(ids, fp.make_cache(1, 16)).float()
    rel = (a - b).abs().max() / a.abs().max()
    assert rel < 0.02, float(rel)

    om = lfmq.build_model(OPTConfig.tiny(), seed=0)
    with pytest.raises(NotImplementedError):
        export_hf_state(om)
