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
