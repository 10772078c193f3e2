"""GPU end-to-end: quantize -> save -> load -> decode on a 2-layer
8B-shaped random Llama; logit KL vs the CPU reference
path under threshold."""
import pytest
import torch

import lfmq
from lfmq.models.llama import LlamaConfig

pytestmark = pytest.mark.gpu
DEV = "cuda:0"


def _kl(lp, lq):
    pf = torch.log_softmax(lp.float(), -1)
    pq = torch.log_softmax(lq.float(), -1)
    return float((pf.exp() * (pf - pq)).sum(-1).mean())


@pytest.mark.parametrize("scheme", ["rtn", "gptq", "awq"])
def test_e2e_gpu_decode(tmp_path, scheme):
    cfg = LlamaConfig(vocab_size=2048, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=256)
    qcfg = lfmq.QuantConfig(scheme=scheme, bits=4, group_size=64,
                            calib_n_seqs=2, calib_seq_len=64, awq_grid=6)
    model = lfmq.build_model(cfg, seed=1)
    qm = lfmq.quantize(model, qcfg)
    p = str(tmp_path / scheme)
    lfmq.save(qm, p, qcfg)

    # CPU reference forward of the SAME quantized model
    qc = lfmq.load(p)
    ids = torch.randint(0, cfg.vocab_size, (1, 12))
    logits_cpu = qc(ids, qc.make_cache(1, 64))

    qg = lfmq.load(p, device=DEV)
    logits_gpu = qg(ids.to(DEV), qg.make_cache(1, 64))
    kl = _kl(logits_cpu, logits_gpu.cpu())
    assert kl < 0.05, f"{scheme}: GPU-vs-CPU logit KL {kl}"

    # decode path (s=1 kernels: gemv + rope + kv_append + attn_decode)
    out = qg.generate(ids.to(DEV), max_new_tokens=8)
    assert out.shape == (1, 20)


def test_gpu_quantize_gptq():
    """GPTQ calibration ON the GPU (hessian_accum HIP kernel in the loop)."""
    cfg = LlamaConfig(vocab_size=1024, hidden_size=256, intermediate_size=512,
                      n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                      max_seq=256)
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            calib_n_seqs=2, calib_seq_len=64)
    model = lfmq.build_model(cfg, seed=2).to(DEV)
    qm = lfmq.quantize(model, qcfg, device=torch.device(DEV))
    ids = torch.randint(0, cfg.vocab_size, (1, 8), device=DEV)
    logits = qm(ids, qm.make_cache(1, 32))
    assert torch.isfinite(logits.float()).all()


def test_gpu_quantize_awq_clip():
    """AWQ with the clip search ON the GPU — the clip path's tensors must
    live on the weight's device (device-mismatch regression guard)."""
    cfg = LlamaConfig(vocab_size=1024, hidden_size=256, intermediate_size=512,
                      n_layers=1, n_heads=4, n_kv_heads=2, head_dim=64,
                      max_seq=256)
    qcfg = lfmq.QuantConfig(scheme="awq", bits=4, group_size=64,
                            awq_clip=True, awq_grid=4,
                            calib_n_seqs=2, calib_seq_len=64)
    model = lfmq.build_model(cfg, seed=6).to(DEV)
    qm = lfmq.quantize(model, qcfg, device=torch.device(DEV))
    ids = torch.randint(0, cfg.vocab_size, (1, 8), device=DEV)
    logits = qm(ids, qm.make_cache(1, 32))
    assert torch.isfinite(logits.float()).all()


def test_decode_matches_prefill_gpu():
    cfg = LlamaConfig(vocab_size=1024, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=128)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=4), qcfg).to(DEV)
    ids = torch.randint(0, cfg.vocab_size, (1, 8), device=DEV)
    lp = qm(ids, qm.make_cache(1, 32)).float()
    cache = qm.make_cache(1, 32)
    outs = [qm(ids[:, t:t + 1], cache).float() for t in range(8)]
    ld = torch.cat(outs, dim=1)
    rel = (lp - ld).abs().max() / lp.abs().max()
    assert rel < 0.05, f"decode/prefill mismatch rel={float(rel)}"


def test_graph_decoder_matches_eager():
    """hipGraph-captured decode must reproduce eager greedy decode
    (fold_rms=False so both sides run the identical unfused kernels)."""
    from lfmq.models.graph import GraphDecoder
    cfg = LlamaConfig(vocab_size=2048, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=256)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64,
                            fold_rms=False)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=13), qcfg).to(DEV)
    ids = torch.randint(0, cfg.vocab_size, (1, 12), device=DEV)
    eager = qm.generate(ids, max_new_tokens=10)[:, 12:]
    dec = GraphDecoder(qm, batch=1, max_seq=64, max_new_tokens=16)
    dec.prefill(ids)
    got = dec.decode(10)
    assert torch.equal(got.cpu(), eager.cpu()), (got, eager)
    # continue decoding — state must carry across calls
    more = dec.decode(3)
    eager2 = qm.generate(ids, max_new_tokens=13)[:, 12 + 10:]
    assert torch.equal(more.cpu(), eager2.cpu())


def test_fused_fold_decode_matches_unfused():
    """Folded-norm fused decode (6-kernel layer) vs the unfused path on
    the same underlying weights: logits must agree closely."""
    from lfmq.models.graph import GraphDecoder
    cfg = LlamaConfig(vocab_size=2048, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=256)
    torch.manual_seed(31)
    ids = torch.randint(0, cfg.vocab_size, (1, 12), device=DEV)

    qcfg_f = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64, fold_rms=True)
    mf = lfmq.quantize(lfmq.build_model(cfg, seed=7), qcfg_f).to(DEV)
    assert mf._decode_fused_ready()
    df = GraphDecoder(mf, 1, max_seq=64, max_new_tokens=16)
    df.prefill(ids)
    toks_f = df.decode(8)

    qcfg_u = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64, fold_rms=False)
    mu = lfmq.quantize(lfmq.build_model(cfg, seed=7), qcfg_u).to(DEV)
    du = GraphDecoder(mu, 1, max_seq=64, max_new_tokens=16)
    du.prefill(ids)
    toks_u = du.decode(8)
    agree = (toks_f == toks_u).float().mean().item()
    assert agree >= 0.75, f"fused vs unfused token agreement {agree}"


def test_mixtral_gpu_decode():
    """Mixtral on GPU: grouped-kernel decode vs CPU reference logits."""
    from lfmq.models.mixtral import MixtralConfig
    cfg = MixtralConfig(vocab_size=512, hidden_size=256, intermediate_size=512,
                        n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                        n_experts=4, top_k=2, max_seq=128)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=19), qcfg)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    l_cpu = qm(ids, qm.make_cache(1, 32)).float()
    qg = qm.to(DEV).consolidate_moe()
    l_gpu = qg(ids.to(DEV), qg.make_cache(1, 32)).float()
    rel = (l_gpu.cpu() - l_cpu).abs().max() / l_cpu.abs().max()
    assert rel < 0.05, float(rel)
    out = qg.generate(ids.to(DEV), max_new_tokens=6)
    assert out.shape == (1, 14)


def test_graph_decoder_batch8():
    """batch-8 decode (secondary BASELINE metric shape) through the graph
    path (fused kernels are m<=4: the b=8 step must take the general
    path, not crash)."""
    from lfmq.models.graph import GraphDecoder
    cfg = LlamaConfig(vocab_size=1024, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=128)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64, fold_rms=False)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=23), qcfg).to(DEV)
    ids = torch.randint(0, cfg.vocab_size, (8, 10), device=DEV)
    eager = qm.generate(ids, max_new_tokens=6)[:, 10:]
    dec = GraphDecoder(qm, batch=8, max_seq=64, max_new_tokens=8)
    dec.prefill(ids)
    got = dec.decode(6)
    assert torch.equal(got.cpu(), eager.cpu())


def test_w8_decode_gpu():
    """W8A16 end-to-end decode on GPU (config class: w8 kernels wired)."""
    cfg = LlamaConfig(vocab_size=512, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=8, n_kv_heads=2, head_dim=64,
                      max_seq=64)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=8, group_size=128)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=27), qcfg)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    l_cpu = qm(ids, qm.make_cache(1, 32)).float()
    qg = qm.to(DEV)
    l_gpu = qg(ids.to(DEV), qg.make_cache(1, 32)).float()
    rel = (l_gpu.cpu() - l_cpu).abs().max() / l_cpu.abs().max()
    assert rel < 0.05, float(rel)
    out = qg.generate(ids.to(DEV), max_new_tokens=6)
    assert out.shape == (1, 14)


def test_mixtral_graph_decoder():
    """hipGraph capture of the MoE decode step (argsort/scatter/grouped
    kernels in-graph).  Skips with the reason if ROCm refuses capture."""
    from lfmq.models.graph import GraphDecoder
    from lfmq.models.mixtral import MixtralConfig
    cfg = MixtralConfig(vocab_size=512, hidden_size=256, intermediate_size=512,
                        n_layers=2, n_heads=4, n_kv_heads=2, head_dim=64,
                        n_experts=4, top_k=2, max_seq=128)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=33), qcfg).to(DEV)
    qm.consolidate_moe()
    ids = torch.randint(0, cfg.vocab_size, (1, 8), device=DEV)
    eager = qm.generate(ids, max_new_tokens=6)[:, 8:]
    dec = GraphDecoder(qm, batch=1, max_seq=64, max_new_tokens=8)
    dec.prefill(ids)
    try:
        got = dec.decode(6)
    except Exception as e:
        pytest.skip(f"MoE graph capture unsupported here: {e}")
    assert torch.equal(got.cpu(), eager.cpu())


def test_cli_quantize_eval_gpu(tmp_path):
    """CLI end-to-end on GPU: quantize -> eval -> bench."""
    import subprocess, sys, json, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ck = str(tmp_path / "ck")
    r = subprocess.run([sys.executable, "-m", "lfmq.cli", "quantize",
                        "--model", "tiny-llama", "--scheme", "gptq",
                        "--group-size", "64", "--calib-seqs", "2",
                        "--calib-len", "32", "--out", ck],
                       capture_output=True, text=True, cwd=repo)
    assert r.returncode == 0, r.stderr[-1500:]
    r = subprocess.run([sys.executable, "-m", "lfmq.cli", "eval",
                        "--ckpt", ck, "--seq-len", "128", "--n-seqs", "1"],
                       capture_output=True, text=True, cwd=repo)
    assert r.returncode == 0, r.stderr[-1500:]
    out = json.loads(r.stdout[r.stdout.index("{"):])
    assert "delta_ppl" in out and "logit_kl" in out


def test_decode_matches_prefill_gpu_d128():
    """Same consistency check at head_dim 128 — the flash-prefill
    (csrc/attn_prefill.hip) default path vs the fused decode kernels."""
    cfg = LlamaConfig(vocab_size=1024, hidden_size=512, intermediate_size=1024,
                      n_layers=2, n_heads=4, n_kv_heads=2, head_dim=128,
                      max_seq=256)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=9), qcfg).to(DEV)
    ids = torch.randint(0, cfg.vocab_size, (1, 72), device=DEV)
    lp = qm(ids, qm.make_cache(1, 128)).float()          # flash prefill
    cache = qm.make_cache(1, 128)
    outs = [qm(ids[:, t:t + 1], cache).float() for t in range(72)]
    ld = torch.cat(outs, dim=1)                          # decode kernels
    rel = (lp - ld).abs().max() / lp.abs().max()
    assert rel < 0.05, f"decode/flash-prefill mismatch rel={float(rel)}"
    # chunked continuation (pos0 > 0) through the same kernel
    more = torch.randint(0, cfg.vocab_size, (1, 40), device=DEV)
    cont = qm(more, cache).float()
    cache2 = qm.make_cache(1, 128)
    qm(torch.cat([ids, more], dim=1)[:, :72], cache2)
    outs2 = [qm(more[:, t:t + 1], cache2).float() for t in range(40)]
    rel2 = (cont - torch.cat(outs2, dim=1)).abs().max() / cont.abs().max()
    assert rel2 < 0.05, f"continuation mismatch rel={float(rel2)}"
