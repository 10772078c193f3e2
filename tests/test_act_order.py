"""act-order GPTQ end-to-end (group-contiguous + perm layout v2)."""
import pytest
import torch

import lfmq
from lfmq.models.llama import LlamaConfig
from lfmq.quant.calib import CalibStream


def test_act_order_e2e_cpu(tmp_path):
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            act_order=True, calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    fp = lfmq.build_model(cfg, seed=12)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=12), qcfg, calib=calib)
    assert hasattr(qm.layers[0].self_attn.qkv_proj, "perm")
    p = str(tmp_path / "ao")
    lfmq.save(qm, p, qcfg)
    m2 = lfmq.load(p)
    assert hasattr(m2.layers[0].self_attn.qkv_proj, "perm")
    ids = torch.randint(0, cfg.vocab_size, (1, 12))
    lf = fp(ids, fp.make_cache(1, 24)).float()
    lq = m2(ids, m2.make_cache(1, 24)).float()
    pf = torch.log_softmax(lf, -1); pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, float(kl)


@pytest.mark.gpu
def test_act_order_gpu_decode_parity(tmp_path):
    """GPU decode of an act-order checkpoint (the fused
    kernels consume the group-contiguous layout; the input is gathered by
    perm).  CPU forward of the same checkpoint is the oracle."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            act_order=True, calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=12), qcfg, calib=calib)
    p = str(tmp_path / "ao")
    lfmq.save(qm, p, qcfg)
    cpu = lfmq.load(p)
    gpu = lfmq.load(p, device="cuda")
    assert hasattr(gpu.layers[0].self_attn.qkv_proj, "perm")
    ids = torch.randint(0, cfg.vocab_size, (1, 9))
    ref = cpu(ids, cpu.make_cache(1, 40)).float()
    # prefill + a few decode steps on GPU
    cache = gpu.make_cache(1, 40, device="cuda")
    got = gpu(ids.cuda(), cache).float().cpu()
    rel = (got - ref).abs().max() / ref.abs().max()
    assert float(rel) < 2e-2, f"act-order GPU prefill rel={float(rel)}"
    tok = ids[:, -1:].cuda()
    cpu_cache = cpu.make_cache(1, 40)
    cpu(ids, cpu_cache)
    for _ in range(4):
        g = gpu(tok.cuda(), cache).float().cpu()
        c = cpu(tok.cpu(), cpu_cache).float()
        rel = (g - c).abs().max() / c.abs().max()
        assert float(rel) < 2e-2, f"act-order GPU decode rel={float(rel)}"
        tok = c[:, -1:].argmax(dim=-1)


def test_act_order_lazy_tp_load_raises(tmp_path):
    """Lazy TP load slices row-parallel k — invalid under a k-permutation:
    must refuse loudly and point to the in-memory shard path."""
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            act_order=True, calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=12), qcfg, calib=calib)
    p = str(tmp_path / "ao")
    lfmq.save(qm, p, qcfg)
    with pytest.raises(NotImplementedError, match="act-order"):
        lfmq.load(p, tp_rank=0, tp_world=2)


def test_perm_fused_primitives_cpu(monkeypatch):
    """forward_resid / forward_rms on an act-order QLinear must apply the
    perm gather before the fused GEMV (rmsnorm commutes with the
    permutation, so gather-first is exact).  The fused epilogues are
    GPU-only, so pin the contract with a capture stub: the x that reaches
    dq_linear must be x[perm]."""
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=64,
                            act_order=True, calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=12), qcfg, calib=calib)
    ql = qm.layers[0].self_attn.qkv_proj
    assert hasattr(ql, "perm")
    torch.manual_seed(5)
    x = (torch.randn(1, ql.in_features) * 0.1).to(torch.bfloat16)
    want = x.index_select(-1, ql.perm)

    seen = {}

    def stub(xx, qw, sc, zz, bits, g, bias, rms_eps=-1.0, resid=None):
        seen["x"] = xx
        return torch.zeros(xx.shape[0], ql.out_features,
                           dtype=xx.dtype)

    import lfmq.models.qlinear as qlin_mod
    monkeypatch.setattr(qlin_mod.F, "dq_linear", stub)

    ql.forward_resid(x, torch.zeros(1, ql.out_features).to(torch.bfloat16))
    assert torch.equal(seen["x"], want), "forward_resid skipped the gather"
    ql.forward_rms(x, 1e-5)
    assert torch.equal(seen["x"], want), "forward_rms skipped the gather"
    ql.forward(x)
    assert torch.equal(seen["x"], want)
