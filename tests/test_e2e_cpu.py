"""End-to-end CPU configs.

Config #1 (BASELINE.json): OPT-125m-shaped per-tensor int8 RTN quantize +
CPU matmul decode — quantize -> save -> load -> greedy decode equivalence
vs fp32 within tolerance.  Plus tiny-Llama GPTQ/AWQ end-to-end fidelity.
"""
import torch

from lfmq import QuantConfig, build_model, load, quantize, save
from lfmq.models.llama import LlamaConfig
from lfmq.models.opt import OPTConfig
from lfmq.quant.calib import CalibStream


def test_config1_opt_int8_cpu_decode(tmp_path):
    cfg = OPTConfig.tiny()
    fp = build_model(cfg, seed=11)
    ids = torch.randint(0, cfg.vocab_size, (2, 12))
    ref_out = fp.generate(ids, max_new_tokens=8)
    ref_logits = fp(ids, fp.make_cache(2, 64))

    qcfg = QuantConfig(scheme="rtn", bits=8, per_tensor=True, group_size=0)
    qm = quantize(build_model(cfg, seed=11), qcfg)
    p = str(tmp_path / "opt")
    save(qm, p, qcfg)
    qm2 = load(p)

    q_logits = qm2(ids, qm2.make_cache(2, 64))
    # logits close in relative terms
    rel = (q_logits - ref_logits).abs().mean() / ref_logits.abs().mean()
    assert rel < 0.05
    q_out = qm2.generate(ids, max_new_tokens=8)
    agree = (q_out == ref_out).float().mean()
    assert agree > 0.85  # greedy tokens nearly identical


def test_tiny_llama_gptq_e2e(tmp_path):
    cfg = LlamaConfig.tiny()
    fp = build_model(cfg, seed=3)
    qcfg = QuantConfig(scheme="gptq", bits=4, group_size=64,
                       calib_n_seqs=4, calib_seq_len=64)
    calib = CalibStream(cfg.vocab_size, 64, 4, qcfg.calib_seed)
    qm = quantize(build_model(cfg, seed=3), qcfg, calib=calib)
    p = str(tmp_path / "llama_gptq")
    save(qm, p, qcfg)
    qm2 = load(p)

    ids = torch.randint(0, cfg.vocab_size, (1, 16))
    lf = fp(ids, fp.make_cache(1, 32)).float()
    lq = qm2(ids, qm2.make_cache(1, 32)).float()
    # logit KL under threshold (random-init tiny model: loose bound)
    pf = torch.log_softmax(lf, -1)
    pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, f"KL={kl}"


def test_tiny_llama_awq_e2e(tmp_path):
    cfg = LlamaConfig.tiny()
    fp = build_model(cfg, seed=5)
    qcfg = QuantConfig(scheme="awq", bits=4, group_size=64,
                       calib_n_seqs=2, calib_seq_len=64, awq_grid=8)
    calib = CalibStream(cfg.vocab_size, 64, 2, qcfg.calib_seed)
    qm = quantize(build_model(cfg, seed=5), qcfg, calib=calib)
    p = str(tmp_path / "llama_awq")
    save(qm, p, qcfg)
    qm2 = load(p)
    ids = torch.randint(0, cfg.vocab_size, (1, 16))
    lf = fp(ids, fp.make_cache(1, 32)).float()
    lq = qm2(ids, qm2.make_cache(1, 32)).float()
    pf = torch.log_softmax(lf, -1)
    pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, f"KL={kl}"


def test_quantized_decode_prefill_consistency():
    """Decode one-token-at-a-time must equal prefill logits (cache math)."""
    cfg = LlamaConfig.tiny()
    qcfg = QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = quantize(build_model(cfg, seed=9), qcfg)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    lp = qm(ids, qm.make_cache(1, 32)).float()
    cache = qm.make_cache(1, 32)
    outs = []
    for t in range(8):
        outs.append(qm(ids[:, t:t + 1], cache).float())
    ld = torch.cat(outs, dim=1)
    assert torch.allclose(lp, ld, atol=2e-2, rtol=1e-2)


def test_opt_awq_e2e(tmp_path):
    """AWQ on the OPT arch (LayerNorm/v_proj/fc1 scale plan)."""
    from lfmq.quant.calib import CalibStream
    cfg = OPTConfig.tiny()
    fp = build_model(cfg, seed=14)
    qcfg = QuantConfig(scheme="awq", bits=4, group_size=32, awq_grid=6,
                       calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, qcfg.calib_seed)
    qm = quantize(build_model(cfg, seed=14), qcfg, calib=calib)
    ids = torch.randint(0, cfg.vocab_size, (1, 10))
    lf = fp(ids, fp.make_cache(1, 24)).float()
    lq = qm(ids, qm.make_cache(1, 24)).float()
    pf = torch.log_softmax(lf, -1); pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, float(kl)


def test_sym_quant_e2e_round_trip(tmp_path):
    """sym=True end-to-end: quantize -> save -> load -> decode; the
    packed layout stores the constant sym zero like any other zero."""
    import torch

    import lfmq
    from lfmq.models.llama import LlamaConfig
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32, sym=True)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=47), qcfg)
    p = str(tmp_path / "sym")
    lfmq.save(qm, p, qcfg)
    back = lfmq.load(p)
    ids = torch.arange(8).remainder(cfg.vocab_size).reshape(1, 8)
    a = qm(ids, qm.make_cache(1, 16))
    b = back(ids, back.make_cache(1, 16))
    assert torch.equal(a, b)
    out = back.generate(ids[:, :4], max_new_tokens=4)
    assert out.shape == (1, 8)


def test_per_channel_quant_e2e(tmp_path):
    """group_size=-1 (per-channel: one group = the whole row) through
    quantize -> save -> load -> forward."""
    import torch

    import lfmq
    from lfmq.models.llama import LlamaConfig
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=-1)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=49), qcfg)
    p = str(tmp_path / "pc")
    lfmq.save(qm, p, qcfg)
    back = lfmq.load(p)
    ids = torch.arange(8).remainder(cfg.vocab_size).reshape(1, 8)
    assert torch.equal(qm(ids, qm.make_cache(1, 16)),
                       back(ids, back.make_cache(1, 16)))
