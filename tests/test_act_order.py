"""act-order GPTQ end-to-end on the CPU path (g_idx layout)."""
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
    assert hasattr(qm.layers[0].self_attn.qkv_proj, "g_idx")
    p = str(tmp_path / "ao")
    lfmq.save(qm, p, qcfg)
    m2 = lfmq.load(p)
    assert hasattr(m2.layers[0].self_attn.qkv_proj, "g_idx")
    ids = torch.randint(0, cfg.vocab_size, (1, 12))
    lf = fp(ids, fp.make_cache(1, 24)).float()
    lq = m2(ids, m2.make_cache(1, 24)).float()
    pf = torch.log_softmax(lf, -1); pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, float(kl)
