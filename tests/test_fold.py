"""RMSNorm fold transform: exactness + flag detection (CPU)."""
import torch

import lfmq
from lfmq.models.llama import LlamaConfig
from lfmq.quant.fold import fold_rms_norms, refresh_norm_fold_flags


def test_fold_preserves_function():
    cfg = LlamaConfig.tiny()
    m1 = lfmq.build_model(cfg, seed=5)
    m2 = lfmq.build_model(cfg, seed=5)
    fold_rms_norms(m2)
    assert m2.layers[0]._in_norm_folded and m2.layers[0]._post_norm_folded
    assert (m2.layers[0].input_layernorm.weight == 1).all()
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    l1 = m1(ids, m1.make_cache(1, 16)).float()
    l2 = m2(ids, m2.make_cache(1, 16)).float()
    rel = (l1 - l2).abs().max() / l1.abs().max()
    assert rel < 0.03, float(rel)


def test_fold_flags_roundtrip(tmp_path):
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64, fold_rms=True)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=2), qcfg)
    p = str(tmp_path / "ck")
    lfmq.save(qm, p, qcfg)
    m2 = lfmq.load(p)
    assert m2.layers[0]._in_norm_folded


def test_unfolded_flags_false():
    cfg = LlamaConfig.tiny()
    m = lfmq.build_model(cfg, seed=3)
    refresh_norm_fold_flags(m)
    assert not m.layers[0]._in_norm_folded  # random weights != 1
