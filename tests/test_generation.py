"""GenerationConfig paths through the shared autoregressive generator
(lfmq/models/generation.py): greedy default equivalence, seeded
temperature sampling, eos early-stop."""
import torch

import lfmq
from lfmq.config import GenerationConfig
from lfmq.models.llama import LlamaConfig


def _model():
    cfg = LlamaConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    return lfmq.quantize(lfmq.build_model(cfg, seed=5), qcfg), cfg


def test_greedy_default_matches_explicit_cfg():
    qm, cfg = _model()
    ids = torch.arange(8).remainder(cfg.vocab_size).reshape(1, 8)
    a = qm.generate(ids, max_new_tokens=6)
    b = qm.generate(ids, gen_cfg=GenerationConfig(max_new_tokens=6))
    assert torch.equal(a, b)
    assert a.shape == (1, 14)


def test_temperature_sampling_seeded_and_in_vocab():
    qm, cfg = _model()
    ids = torch.arange(8).remainder(cfg.vocab_size).reshape(2, 4)
    gc = GenerationConfig(max_new_tokens=6, greedy=False, temperature=0.8)
    g1 = torch.Generator().manual_seed(11)
    g2 = torch.Generator().manual_seed(11)
    a = qm.generate(ids, gen_cfg=gc, generator=g1)
    b = qm.generate(ids, gen_cfg=gc, generator=g2)
    assert torch.equal(a, b)                 # same seed, same draw
    assert a.shape == (2, 10)
    new = a[:, 4:]
    assert int(new.min()) >= 0 and int(new.max()) < cfg.vocab_size


def test_eos_early_stop():
    qm, cfg = _model()
    ids = torch.arange(6).remainder(cfg.vocab_size).reshape(1, 6)
    # learn the first greedy continuation token, then declare it eos
    first = int(qm.generate(ids, max_new_tokens=1)[0, -1])
    gc = GenerationConfig(max_new_tokens=16, eos_token_id=first)
    out = qm.generate(ids, gen_cfg=gc)
    assert out.shape == (1, 7), out.shape    # stopped right after eos
    assert int(out[0, -1]) == first
