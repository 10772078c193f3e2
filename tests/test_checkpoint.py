"""Packed-checkpoint format: save/load roundtrip + schema stability."""
import json
import os

import pytest
import torch

import lfmq
from lfmq import QuantConfig, build_model, load, quantize, save
from lfmq.checkpoint import CONFIG_FILE, MODEL_FILE, load_meta, load_packed
from lfmq.models.llama import LlamaConfig
from lfmq.models.opt import OPTConfig


def _tiny_quantized(scheme="rtn"):
    cfg = LlamaConfig.tiny()
    model = build_model(cfg, seed=7)
    qcfg = QuantConfig(scheme=scheme, bits=4, group_size=64,
                       calib_n_seqs=2, calib_seq_len=32)
    return quantize(model, qcfg), qcfg, cfg


def test_save_load_roundtrip(tmp_path):
    model, qcfg, cfg = _tiny_quantized()
    p = str(tmp_path / "ckpt")
    save(model, p, qcfg)
    assert os.path.exists(os.path.join(p, MODEL_FILE))
    assert os.path.exists(os.path.join(p, CONFIG_FILE))
    m2 = load(p)
    sd1, sd2 = model.state_dict(), m2.state_dict()
    assert set(sd1) == set(sd2)
    for k in sd1:
        assert torch.equal(sd1[k].cpu(), sd2[k].cpu()), k
    # decode equivalence
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    out1 = model.generate(ids, max_new_tokens=4)
    out2 = m2.generate(ids, max_new_tokens=4)
    assert torch.equal(out1, out2)


def test_checkpoint_schema_golden(tmp_path):
    """Format stability: tensor names/dtypes/shapes for a known config."""
    model, qcfg, cfg = _tiny_quantized()
    p = str(tmp_path / "ckpt")
    save(model, p, qcfg)
    meta = load_meta(p)
    assert meta["format_version"] == lfmq.FORMAT_VERSION
    assert meta["quant"]["scheme"] == "rtn"
    state = load_packed(p)
    # one quantized layer spot-check: qkv_proj of layer 0
    h, hq, hkv, d = cfg.hidden_size, cfg.n_heads, cfg.n_kv_heads, cfg.head_dim
    qkv_out = (hq + 2 * hkv) * d
    t = state["layers.0.self_attn.qkv_proj.qweight"]
    assert t.dtype == torch.int32 and t.shape == (qkv_out, h // 8)
    s = state["layers.0.self_attn.qkv_proj.scales"]
    assert s.dtype == torch.bfloat16 and s.shape == (qkv_out, h // 64)
    z = state["layers.0.self_attn.qkv_proj.qzeros"]
    assert z.dtype == torch.int32 and z.shape == (qkv_out, (h // 64 + 7) // 8)
    # norms/embeddings stay fp
    assert state["norm.weight"].dtype == torch.bfloat16
    assert "lm_head.weight" in state  # skipped module stays dense


def test_load_rejects_version_mismatch(tmp_path):
    model, qcfg, _ = _tiny_quantized()
    p = str(tmp_path / "ckpt")
    save(model, p, qcfg)
    cfgf = os.path.join(p, CONFIG_FILE)
    with open(cfgf) as f:
        meta = json.load(f)
    meta["format_version"] = 999
    with open(cfgf, "w") as f:
        json.dump(meta, f)
    with pytest.raises(ValueError):
        load(p)


def test_sharded_read(tmp_path):
    """load_packed shard_fn slices read only the requested rows."""
    model, qcfg, _ = _tiny_quantized()
    p = str(tmp_path / "ckpt")
    save(model, p, qcfg)
    full = load_packed(p)

    def shard(name, shape):
        if name.endswith("qweight"):
            return (slice(0, shape[0] // 2),)
        return None
    half = load_packed(p, shard_fn=shard)
    for k in full:
        if k.endswith("qweight"):
            assert torch.equal(half[k], full[k][: full[k].shape[0] // 2])
        else:
            assert torch.equal(half[k], full[k])


def test_atomic_write_no_partial(tmp_path):
    model, qcfg, _ = _tiny_quantized()
    p = str(tmp_path / "ckpt")
    save(model, p, qcfg)
    save(model, p, qcfg)  # overwrite in place must be safe
    assert not [f for f in os.listdir(p) if f.endswith(".tmp")]
