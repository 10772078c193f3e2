"""HF-layout weight import (lfmq/models/import_hf.py): round-trip a
model's own weights through the HF naming/fusion conventions and
require bit-identical logits."""
import torch

import lfmq
from lfmq.models.import_hf import convert_hf_state, import_hf_state, read_hf_dir
from lfmq.models.llama import LlamaConfig
from lfmq.models.mixtral import MixtralConfig


def _to_hf_llama(state):
    """Our llama state dict -> HF-transformers naming (split fused)."""
    hf = {}
    for k, v in state.items():
        if k.endswith("qkv_proj.weight"):
            base = k[:-len(".qkv_proj.weight")]
            n = v.shape[0]
            # tiny(): hq=4, hkv=2, d=64 -> q 256, k 128, v 128
            q, kk, vv = v.split([n * 2 // 4, n // 4, n // 4], dim=0)
            hf[f"model.{base}.q_proj.weight"] = q
            hf[f"model.{base}.k_proj.weight"] = kk
            hf[f"model.{base}.v_proj.weight"] = vv
        elif k.endswith("gate_up_proj.weight"):
            base = k[:-len(".gate_up_proj.weight")]
            g, u = v.chunk(2, dim=0)
            hf[f"model.{base}.gate_proj.weight"] = g
            hf[f"model.{base}.up_proj.weight"] = u
        elif k == "lm_head.weight":
            hf[k] = v
        else:
            hf[f"model.{k}"] = v
    return hf


def test_llama_import_round_trip():
    cfg = LlamaConfig.tiny()
    src = lfmq.build_model(cfg, seed=31)
    hf = _to_hf_llama(src.state_dict())
    dst = lfmq.build_model(cfg)          # uninitialized skeleton
    import_hf_state(dst, hf)
    ids = torch.arange(10).remainder(cfg.vocab_size).reshape(1, 10)
    a = src(ids, src.make_cache(1, 16))
    b = dst(ids, dst.make_cache(1, 16))
    assert torch.equal(a, b)


def test_llama_tied_embeddings_fallback():
    cfg = LlamaConfig.tiny()
    src = lfmq.build_model(cfg, seed=33)
    hf = _to_hf_llama(src.state_dict())
    del hf["lm_head.weight"]             # tied checkpoint: no lm_head
    out = convert_hf_state(hf)
    assert torch.equal(out["lm_head.weight"], out["embed_tokens.weight"])


def test_mixtral_expert_fusion():
    cfg = MixtralConfig.tiny()
    src = lfmq.build_model(cfg, seed=37)
    hf = {}
    for k, v in src.state_dict().items():
        if ".moe.experts." in k and k.endswith("w13.weight"):
            base = k.replace(".moe.experts.", ".block_sparse_moe.experts.")
            base = base[:-len(".w13.weight")]
            w1, w3 = v.chunk(2, dim=0)
            hf[f"model.{base}.w1.weight"] = w1
            hf[f"model.{base}.w3.weight"] = w3
        elif ".moe." in k:
            hf["model." + k.replace(".moe.", ".block_sparse_moe.")] = v
        elif k.endswith("qkv_proj.weight"):
            base = k[:-len(".qkv_proj.weight")]
            n = v.shape[0]
            q, kk, vv = v.split([n * 2 // 4, n // 4, n // 4], dim=0)
            hf[f"model.{base}.q_proj.weight"] = q
            hf[f"model.{base}.k_proj.weight"] = kk
            hf[f"model.{base}.v_proj.weight"] = vv
        elif k == "lm_head.weight":
            hf[k] = v
        else:
            hf[f"model.{k}"] = v
    dst = lfmq.build_model(cfg)
    import_hf_state(dst, hf)
    ids = torch.arange(8).remainder(cfg.vocab_size).reshape(1, 8)
    assert torch.equal(src(ids, src.make_cache(1, 16)),
                       dst(ids, dst.make_cache(1, 16)))


def test_read_hf_dir_and_quantize(tmp_path):
    """Full journey: HF shard dir -> import -> quantize -> decode."""
    from safetensors.torch import save_file
    cfg = LlamaConfig.tiny()
    src = lfmq.build_model(cfg, seed=41)
    hf = _to_hf_llama(src.state_dict())
    hf = {k: v.contiguous() for k, v in hf.items()}
    keys = sorted(hf)
    half = len(keys) // 2                # two shards, as HF dirs have
    save_file({k: hf[k] for k in keys[:half]},
              str(tmp_path / "model-00001-of-00002.safetensors"))
    save_file({k: hf[k] for k in keys[half:]},
              str(tmp_path / "model-00002-of-00002.safetensors"))
    dst = lfmq.build_model(cfg)
    import_hf_state(dst, str(tmp_path))
    qm = lfmq.quantize(dst, lfmq.QuantConfig(scheme="rtn", bits=4,
                                             group_size=32))
    out = qm.generate(torch.arange(6).reshape(1, 6), max_new_tokens=4)
    assert out.shape == (1, 10)
