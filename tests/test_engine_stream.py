"""Persistent-engine weight-stream builder: layout invariants on CPU.

The kernel (csrc/engine.hip) walks records with the documented rule; this
re-derives the walk independently from the spec in lfmq/models/engine.py's
docstring and checks the built bytes.  GPU execution parity lives in
tests/test_gpu_engine.py.
"""
import os

import torch

import lfmq

os.environ["LFMQ_ENGINE"] = "1"  # the engine is opt-in; tests opt in
from lfmq.models.llama import LlamaConfig
from lfmq.models.engine import (DecodeEngine, engine_supported, _rec_bytes,
                                SLOT, NCU)


def _engine_model():
    # smallest config the engine supports: d=128, k dims % 2048 == 0
    cfg = LlamaConfig(vocab_size=1024, hidden_size=2048,
                      intermediate_size=2048, n_layers=2, n_heads=16,
                      n_kv_heads=4, head_dim=128, max_seq=512)
    m = lfmq.build_model(cfg, seed=3)
    q = lfmq.quantize(m, lfmq.QuantConfig(scheme="rtn", bits=4, group_size=128))
    return q


def test_engine_supported_gates():
    q = _engine_model()
    assert engine_supported(q, batch=1)
    assert not engine_supported(q, batch=2)
    # unsupported: group too small for the lane-chunk layout
    cfg = LlamaConfig(vocab_size=256, hidden_size=2048,
                      intermediate_size=2048, n_layers=1, n_heads=16,
                      n_kv_heads=4, head_dim=128, max_seq=256)
    q2 = lfmq.quantize(lfmq.build_model(cfg, seed=1),
                       lfmq.QuantConfig(scheme="rtn", bits=4, group_size=16))
    assert not engine_supported(q2, batch=1)
    # head_dim != 128 unsupported
    cfg3 = LlamaConfig.tiny()
    q3 = lfmq.quantize(lfmq.build_model(cfg3, seed=1),
                       lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32))
    assert not engine_supported(q3, batch=1)


def test_stream_layout_roundtrip():
    q = _engine_model()
    cache = q.make_cache(1, 256)
    eng = DecodeEngine(q, cache, 256)
    cfg = q.cfg
    g = eng.g
    ring = eng.nslot * SLOT
    stream = eng.stream.reshape(NCU, eng.per_cu)

    assert eng.per_cu % SLOT == 0

    # independent offset walk per the documented spec
    def walk():
        cur = 0
        for layer in q.layers:
            for op, inter in ((layer.self_attn.qkv_proj, False),
                              (layer.self_attn.o_proj, False),
                              (layer.mlp.gate_up_proj, True),
                              (layer.mlp.down_proj, False)):
                n, k = op.out_features, op.in_features
                G = k // g
                rb = _rec_bytes(k, G)
                npc = n // NCU
                for r in range(npc):
                    if cur % ring + rb > ring:
                        cur += ring - cur % ring
                    yield op, inter, r, cur, rb, k, G
                    cur += rb

    for op, inter_leaved, r, off, rb, k, G in walk():
        # never straddles the ring wrap
        assert off % ring + rb <= ring
        assert off % 16 == 0
        iters = k // 512
        # spot-check CU 0 and CU 17
        for cu in (0, 17):
            if inter_leaved:
                m = op.out_features // 2 // NCU
                row = (cu * m + r // 2) + (op.out_features // 2) * (r % 2)
            else:
                row = cu * (op.out_features // NCU) + r
            got = stream[cu, off:off + rb]
            # v2 nibbles: lane-major dwords = transposed [iters, 64] grid
            qw_t = op.qweight[row].reshape(iters, 64).t().contiguous()
            assert torch.equal(got[:k // 2],
                               qw_t.view(torch.uint8).reshape(k // 2))
            # v2 fp32 (s, s*(z+128)) pairs
            s = op.scales[row].float()
            gidx = torch.arange(G)
            z = (op.qzeros[row][gidx // 8] >> (4 * (gidx % 8))) & 0xF
            pairs = torch.stack([s, s * (z.float() + 128.0)], -1).reshape(-1)
            assert torch.equal(
                got[k // 2:k // 2 + 8 * G].view(torch.float32), pairs)


def test_kvcache_stacked_views():
    q = _engine_model()
    cache = q.make_cache(1, 64)
    # per-layer views alias the stacked tensor (static addresses for graphs
    # and a single base pointer + stride for the engine)
    cache.k[1][0, 0, 3, 5] = 7.0
    assert float(cache.k_all[1, 0, 0, 3, 5]) == 7.0
    assert cache.k[0].is_contiguous()


def test_stream_ring_wrap_padding():
    """A stream larger than the LDS ring must pad records at ring-wrap
    boundaries (the kernel mirrors this rule; a straddling record would
    read across the LDS ring seam)."""
    cfg = LlamaConfig(vocab_size=512, hidden_size=2048,
                      intermediate_size=2048, n_layers=8, n_heads=16,
                      n_kv_heads=4, head_dim=128, max_seq=256)
    q = lfmq.quantize(lfmq.build_model(cfg, seed=4),
                      lfmq.QuantConfig(scheme="rtn", bits=4, group_size=128))
    cache = q.make_cache(1, 128)
    eng = DecodeEngine(q, cache, 128)
    ring = eng.nslot * SLOT
    assert eng.per_cu > ring, "test needs a stream larger than the ring"
    g = eng.g
    cur = 0
    pads = 0
    for layer in q.layers:
        for op in (layer.self_attn.qkv_proj, layer.self_attn.o_proj,
                   layer.mlp.gate_up_proj, layer.mlp.down_proj):
            rb = _rec_bytes(op.in_features, op.in_features // g)
            for _ in range(op.out_features // NCU):
                if cur % ring + rb > ring:
                    pads += 1
                    cur += ring - cur % ring
                assert cur % ring + rb <= ring
                cur += rb
    assert pads > 0, "no wrap pad exercised — grow the model"
    assert eng.per_cu == (cur + SLOT - 1) // SLOT * SLOT


def test_engine_env_gate_not_sticky():
    """LFMQ_ENGINE=0 must not latch _engine_unsupported: the env opt-in
    is re-read every step (toggling within one process is how the parity
    tools compare engine vs launched paths)."""
    q = _engine_model()
    cache = q.make_cache(1, 64)
    old = os.environ.get("LFMQ_ENGINE")
    try:
        os.environ["LFMQ_ENGINE"] = "0"
        assert q._engine_for(cache, 1) is False
        assert not getattr(q, "_engine_unsupported", False)
        assert getattr(q, "_engine", None) is None
    finally:
        os.environ["LFMQ_ENGINE"] = old if old is not None else "1"


def test_stream_pair_divisibility():
    """The record phases consume PAIRS (quads for gate_up) — the
    engine_supported shape gates must imply npc_qkv/npc_o/npc_dn even
    and npc_gu % 4 == 0 for every supported config."""
    q = _engine_model()
    cfg = q.cfg
    nqkv = (cfg.n_heads + 2 * cfg.n_kv_heads) * cfg.head_dim
    assert (nqkv // NCU) % 2 == 0
    assert (cfg.hidden_size // NCU) % 2 == 0          # o / down outputs
    assert (2 * cfg.intermediate_size // NCU) % 4 == 0  # gate_up quads
