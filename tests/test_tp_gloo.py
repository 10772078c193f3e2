"""TP numerics on CPU with the gloo backend: shard ->
forward -> compare against the unsharded model.  The RCCL path is the
same code with the backend string swapped."""
import os

import pytest
import torch
import torch.multiprocessing as mp

import lfmq
from lfmq.models.llama import LlamaConfig


def _tiny_cfg():
    return LlamaConfig(vocab_size=256, hidden_size=128, intermediate_size=256,
                       n_layers=2, n_heads=4, n_kv_heads=2, head_dim=32,
                       max_seq=64)


def _worker(rank, world, path, port, out_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model = lfmq.load(path, tp_rank=rank, tp_world=world)
        ids = torch.arange(12).remainder(256).reshape(1, 12)
        logits = model(ids, model.make_cache(1, 32))
        out = model.generate(ids[:, :4], max_new_tokens=4)
        if rank == 0:
            torch.save({"logits": logits, "gen": out},
                       os.path.join(out_dir, "tp_out.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2])
def test_tp_gloo_matches_unsharded(tmp_path, world):
    cfg = _tiny_cfg()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=21), qcfg)
    path = str(tmp_path / "ckpt")
    lfmq.save(qm, path, qcfg)

    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref_logits = qm(ids, qm.make_cache(1, 32))
    ref_gen = qm.generate(ids[:, :4], max_new_tokens=4)

    port = 29600 + world
    mp.spawn(_worker, args=(world, path, port, str(tmp_path)), nprocs=world,
             join=True)
    out = torch.load(str(tmp_path / "tp_out.pt"), weights_only=False)
    rel = (out["logits"].float() - ref_logits.float()).abs().max() / \
        ref_logits.float().abs().max()
    assert rel < 0.05, f"TP logits rel err {float(rel)}"
    assert torch.equal(out["gen"], ref_gen)


def test_shard_llama_in_memory():
    """Single-process sanity: in-memory shard_llama halves module shapes."""
    from lfmq.dist.tp import shard_llama
    cfg = _tiny_cfg()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=3), qcfg)
    sh = shard_llama(qm, rank=1, world=2)
    a = sh.layers[0].self_attn
    assert a.qkv_proj.out_features == (4 + 2 + 2) * 32 // 2
    assert a.o_proj.in_features == 4 * 32 // 2
    assert a.n_heads == 2 and a.n_kv_heads == 1
    m = sh.layers[0].mlp
    assert m.gate_up_proj.out_features == 256
    assert m.down_proj.in_features == 128
    # world=1 forward still works (no collectives fired at world 1)
    ids = torch.arange(8).reshape(1, 8)
    logits = sh(ids, sh.make_cache(1, 16))
    assert torch.isfinite(logits.float()).all()


def test_row_shard_group_divisibility_guard():
    from lfmq.dist.tp import _shard_row
    from lfmq.models.qlinear import QLinear
    q = QLinear(96, 8, 4, 96)  # k=96, g=96: k/2=48 not divisible by 96
    with pytest.raises(AssertionError):
        _shard_row(q, 0, 2)


@pytest.mark.parametrize("world", [4])
def test_tp_gloo_world4(tmp_path, world):
    """4-way TP numerics (needs n_kv_heads % 4 == 0)."""
    cfg = LlamaConfig(vocab_size=256, hidden_size=256, intermediate_size=512,
                      n_layers=2, n_heads=8, n_kv_heads=4, head_dim=32,
                      max_seq=64)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=64)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=41), qcfg)
    path = str(tmp_path / "ckpt4")
    lfmq.save(qm, path, qcfg)
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref_logits = qm(ids, qm.make_cache(1, 32))
    port = 29650
    mp.spawn(_worker, args=(world, path, port, str(tmp_path)), nprocs=world,
             join=True)
    out = torch.load(str(tmp_path / "tp_out.pt"), weights_only=False)
    rel = (out["logits"].float() - ref_logits.float()).abs().max() / \
        ref_logits.float().abs().max()
    assert rel < 0.05, f"TP4 logits rel err {float(rel)}"


def test_tp_gloo_world8(tmp_path):
    """8-way TP numerics — the driver's largest scale point.  One q/kv
    head per rank, row-shard k slices of exactly one group (the
    tightest legal group/shard alignment)."""
    world = 8
    cfg = LlamaConfig(vocab_size=256, hidden_size=256, intermediate_size=512,
                      n_layers=2, n_heads=8, n_kv_heads=8, head_dim=32,
                      max_seq=64)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=43), qcfg)
    path = str(tmp_path / "ckpt8")
    lfmq.save(qm, path, qcfg)
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref_logits = qm(ids, qm.make_cache(1, 32))
    mp.spawn(_worker, args=(world, path, 29658, str(tmp_path)), nprocs=world,
             join=True)
    out = torch.load(str(tmp_path / "tp_out.pt"), weights_only=False)
    rel = (out["logits"].float() - ref_logits.float()).abs().max() /         ref_logits.float().abs().max()
    assert rel < 0.05, f"TP8 logits rel err {float(rel)}"


def test_tp_gloo_quantized_lm_head(tmp_path):
    """TP with a QUANTIZED lm_head (skip_modules=()): column-shard +
    all-gather over the packed head."""
    cfg = _tiny_cfg()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32,
                            skip_modules=())
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=51), qcfg)
    path = str(tmp_path / "ckpt_qh")
    lfmq.save(qm, path, qcfg)
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref_logits = qm(ids, qm.make_cache(1, 32))
    mp.spawn(_worker, args=(2, path, 29741, str(tmp_path)), nprocs=2,
             join=True)
    out = torch.load(str(tmp_path / "tp_out.pt"), weights_only=False)
    rel = (out["logits"].float() - ref_logits.float()).abs().max() / \
        ref_logits.float().abs().max()
    assert rel < 0.05, f"quantized-head TP rel err {float(rel)}"


def test_act_order_row_shard_uses_gather_plan():
    """Row-parallel act-order layers must route to the gather-column plan
    (the k-permutation crosses k-shard boundaries); a direct row shard is
    refused loudly."""
    from lfmq.dist.tp import _shard_row, _shard_row_or_gather
    from lfmq.models.qlinear import GatherColumnParallelQLinear, QLinear
    q = QLinear(64, 32, 4, 32)
    q.register_buffer("perm", torch.randperm(64))
    with pytest.raises(AssertionError):
        _shard_row(q, 0, 2)
    out = _shard_row_or_gather(q, 0, 2)
    assert isinstance(out, GatherColumnParallelQLinear)
    assert torch.equal(out.perm, q.perm)


def _worker_act_order_tp(rank, world, port, tmpdir):
    """Full act-order model TP-sharded under gloo vs the unsharded model
    (VERDICT r1 item 5 done-criterion)."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lfmq.dist.tp import shard_llama
        from lfmq.quant.calib import CalibStream
        cfg = LlamaConfig(vocab_size=256, hidden_size=256,
                          intermediate_size=512, n_layers=2, n_heads=8,
                          n_kv_heads=8, head_dim=32, max_seq=64)
        qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=32,
                                act_order=True, calib_n_seqs=2,
                                calib_seq_len=16)
        calib = CalibStream(cfg.vocab_size, 16, 2, qcfg.calib_seed)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=31), qcfg, calib=calib)
        assert hasattr(qm.layers[0].self_attn.o_proj, "perm")
        ids = torch.arange(10).remainder(256).reshape(1, 10)
        ref_logits = qm(ids, qm.make_cache(1, 32))
        shard_llama(qm, rank, world)
        from lfmq.models.qlinear import GatherColumnParallelQLinear
        assert isinstance(qm.layers[0].self_attn.o_proj,
                          GatherColumnParallelQLinear)
        tp_logits = qm(ids, qm.make_cache(1, 32))
        rel = (tp_logits.float() - ref_logits.float()).abs().max() /             ref_logits.float().abs().max()
        assert rel < 0.02, f"act-order TP rel={float(rel)}"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_act_order_tp_gloo(tmp_path):
    mp.spawn(_worker_act_order_tp, args=(2, 29871, str(tmp_path)), nprocs=2,
             join=True)


def _worker_bench_path(rank, world, port, out_dir):
    """Mirror of the bench.py N-GPU route: build+quantize in every rank
    (deterministic), shard_llama in memory, forward under gloo."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lfmq.dist.tp import shard_llama
        cfg = LlamaConfig(vocab_size=256, hidden_size=256,
                          intermediate_size=512, n_layers=2, n_heads=8,
                          n_kv_heads=8, head_dim=32, max_seq=64)
        qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=77), qcfg)
        qm = shard_llama(qm, rank, world)
        ids = torch.arange(12).remainder(256).reshape(1, 12)
        logits = qm(ids, qm.make_cache(1, 32))
        if rank == 0:
            torch.save({"logits": logits}, os.path.join(out_dir, "tp8.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_tp_gloo_world8_bench_path(tmp_path):
    """8-way TP through the exact bench.py shard route (the width the
    driver's SCALE run uses on llama3-8b: hkv 8 => kv head per rank)."""
    cfg = LlamaConfig(vocab_size=256, hidden_size=256, intermediate_size=512,
                      n_layers=2, n_heads=8, n_kv_heads=8, head_dim=32,
                      max_seq=64)
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=77), qcfg)
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref = qm(ids, qm.make_cache(1, 32))
    mp.spawn(_worker_bench_path, args=(8, 29780, str(tmp_path)), nprocs=8,
             join=True)
    out = torch.load(str(tmp_path / "tp8.pt"), weights_only=False)
    rel = (out["logits"].float() - ref.float()).abs().max() / \
        ref.float().abs().max()
    assert rel < 0.05, f"TP8 logits rel err {float(rel)}"


def test_collective_call_sites_no_spawn(monkeypatch):
    """SURVEY.md §4.3 asserting-stub check: a rank-0 TP shard must fire
    exactly one all-reduce per row-parallel projection (o_proj,
    down_proj: 2/layer) and one lm_head all-gather per forward — counted
    with stubbed collectives, no process group."""
    from lfmq.dist.tp import shard_llama
    from lfmq.models import qlinear as ql
    cfg = _tiny_cfg()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=61), qcfg)
    qm = shard_llama(qm, rank=0, world=2)

    calls = {"all_reduce": 0, "all_gather": 0}

    def fake_reduce(t):
        calls["all_reduce"] += 1
        return t  # rank-0 partial; numerics not asserted here

    def fake_gather(t, dim=-1):
        calls["all_gather"] += 1
        return torch.cat([t, torch.zeros_like(t)], dim=dim)

    monkeypatch.setattr(ql.lfmq_dist, "all_reduce_sum", fake_reduce)
    monkeypatch.setattr(ql.lfmq_dist, "all_gather_cat", fake_gather)
    monkeypatch.setattr(ql.lfmq_dist, "get_world_size", lambda: 2)

    ids = torch.arange(6).remainder(256).reshape(1, 6)
    qm(ids, qm.make_cache(1, 16))
    assert calls["all_reduce"] == 2 * cfg.n_layers, calls
    assert calls["all_gather"] == 1, calls


def test_tied_embeddings_save_load_roundtrip(tmp_path):
    """Tied lm_head aliases embed_tokens; safetensors refuses shared
    storage, so save() drops the alias and load() re-ties.  Logits must
    round-trip bit-exact and the loaded module must be truly tied."""
    cfg = _tiny_cfg()
    cfg.tie_embeddings = True
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=9), qcfg)
    assert qm.lm_head.weight is qm.embed_tokens.weight
    path = str(tmp_path / "tied")
    lfmq.save(qm, path, qcfg)
    m2 = lfmq.load(path)
    assert m2.lm_head.weight is m2.embed_tokens.weight
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    a = qm(ids, qm.make_cache(1, 32))
    b = m2(ids, m2.make_cache(1, 32))
    assert torch.equal(a, b)


def test_tied_embeddings_tp_gloo(tmp_path):
    """TP load of a tied checkpoint: the single stored copy must be
    row-sliced into each rank's vocab shard of the gathered head."""
    cfg = _tiny_cfg()
    cfg.tie_embeddings = True
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=9), qcfg)
    path = str(tmp_path / "tied")
    lfmq.save(qm, path, qcfg)
    ids = torch.arange(12).remainder(256).reshape(1, 12)
    ref_logits = qm(ids, qm.make_cache(1, 32))
    ref_gen = qm.generate(ids[:, :4], max_new_tokens=4)
    mp.spawn(_worker, args=(2, path, 29671, str(tmp_path)), nprocs=2,
             join=True)
    out = torch.load(str(tmp_path / "tp_out.pt"), weights_only=False)
    rel = (out["logits"].float() - ref_logits.float()).abs().max() / \
        ref_logits.float().abs().max()
    assert rel < 0.05, f"tied TP logits rel err {float(rel)}"
    assert torch.equal(out["gen"], ref_gen)


def test_shard_llama_8b_widths_in_memory():
    """Shard the REAL llama3-8b layer widths (hidden 4096, inter 14336,
    32q/8kv heads, vocab 128256, g=128, W8 head — exactly what the
    driver's scale run builds) at world 8, the largest scale point
    (single layer so it stays CI-light).  Catches 8B-specific
    divisibility/shape bugs without needing 8 processes."""
    from lfmq.bench.decode import build_quantized_llama
    from lfmq.dist.tp import shard_llama
    world = 8
    qm, cfg, _ = build_quantized_llama(
        "llama3-8b", "rtn", 4, 128, "cpu", seed=0, n_layers=1,
        quant_lm_head="w8")
    rank = world - 1  # the last rank has the edge-case offsets
    shard_llama(qm, rank, world)
    at = qm.layers[0].self_attn
    assert at.qkv_proj.out_features == (32 // world + 2 * (8 // world)) * 128
    assert at.o_proj.in_features == 4096 // world
    mlp = qm.layers[0].mlp
    assert mlp.gate_up_proj.out_features == 2 * 14336 // world
    assert mlp.down_proj.in_features == 14336 // world
    assert mlp.down_proj.group_size == 128  # group survives the k-shard
    assert qm.lm_head.out_features == 128256 // world
    # forward shapes hold at world 1 semantics (no pg: collectives no-op)
    import lfmq.dist as d
    assert d.get_world_size() == 1
