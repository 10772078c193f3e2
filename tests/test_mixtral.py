"""Mixtral MoE: CPU e2e + routing math + EP all-to-all (gloo)."""
import os

import pytest
import torch
import torch.multiprocessing as mp

import lfmq
from lfmq.models.mixtral import MixtralConfig
from lfmq.quant.calib import CalibStream


def test_mixtral_cpu_e2e_rtn(tmp_path):
    cfg = MixtralConfig.tiny()
    fp = lfmq.build_model(cfg, seed=8)
    ids = torch.randint(0, cfg.vocab_size, (1, 10))
    lf = fp(ids, fp.make_cache(1, 32)).float()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=8), qcfg)
    p = str(tmp_path / "mx")
    lfmq.save(qm, p, qcfg)
    m2 = lfmq.load(p)
    lq = m2(ids, m2.make_cache(1, 32)).float()
    pf = torch.log_softmax(lf, -1)
    pq = torch.log_softmax(lq, -1)
    kl = (pf.exp() * (pf - pq)).sum(-1).mean()
    assert kl < 0.5, float(kl)
    out = m2.generate(ids, max_new_tokens=4)
    assert out.shape == (1, 14)


def test_mixtral_gptq_quantize():
    cfg = MixtralConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="gptq", bits=4, group_size=32,
                            calib_n_seqs=2, calib_seq_len=32)
    calib = CalibStream(cfg.vocab_size, 32, 2, 7)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=4), qcfg, calib=calib)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    logits = qm(ids, qm.make_cache(1, 16))
    assert torch.isfinite(logits.float()).all()


def test_router_topk_weights():
    cfg = MixtralConfig.tiny()
    m = lfmq.build_model(cfg, seed=1)
    moe = m.layers[0].moe
    x = torch.randn(5, cfg.hidden_size).to(torch.bfloat16)
    w, ids = moe.route(x)
    assert w.shape == (5, cfg.top_k) and ids.shape == (5, cfg.top_k)
    assert torch.allclose(w.sum(-1), torch.ones(5), atol=1e-5)
    assert ids.max() < cfg.n_experts


def _ep_worker(rank, world, port, tmpdir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = MixtralConfig.tiny()
        qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=17), qcfg)
        ids = torch.arange(10).remainder(cfg.vocab_size).reshape(1, 10)
        ref_logits = qm(ids, qm.make_cache(1, 32))  # unsharded reference
        from lfmq.dist.ep import apply_expert_parallel
        apply_expert_parallel(qm, rank, world)
        ep_logits = qm(ids, qm.make_cache(1, 32))
        rel = (ep_logits.float() - ref_logits.float()).abs().max() / \
            ref_logits.float().abs().max()
        assert rel < 0.03, f"EP mismatch rel={float(rel)}"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_ep_gloo_matches_dense(tmp_path):
    mp.spawn(_ep_worker, args=(2, 29712, str(tmp_path)), nprocs=2, join=True)


def test_ep_gloo_world4(tmp_path):
    """One expert per rank at world 4 (tiny config has 4 experts) — the
    all-to-all splits are maximally uneven at this world; pre-validates
    the EP rank math beyond the usual world-2 double."""
    mp.spawn(_ep_worker, args=(4, 29761, str(tmp_path)), nprocs=4, join=True)


def _tp_worker(rank, world, port, tmpdir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = MixtralConfig.tiny()
        qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=44), qcfg)
        ids = torch.arange(10).remainder(cfg.vocab_size).reshape(1, 10)
        ref_logits = qm(ids, qm.make_cache(1, 32))
        from lfmq.dist.tp import shard_mixtral
        shard_mixtral(qm, rank, world)
        tp_logits = qm(ids, qm.make_cache(1, 32))
        rel = (tp_logits.float() - ref_logits.float()).abs().max() / \
            ref_logits.float().abs().max()
        assert rel < 0.03, f"mixtral TP rel={float(rel)}"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_mixtral_tp_fallback_gloo(tmp_path):
    """TP-only Mixtral (attention sharded, experts replicated)."""
    mp.spawn(_tp_worker, args=(2, 29733, str(tmp_path)), nprocs=2, join=True)


def test_mixtral_tp_load_raises_cleanly(tmp_path):
    cfg = MixtralConfig.tiny()
    qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
    qm = lfmq.quantize(lfmq.build_model(cfg, seed=9), qcfg)
    p = str(tmp_path / "mx")
    lfmq.save(qm, p, qcfg)
    with pytest.raises(NotImplementedError):
        lfmq.load(p, tp_rank=0, tp_world=2)


def _ep_worker8(rank, world, port, tmpdir):
    """EP at the 8-rank width the 8-GPU node would use: 8 experts, one
    per rank (the mixtral 8x7B deployment shape)."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = MixtralConfig.tiny(n_experts=8)
        qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=53), qcfg)
        ids = torch.arange(10).remainder(cfg.vocab_size).reshape(1, 10)
        ref_logits = qm(ids, qm.make_cache(1, 32))
        from lfmq.dist.ep import apply_expert_parallel
        apply_expert_parallel(qm, rank, world)
        ep_logits = qm(ids, qm.make_cache(1, 32))
        rel = (ep_logits.float() - ref_logits.float()).abs().max() / \
            ref_logits.float().abs().max()
        assert rel < 0.03, f"EP8 mismatch rel={float(rel)}"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_ep_gloo_world8_one_expert_per_rank(tmp_path):
    mp.spawn(_ep_worker8, args=(8, 29791, str(tmp_path)), nprocs=8, join=True)


def _ep_paths_worker(rank, world, port, tmpdir):
    """Same model through BOTH dispatch protocols: the static
    (capacity-padded, sync-free) decode path and the dynamic ragged path
    must both match the unsharded reference, including decode t=1."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import lfmq.dist.ep as ep
        cfg = MixtralConfig.tiny()
        qcfg = lfmq.QuantConfig(scheme="rtn", bits=4, group_size=32)
        qm = lfmq.quantize(lfmq.build_model(cfg, seed=23), qcfg)
        ids = torch.arange(9).remainder(cfg.vocab_size).reshape(1, 9)
        one = torch.tensor([[3]])
        ref_logits = qm(ids, qm.make_cache(1, 32))
        ref_one = qm(one, qm.make_cache(1, 32))
        ep.apply_expert_parallel(qm, rank, world)

        saved = ep.STATIC_SLOT_LIMIT
        try:
            ep.STATIC_SLOT_LIMIT = 10 ** 9        # force static
            st = qm(ids, qm.make_cache(1, 32))
            st1 = qm(one, qm.make_cache(1, 32))   # decode shape t=1
            ep.STATIC_SLOT_LIMIT = 0              # force dynamic
            dy = qm(ids, qm.make_cache(1, 32))
        finally:
            ep.STATIC_SLOT_LIMIT = saved
        for got, want, tag in ((st, ref_logits, "static"),
                               (st1, ref_one, "static-t1"),
                               (dy, ref_logits, "dynamic")):
            rel = (got.float() - want.float()).abs().max() / \
                want.float().abs().max()
            assert rel < 0.03, f"EP {tag} mismatch rel={float(rel)}"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_ep_gloo_static_and_dynamic_paths(tmp_path):
    mp.spawn(_ep_paths_worker, args=(2, 29815, str(tmp_path)), nprocs=2,
             join=True)
