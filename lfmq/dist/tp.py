"""Tensor-parallel sharding of packed quantized Llama models.

Plan: column-parallel qkv_proj / gate_up_proj / lm_head,
row-parallel o_proj / down_proj with an all-reduce on RCCL over xGMI.
Packed-tensor slicing rules (layout: lfmq/ops/ref.py):
  - column (n) shard: slice rows of qweight/scales/qzeros.
  - row (k) shard: slice qweight words (k/8 per word-group), scales and
    zeros by group — requires k_shard % group_size == 0 (asserted; 70B
    down_proj: 28672/8 = 3584 = 28 groups of 128, exactly).
Fused projections shard per logical part: qkv = [q|k|v] slices one
head-range of each part; gate_up = [gate|up] one inter-range of each.

Every rank can also shard lazily at load time (load_tp): the shard_fn
hands safetensors ONLY the needed row/word slices — no full read, no
broadcast.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.nn as nn

from ..models.qlinear import (ColumnParallelQLinear,
                              GatherColumnParallelQLinear, QLinear,
                              RowParallelQLinear)
from ..quant.common import get_module, replace_module


def _row_slices(parts: List[Tuple[int, int]], rank: int, world: int) -> List[slice]:
    """parts: [(offset, size)] of a fused output dim -> this rank's slices."""
    out = []
    for off, size in parts:
        assert size % world == 0, f"fused part {size} % tp {world} != 0"
        sh = size // world
        out.append(slice(off + rank * sh, off + (rank + 1) * sh))
    return out


def _cat_rows(t: torch.Tensor, slices: List[slice]) -> torch.Tensor:
    return torch.cat([t[s] for s in slices], dim=0)


def _shard_col(q: QLinear, parts: List[Tuple[int, int]], rank: int, world: int,
               gather_output: bool = False) -> ColumnParallelQLinear:
    sl = _row_slices(parts, rank, world)
    out = ColumnParallelQLinear(q.in_features, q.out_features // world, q.bits,
                                q.group_size, bias=q.bias is not None,
                                gather_output=gather_output)
    out.qweight.copy_(_cat_rows(q.qweight, sl))
    out.scales.copy_(_cat_rows(q.scales, sl))
    out.qzeros.copy_(_cat_rows(q.qzeros, sl))
    if q.bias is not None:
        out.bias.copy_(_cat_rows(q.bias, sl))
    # act-order: the perm gathers on the (un-sharded) INPUT dim — replicate
    if hasattr(q, "perm"):
        out.register_buffer("perm", q.perm.clone())
    return out


def _shard_gather_col(q: QLinear, rank: int,
                      world: int) -> GatherColumnParallelQLinear:
    """Column-shard an act-order layer that would normally be row-parallel
    (its input arrives k-sharded and is all-gathered in forward)."""
    sl = _row_slices([(0, q.out_features)], rank, world)
    out = GatherColumnParallelQLinear(q.in_features, q.out_features // world,
                                      q.bits, q.group_size,
                                      bias=q.bias is not None)
    out.qweight.copy_(_cat_rows(q.qweight, sl))
    out.scales.copy_(_cat_rows(q.scales, sl))
    out.qzeros.copy_(_cat_rows(q.qzeros, sl))
    if q.bias is not None:
        out.bias.copy_(_cat_rows(q.bias, sl))
    out.register_buffer("perm", q.perm.clone())
    return out


def _shard_row_or_gather(q: QLinear, rank: int, world: int):
    return _shard_gather_col(q, rank, world) if hasattr(q, "perm")         else _shard_row(q, rank, world)


def _shard_row(q: QLinear, rank: int, world: int) -> RowParallelQLinear:
    assert not hasattr(q, "perm"), (
        "act-order row-parallel layers use the gather-column plan "
        "(_shard_row_act_order), not a k-shard: the permutation crosses "
        "k-shard boundaries")
    k = q.in_features
    assert k % world == 0, f"in_features {k} % tp {world} != 0"
    ks = k // world
    g = q.group_size
    assert ks % g == 0, (
        f"k shard {ks} not a multiple of group_size {g} — pick tp so that "
        f"(k/tp) % g == 0")
    per = 32 // q.bits
    wslice = slice(rank * ks // per, (rank + 1) * ks // per)
    Gs = ks // g
    gslice = slice(rank * Gs, (rank + 1) * Gs)
    out = RowParallelQLinear(ks, q.out_features, q.bits, g,
                             bias=q.bias is not None)
    out.qweight.copy_(q.qweight[:, wslice])
    out.scales.copy_(q.scales[:, gslice])
    # zeros are packed along G: need Gs % per == 0 OR repack
    if (rank * Gs) % per == 0 and Gs % per == 0:
        out.qzeros.copy_(q.qzeros[:, rank * Gs // per:(rank + 1) * Gs // per])
    else:  # repack the group range
        from ..ops import ref
        G = q.in_features // g
        z = ref.unpack_zeros(q.qzeros.cpu(), q.bits, G)[:, gslice]
        out.qzeros.copy_(ref.pack_zeros(z, q.bits).to(q.qzeros.device))
    if q.bias is not None:
        # bias must be added once, after the reduce: keep it on rank 0 only
        if rank == 0:
            out.bias.copy_(q.bias)
        else:
            out.bias.zero_()
    return out


@torch.no_grad()
def shard_llama(model: nn.Module, rank: int, world: int) -> nn.Module:
    """Shard an (already quantized) LlamaForCausalLM in place."""
    if world == 1:
        return model
    cfg = model.cfg
    d = cfg.head_dim
    assert cfg.n_heads % world == 0, "n_heads % tp != 0"
    assert cfg.n_kv_heads % world == 0, (
        f"n_kv_heads {cfg.n_kv_heads} % tp {world} != 0 (KV-head replication "
        "not implemented; pick tp <= n_kv_heads)")
    hq, hkv = cfg.n_heads, cfg.n_kv_heads
    inter = cfg.intermediate_size

    for li, layer in enumerate(model.layers):
        attn = layer.self_attn
        qkv_parts = [(0, hq * d), (hq * d, hkv * d), ((hq + hkv) * d, hkv * d)]
        replace_module(attn, "qkv_proj",
                       _shard_col(attn.qkv_proj, qkv_parts, rank, world))
        replace_module(attn, "o_proj",
                       _shard_row_or_gather(attn.o_proj, rank, world))
        attn.n_heads = hq // world
        attn.n_kv_heads = hkv // world
        mlp = layer.mlp
        gu_parts = [(0, inter), (inter, inter)]
        replace_module(mlp, "gate_up_proj",
                       _shard_col(mlp.gate_up_proj, gu_parts, rank, world))
        replace_module(mlp, "down_proj",
                       _shard_row_or_gather(mlp.down_proj, rank, world))
        mlp.inter = inter // world

    # lm_head: vocab column-shard + all-gather (dense bf16 or quantized)
    lm = model.lm_head
    if isinstance(lm, QLinear):
        replace_module(model, "lm_head",
                       _shard_col(lm, [(0, lm.out_features)], rank, world,
                                  gather_output=True))
    else:
        assert lm.out_features % world == 0
        vs = lm.out_features // world
        new = nn.Linear(lm.in_features, vs, bias=False, dtype=lm.weight.dtype)
        new.weight.data.copy_(lm.weight.data[rank * vs:(rank + 1) * vs])
        new = _GatheredLinear(new)
        replace_module(model, "lm_head", new)
    return model


@torch.no_grad()
def shard_mixtral(model: nn.Module, rank: int, world: int) -> nn.Module:
    """TP-only Mixtral fallback: attention column/row
    sharded exactly like Llama, experts REPLICATED (the mandated
    expert-parallel path is lfmq.dist.ep; this keeps memory higher but
    needs no all-to-all).  The row-parallel o_proj all-reduce makes the
    hidden stream identical on every rank, so the replicated MoE stays
    consistent."""
    if world == 1:
        return model
    cfg = model.cfg
    d, hq, hkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
    assert hq % world == 0 and hkv % world == 0
    for layer in model.layers:
        attn = layer.self_attn
        qkv_parts = [(0, hq * d), (hq * d, hkv * d), ((hq + hkv) * d, hkv * d)]
        replace_module(attn, "qkv_proj",
                       _shard_col(attn.qkv_proj, qkv_parts, rank, world))
        replace_module(attn, "o_proj",
                       _shard_row_or_gather(attn.o_proj, rank, world))
        attn.n_heads = hq // world
        attn.n_kv_heads = hkv // world
    lm = model.lm_head
    if not isinstance(lm, QLinear):
        assert lm.out_features % world == 0
        vs = lm.out_features // world
        new = nn.Linear(lm.in_features, vs, bias=False, dtype=lm.weight.dtype)
        new.weight.data.copy_(lm.weight.data[rank * vs:(rank + 1) * vs])
        replace_module(model, "lm_head", _GatheredLinear(new))
    return model


class _GatheredLinear(nn.Module):
    """Dense column-parallel linear + all-gather (the bf16 lm_head)."""

    def __init__(self, local: nn.Linear):
        super().__init__()
        self.local = local

    def forward(self, x):
        from . import all_gather_cat
        return all_gather_cat(self.local(x), dim=-1)


@torch.no_grad()
def load_tp(path: str, meta: dict, device: str, rank: int, world: int,
            dtype=None) -> nn.Module:
    """Build this rank's sharded model, reading only its slices from the
    packed checkpoint (no broadcast, lazy slices)."""
    from ..api import _build_quantized_skeleton
    from ..checkpoint import load_packed

    model = _build_quantized_skeleton(meta, dtype=dtype)
    cfg = model.cfg
    if getattr(cfg, "arch", "llama") != "llama":
        raise NotImplementedError(
            f"TP load for arch {cfg.arch!r} is not implemented: use "
            "lfmq.load(..., tp_world=1) + lfmq.dist.ep.apply_expert_parallel "
            "(Mixtral) or lfmq.dist.tp.shard_mixtral on the loaded model")
    if any(spec.get("perm") or spec.get("g_idx")
           for spec in meta.get("manifest", {}).values()
           if isinstance(spec, dict)):
        raise NotImplementedError(
            "lazy TP load of an act-order checkpoint is not supported (the "
            "k-permutation crosses the lazy row-shard slices): load with "
            "tp_world=1 and shard in memory via lfmq.dist.tp.shard_llama, "
            "which routes act-order row-parallel layers through the "
            "gather-column plan")
    d, hq, hkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
    inter, vocab = cfg.intermediate_size, cfg.vocab_size

    col_parts: Dict[str, List[Tuple[int, int]]] = {}
    row_mods: List[str] = []
    for name, mod in model.named_modules():
        if name.endswith("qkv_proj"):
            col_parts[name] = [(0, hq * d), (hq * d, hkv * d),
                               ((hq + hkv) * d, hkv * d)]
        elif name.endswith("gate_up_proj"):
            col_parts[name] = [(0, inter), (inter, inter)]
        elif name.endswith("o_proj") or name.endswith("down_proj"):
            row_mods.append(name)
        elif name == "lm_head":
            col_parts[name] = [(0, vocab)]

    # first shard the module tree (reads full-size metadata from skeleton)
    shard_llama_skeleton(model, rank, world)

    def shard_fn(tname: str, shape: torch.Size):
        mod_name = tname.rsplit(".", 1)[0]
        field = tname.rsplit(".", 1)[1]
        if mod_name in col_parts:
            if field in ("qweight", "scales", "qzeros", "bias", "weight"):
                # safetensors lazy slices need contiguous ranges: single-part
                # shards slice at read time; multi-part (fused qkv/gate_up)
                # read whole and slice in the fix-up pass below
                sls = _row_slices(col_parts[mod_name], rank, world)
                return None if len(sls) > 1 else (sls[0],)
        if mod_name in row_mods:
            mod = get_module(model, mod_name)
            ks = mod.in_features  # already sharded size
            if field == "qweight":
                per_words = ks * mod.bits // 32
                return (slice(None), slice(rank * per_words, (rank + 1) * per_words))
            if field == "scales":
                Gs = ks // mod.group_size
                return (slice(None), slice(rank * Gs, (rank + 1) * Gs))
        return None

    state = load_packed(path, shard_fn=shard_fn, device="cpu")

    # fix up multi-part column shards + row-shard zeros/bias
    from ..ops import ref
    fixed = {}
    for tname, t in state.items():
        mod_name, field = tname.rsplit(".", 1)
        if mod_name in col_parts and len(col_parts[mod_name]) > 1 and \
                field in ("qweight", "scales", "qzeros", "bias", "weight"):
            sls = _row_slices(col_parts[mod_name], rank, world)
            t = _cat_rows(t, sls)
        elif mod_name in row_mods:
            if field == "qzeros":
                mod = get_module(model, mod_name)
                ks, g, bits = mod.in_features, mod.group_size, mod.bits
                Gs = ks // g
                z = ref.unpack_zeros(t, bits, ks * world // g)
                z = z[:, rank * Gs:(rank + 1) * Gs]
                t = ref.pack_zeros(z, bits)
            elif field == "bias" and rank != 0:
                t = torch.zeros_like(t)
        if tname == "lm_head.weight" and isinstance(model.lm_head, _GatheredLinear):
            # shard_fn already row-sliced it (single-part column shard)
            fixed["lm_head.local.weight"] = t
            continue
        fixed[tname] = t
    if (not any(k.startswith("lm_head.") for k in fixed)
            and "embed_tokens.weight" in fixed):
        # tied checkpoint: one stored copy (api.save dropped the alias);
        # the embedding is replicated, the head is vocab-column-sharded —
        # slice this rank's rows out of the replicated table
        sl = _row_slices([(0, vocab)], rank, world)[0]
        t = fixed["embed_tokens.weight"][sl]
        if isinstance(model.lm_head, _GatheredLinear):
            fixed["lm_head.local.weight"] = t
        else:
            fixed["lm_head.weight"] = t
    model.load_state_dict(fixed, strict=True)
    return model.to(device)


@torch.no_grad()
def shard_llama_skeleton(model: nn.Module, rank: int, world: int) -> nn.Module:
    """Resize a quantized skeleton's modules to per-rank shapes (no data)."""
    if world == 1:
        return model
    cfg = model.cfg
    hq, hkv, inter = cfg.n_heads, cfg.n_kv_heads, cfg.intermediate_size
    for layer in model.layers:
        attn, mlp = layer.self_attn, layer.mlp
        q = attn.qkv_proj
        replace_module(attn, "qkv_proj", ColumnParallelQLinear(
            q.in_features, q.out_features // world, q.bits, q.group_size,
            bias=q.bias is not None))
        o = attn.o_proj
        replace_module(attn, "o_proj", RowParallelQLinear(
            o.in_features // world, o.out_features, o.bits, o.group_size,
            bias=o.bias is not None))
        attn.n_heads = hq // world
        attn.n_kv_heads = hkv // world
        gu = mlp.gate_up_proj
        replace_module(mlp, "gate_up_proj", ColumnParallelQLinear(
            gu.in_features, gu.out_features // world, gu.bits, gu.group_size,
            bias=gu.bias is not None))
        dn = mlp.down_proj
        replace_module(mlp, "down_proj", RowParallelQLinear(
            dn.in_features // world, dn.out_features, dn.bits, dn.group_size,
            bias=dn.bias is not None))
        mlp.inter = inter // world
    lm = model.lm_head
    if isinstance(lm, QLinear):
        replace_module(model, "lm_head", ColumnParallelQLinear(
            lm.in_features, lm.out_features // world, lm.bits, lm.group_size,
            bias=lm.bias is not None, gather_output=True))
    else:
        new = nn.Linear(lm.in_features, lm.out_features // world, bias=False,
                        dtype=lm.weight.dtype)
        replace_module(model, "lm_head", _GatheredLinear(new))
    return model
