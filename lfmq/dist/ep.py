"""Expert parallelism for Mixtral: experts partitioned across ranks,
token dispatch/return by all_to_all (BASELINE.json
config #5).

Two dispatch protocols per MoE layer (ranks hold replicated activations;
each rank owns the MoE compute for its token shard and n_experts/world
experts):

STATIC (decode-grade, default for small token counts): every rank sends
a fixed-capacity slab of C = tl_max * top_k slots to every other rank
(worst case: all of a shard's slots route to one rank).  Split sizes are
compile-time constants of (t, world), so there is NO count exchange, NO
host sync, and every collective has a static shape — the whole MoE
forward is hipGraph-capturable.  Expert ids ride inside the row payload
(int32 bitcast to two bf16 columns), so dispatch is ONE all_to_all.
Local compute runs every local expert over the recv slab with a one-hot
mask (e_per_rank-fold redundancy — at decode the slab is tiny and the
collectives dominate, so redundant GEMV beats a host-synced ragged
dispatch).  3 collectives/layer total: dispatch, combine, re-replicate.

DYNAMIC (prefill / large t): ragged all_to_all with an up-front count
exchange.  The exchange tensors live on the computing device (RCCL
requires device tensors; round-1 had them on CPU — ADVICE.md r1) and
the single .tolist() on the counts is the one host sync.

  1. shard tokens: rank r takes its slice of the flattened tokens
  2. route locally (router weights are replicated)
  3. dispatch slots to the rank owning each expert
  4. local expert compute (same per-expert kernels as TP mode)
  5. combine: reverse all_to_all; weighted scatter-add into the token
     shard; padded all_gather re-replicates.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist
import torch.nn as nn

from . import get_rank, get_world_size
from ..models.mixtral import MixtralMoE

# static protocol is used when tokens-per-shard * top_k is at most this
# (decode and small prefills); beyond it the capacity padding would cost
# more bandwidth than the dynamic path's one host sync.
STATIC_SLOT_LIMIT = 64


def _even_sizes(total: int, world: int) -> List[int]:
    base = total // world
    rem = total % world
    return [base + (1 if r < rem else 0) for r in range(world)]


class ExpertParallelMoE(nn.Module):
    """Wraps a (quantized) MixtralMoE; this rank keeps only its expert
    slice; forward runs the all-to-all dispatch/combine protocol."""

    def __init__(self, moe: MixtralMoE, rank: int, world: int):
        super().__init__()
        assert moe.n_experts % world == 0, "n_experts % ep_world != 0"
        self.n_experts = moe.n_experts
        self.top_k = moe.top_k
        self.rank = rank
        self.world = world
        self.e_per_rank = moe.n_experts // world
        self.gate = moe.gate  # replicated router
        lo = rank * self.e_per_rank
        self.experts = nn.ModuleList(moe.experts[lo + i]
                                     for i in range(self.e_per_rank))

    # ---------------------------------------------------------- shared
    def _route(self, xl: torch.Tensor):
        logits = self.gate(xl).float()
        weights, ids = torch.topk(logits, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1)
        return weights, ids

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, s, h = x.shape
        x2 = x.reshape(-1, h)
        t = x2.shape[0]
        world, rank = self.world, self.rank
        sizes = _even_sizes(t, world)
        start = sum(sizes[:rank])
        xl = x2[start:start + sizes[rank]]                  # local token shard

        tl_max = max(sizes)
        if tl_max * self.top_k <= STATIC_SLOT_LIMIT:
            yl = self._moe_static(xl, tl_max, h)
        else:
            yl = self._moe_dynamic(xl, h)

        # re-replicate the shards.  all_gather needs EQUAL sizes on every
        # rank (gloo and RCCL both reject ragged lists), but the token
        # shards are uneven whenever t % world != 0 — pad to the max
        # shard and slice per rank after the collective.
        buf = x2.new_zeros(tl_max, h)
        buf[:yl.shape[0]] = yl
        parts = [x2.new_empty(tl_max, h) for _ in range(world)]
        dist.all_gather(parts, buf)
        return torch.cat([parts[r][:sizes[r]] for r in range(world)],
                         dim=0).reshape(b, s, h)

    # ------------------------------------------- static (decode-grade)
    def _moe_static(self, xl: torch.Tensor, tl_max: int, h: int):
        """Capacity-padded dispatch: fixed C slots per destination rank,
        zero host syncs, static shapes (graph-capturable)."""
        world = self.world
        C = tl_max * self.top_k          # worst case: every slot one rank
        dev = xl.device
        tl = xl.shape[0]
        assert xl.dtype in (torch.bfloat16, torch.float32), \
            "static EP eid bitcast needs bf16/fp32 rows"

        weights, ids = self._route(xl)
        n = tl * self.top_k
        flat_ids = ids.reshape(-1)
        dest = flat_ids // self.e_per_rank

        # slot positions inside each destination slab, all on device
        order = torch.argsort(dest, stable=True)
        sorted_dest = dest[order]
        counts = torch.bincount(dest, minlength=world)
        offs = torch.cumsum(counts, 0) - counts
        pos = torch.arange(n, device=dev) - offs[sorted_dest]
        slot = sorted_dest * C + pos                     # [n] into [world*C]

        # payload: rows + (local expert id + 1) bitcast into 2 bf16 cols;
        # eid column 0 everywhere else marks padding (eid_enc 0 = invalid)
        send = xl.new_zeros(world * C, h + 2)
        src_tok = order // self.top_k
        send[slot, :h] = xl[src_tok]
        # int32 -> [.,2] bf16 bitcast; bf16 subnormals survive a cast to
        # fp32 and back exactly (CPU/gloo tests run fp32 activations)
        eid_enc = torch.zeros(world * C, 1, dtype=torch.int32, device=dev)
        eid_enc[slot, 0] = (flat_ids[order] % self.e_per_rank).to(torch.int32) + 1
        send[:, h:] = eid_enc.view(torch.bfloat16).to(send.dtype)

        recv = xl.new_empty(world * C, h + 2)
        dist.all_to_all_single(recv, send.contiguous())

        rows = recv[:, :h].contiguous()  # ext kernels need contiguous x
        recv_enc = recv[:, h:].to(torch.bfloat16).contiguous() \
            .view(torch.int32).reshape(world * C)
        # every local expert over the whole slab, one-hot masked: no
        # data-dependent shapes, no host sync.
        out_rows = torch.zeros_like(rows)
        for le in range(self.e_per_rank):
            m = (recv_enc == le + 1).unsqueeze(-1).to(rows.dtype)
            out_rows = out_rows + self.experts[le](rows) * m

        back = xl.new_empty(world * C, h)
        dist.all_to_all_single(back, out_rows.contiguous())

        yl = torch.zeros(tl, h, dtype=torch.float32, device=dev)
        slot_w = weights.reshape(-1)[order].float()
        yl.index_add_(0, src_tok, back[slot].float() * slot_w.unsqueeze(-1))
        return yl.to(xl.dtype)

    # ------------------------------------------------ dynamic (prefill)
    def _moe_dynamic(self, xl: torch.Tensor, h: int):
        world = self.world
        dev = xl.device
        weights, ids = self._route(xl)
        tl = xl.shape[0]
        flat_ids = ids.reshape(-1)
        dest = flat_ids // self.e_per_rank
        order = torch.argsort(dest, stable=True)
        send_rows = xl[(order // self.top_k)]
        send_eids = (flat_ids[order] % self.e_per_rank).to(torch.int32)

        # count exchange on the computing device (RCCL needs device
        # tensors); the .tolist() below is the protocol's one host sync.
        cnt_in = torch.bincount(dest, minlength=world).to(
            device=dev, dtype=torch.int64)
        cnt_out = torch.zeros(world, dtype=torch.int64, device=dev)
        dist.all_to_all_single(cnt_out, cnt_in)
        send_counts = cnt_in.tolist()
        recv_counts = cnt_out.tolist()
        n_recv = int(sum(recv_counts))

        recv_rows = xl.new_empty(n_recv, h)
        dist.all_to_all_single(recv_rows, send_rows.contiguous(),
                               recv_counts, send_counts)
        recv_eids = torch.zeros(n_recv, dtype=torch.int32, device=dev)
        dist.all_to_all_single(recv_eids, send_eids.contiguous(),
                               recv_counts, send_counts)

        # local expert compute
        out_rows = torch.zeros_like(recv_rows)
        for le in range(self.e_per_rank):
            m = recv_eids == le
            if bool(m.any()):
                out_rows[m] = self.experts[le](recv_rows[m])

        # return trip (reversed splits)
        back_rows = xl.new_empty(send_rows.shape)
        dist.all_to_all_single(back_rows, out_rows.contiguous(),
                               send_counts, recv_counts)

        # weighted combine into the local shard (undo the sort)
        yl = torch.zeros(tl, h, dtype=torch.float32, device=dev)
        src_tok = (order // self.top_k).long()
        slot_w = weights.reshape(-1)[order].float()
        yl.index_add_(0, src_tok, back_rows.float() * slot_w.unsqueeze(-1))
        return yl.to(xl.dtype)


@torch.no_grad()
def apply_expert_parallel(model, rank: int = None, world: int = None):
    """Replace every MixtralMoE with its expert-parallel shard."""
    from ..quant.common import replace_module
    rank = get_rank() if rank is None else rank
    world = get_world_size() if world is None else world
    if world == 1:
        return model
    for name, mod in list(model.named_modules()):
        if isinstance(mod, MixtralMoE):
            replace_module(model, name, ExpertParallelMoE(mod, rank, world))
    return model
