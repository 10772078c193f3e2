"""Expert parallelism for Mixtral: experts partitioned across ranks,
token dispatch/return by all_to_all (BASELINE.json
config #5).

Protocol per MoE layer (ranks hold replicated activations; each rank
owns the MoE compute for its token shard and n_experts/world experts):
  1. shard tokens: rank r takes its slice of the flattened tokens
  2. route locally (router weights are replicated)
  3. dispatch: each (token, expert) slot goes to the rank owning the
     expert — all_to_all_single on the token rows + a parallel
     all_to_all of the expert-local ids (ragged split sizes from an
     initial count exchange)
  4. local expert compute (the same per-expert kernels as TP mode)
  5. combine: reverse all_to_all returns slot outputs; weighted
     scatter-add into the token shard; all_gather re-replicates.
Split sizes are data-dependent: the count exchange syncs to host — the
cost noted in SURVEY.md §2.5 (decode messages are latency-bound).
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist
import torch.nn as nn

from . import get_rank, get_world_size
from ..models.mixtral import MixtralMoE


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

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, s, h = x.shape
        x2 = x.reshape(-1, h)
        t = x2.shape[0]
        world, rank = self.world, self.rank
        sizes = _even_sizes(t, world)
        start = sum(sizes[:rank])
        xl = x2[start:start + sizes[rank]]                  # local token shard

        logits = self.gate(xl).float()
        weights, ids = torch.topk(logits, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1)

        # slots sorted by destination rank (= expert // e_per_rank)
        tl = xl.shape[0]
        flat_ids = ids.reshape(-1)
        dest = flat_ids // self.e_per_rank
        order = torch.argsort(dest, stable=True)
        send_rows = xl[(order // self.top_k)]
        send_eids = (flat_ids[order] % self.e_per_rank).to(torch.int32)
        send_counts = torch.bincount(dest, minlength=world).tolist()

        # count exchange -> recv splits
        cnt_in = torch.tensor(send_counts, dtype=torch.int64)
        cnt_out = torch.zeros(world, dtype=torch.int64)
        dist.all_to_all_single(cnt_out, cnt_in)
        recv_counts = cnt_out.tolist()
        n_recv = int(sum(recv_counts))

        recv_rows = x2.new_empty(n_recv, h)
        dist.all_to_all_single(recv_rows, send_rows.contiguous(),
                               recv_counts, send_counts)
        recv_eids = torch.zeros(n_recv, dtype=torch.int32,
                                device=send_eids.device)
        dist.all_to_all_single(recv_eids, send_eids.contiguous(),
                               recv_counts, send_counts)

        # local expert compute
        out_rows = torch.zeros_like(recv_rows)
        for le in range(self.e_per_rank):
            m = recv_eids == le
            if bool(m.any()):
                out_rows[m] = self.experts[le](recv_rows[m])

        # return trip (reversed splits)
        back_rows = x2.new_empty(send_rows.shape)
        dist.all_to_all_single(back_rows, out_rows.contiguous(),
                               send_counts, recv_counts)

        # weighted combine into the local shard (undo the sort)
        yl = torch.zeros_like(xl, dtype=torch.float32)
        src_tok = (order // self.top_k).long()
        slot_w = weights.reshape(-1)[order].float()
        yl.index_add_(0, src_tok, back_rows.float() * slot_w.unsqueeze(-1))
        yl = yl.to(x2.dtype)

        # re-replicate the shards.  all_gather needs EQUAL sizes on every
        # rank (gloo and RCCL both reject ragged lists), but the token
        # shards are uneven whenever t % world != 0 — pad to the max
        # shard and slice per rank after the collective.
        mx = max(sizes)
        buf = x2.new_zeros(mx, h)
        buf[:yl.shape[0]] = yl
        parts = [x2.new_empty(mx, h) for _ in range(world)]
        dist.all_gather(parts, buf)
        return torch.cat([parts[r][:sizes[r]] for r in range(world)],
                         dim=0).reshape(b, s, h)


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
