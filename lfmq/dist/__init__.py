"""Process-group plumbing: RCCL over xGMI on GPU, gloo on CPU for tests.

One process per GPU, torch.distributed; the "nccl" backend string IS RCCL
on ROCm.  The same code path runs under gloo with world_size>1 on a CPU
host (tests/test_tp_gloo.py) — the only difference is the backend string,
so multi-rank numerics are testable without a GPU cluster.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_GROUP: Optional[dist.ProcessGroup] = None


def init_dist(backend: Optional[str] = None) -> tuple[int, int]:
    """Initialise torch.distributed from torchrun env vars.

    Returns (rank, world_size).  No-op single-rank mode when WORLD_SIZE
    is absent or 1.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1 and "RANK" not in os.environ:
        return 0, 1
    # under torchrun (RANK set) we initialise even at world 1 so a 1-GPU
    # run exercises the real RCCL init path
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend)
        if backend == "nccl":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return dist.get_rank(), dist.get_world_size()


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def get_tp_group():
    """The tensor-parallel group.  Single-node: the whole world."""
    return _TP_GROUP  # None => default group


def _backend_is_gloo() -> bool:
    try:
        return dist.get_backend() == "gloo"
    except Exception:
        return False


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    if get_world_size() > 1:
        if t.dtype == torch.bfloat16 and _backend_is_gloo():
            f = t.float()
            dist.all_reduce(f, op=dist.ReduceOp.SUM, group=get_tp_group())
            t.copy_(f.to(t.dtype))
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM, group=get_tp_group())
    return t


def all_gather_cat(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    world = get_world_size()
    if world == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(parts, t.contiguous(), group=get_tp_group())
    return torch.cat(parts, dim=dim)


def all_to_all_single(out: torch.Tensor, inp: torch.Tensor,
                      out_splits=None, in_splits=None) -> torch.Tensor:
    if get_world_size() == 1:
        out.copy_(inp)
        return out
    dist.all_to_all_single(out, inp, out_splits, in_splits, group=get_tp_group())
    return out


def barrier() -> None:
    if get_world_size() > 1:
        dist.barrier()


def shutdown() -> None:
    """Tear the process group down explicitly.  Exiting the interpreter
    with a live gloo world aborts intermittently (SIGABRT in the gloo
    connection teardown at high rank counts) — every entry point that
    calls init_dist() should call this on the way out."""
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()
