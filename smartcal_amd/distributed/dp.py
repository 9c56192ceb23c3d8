"""Data-parallel gradient synchronization.

Each rank runs its own env + agent replica; after backward the flat
gradient pools are all-reduced (sum÷world) — ONE RCCL call per network
per step over xGMI, since every agent keeps its parameters in a single
flat buffer (`utils.flatten.FlatParams`). Install via the agents'
``grad_hook`` constructor argument.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

__all__ = ["allreduce_grad_hook", "init_from_env"]


def init_from_env() -> tuple[int, int, torch.device]:
    """init_process_group from torchrun env vars; returns
    (rank, world, device). Backend: nccl(=RCCL) with GPUs, gloo on CPU."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1 and not dist.is_initialized():
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    return rank, world, device


def allreduce_grad_hook(world: int | None = None):
    """Returns a ``grad_hook(fps)`` that averages each FlatParams'
    gradient pool across ranks (no-op when world == 1)."""
    if world is None:
        world = dist.get_world_size() if dist.is_initialized() else 1
    if world <= 1:
        return None
    inv = 1.0 / world

    def hook(fps):
        for fp in fps:
            dist.all_reduce(fp.flat_grad, op=dist.ReduceOp.SUM)
            fp.flat_grad.mul_(inv)
    return hook
