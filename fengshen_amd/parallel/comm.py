"""Collective helpers with gloo fallbacks (RCCL on GPU, gloo in CPU tests)."""
from __future__ import annotations

import torch
import torch.distributed as dist


def backend_is_nccl(group) -> bool:
    return dist.is_initialized() and dist.get_backend(group) == "nccl"


def reduce_scatter_flat(flat: torch.Tensor, shard_out: torch.Tensor, group,
                        rank: int, async_op: bool = False):
    """shard_out <- sum over ranks of flat[rank-th chunk]."""
    if backend_is_nccl(group):
        return dist.reduce_scatter_tensor(shard_out, flat, group=group,
                                          async_op=async_op)
    work = dist.all_reduce(flat, group=group, async_op=async_op)
    n = shard_out.numel()
    if async_op:
        class _W:
            def wait(self_inner):
                work.wait()
                shard_out.copy_(flat[rank * n:(rank + 1) * n])
        return _W()
    shard_out.copy_(flat[rank * n:(rank + 1) * n])
    return None


def all_gather_flat(flat_out: torch.Tensor, shard_in: torch.Tensor, group,
                    world: int, async_op: bool = False):
    """flat_out <- concat of shards over ranks."""
    if backend_is_nccl(group):
        return dist.all_gather_into_tensor(flat_out, shard_in.contiguous(),
                                           group=group, async_op=async_op)
    chunks = list(flat_out.chunk(world))
    return dist.all_gather(chunks, shard_in.contiguous(), group=group,
                           async_op=async_op)
