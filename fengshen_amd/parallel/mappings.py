"""TP region mappings — the four autograd-visible collectives.

Behavioral parity: reference mpu/mappings.py:110-192 (copy/reduce/scatter/
gather region functions).  RCCL all-reduce/all-gather over the TP group;
optional fp32 upcast for bf16 payloads (reference :37-46).
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from fengshen_amd.parallel import groups


def _reduce(input_: torch.Tensor) -> torch.Tensor:
    """All-reduce over the TP group (hot: 2×/layer fwd + 2×/layer bwd)."""
    if groups.get_tensor_model_parallel_world_size() == 1:
        return input_
    group = groups.get_tensor_model_parallel_group()
    if groups.get_fp32_allreduce() and input_.dtype in (torch.bfloat16, torch.float16):
        orig = input_.dtype
        buf = input_.float()
        dist.all_reduce(buf, group=group)
        return buf.to(orig)
    dist.all_reduce(input_, group=group)
    return input_


def _split(input_: torch.Tensor) -> torch.Tensor:
    """Keep this rank's chunk along the last dim."""
    tp = groups.get_tensor_model_parallel_world_size()
    if tp == 1:
        return input_
    last = input_.size(-1)
    assert last % tp == 0, f"last dim {last} not divisible by tp {tp}"
    rank = groups.get_tensor_model_parallel_rank()
    return input_.narrow(-1, rank * (last // tp), last // tp).contiguous()


def _gather(input_: torch.Tensor) -> torch.Tensor:
    """All-gather along the last dim."""
    tp = groups.get_tensor_model_parallel_world_size()
    if tp == 1:
        return input_
    group = groups.get_tensor_model_parallel_group()
    input_ = input_.contiguous()
    tensors = [torch.empty_like(input_) for _ in range(tp)]
    tensors[groups.get_tensor_model_parallel_rank()] = input_
    dist.all_gather(tensors, input_, group=group)
    return torch.cat(tensors, dim=-1)


class _CopyToModelParallelRegion(torch.autograd.Function):
    """Identity fwd, all-reduce bwd (input of a ColumnParallelLinear)."""

    @staticmethod
    def forward(ctx, input_):
        return input_

    @staticmethod
    def backward(ctx, grad_output):
        return _reduce(grad_output.contiguous())


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """All-reduce fwd, identity bwd (output of a RowParallelLinear)."""

    @staticmethod
    def forward(ctx, input_):
        return _reduce(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output


class _ScatterToModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _split(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather(grad_output)


class _GatherFromModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _gather(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _split(grad_output)


def copy_to_tensor_model_parallel_region(input_):
    return _CopyToModelParallelRegion.apply(input_)


def reduce_from_tensor_model_parallel_region(input_):
    return _ReduceFromModelParallelRegion.apply(input_)


def scatter_to_tensor_model_parallel_region(input_):
    return _ScatterToModelParallelRegion.apply(input_)


def gather_from_tensor_model_parallel_region(input_):
    return _GatherFromModelParallelRegion.apply(input_)
