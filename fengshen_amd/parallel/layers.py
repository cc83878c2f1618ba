"""Tensor-parallel layers: Column/RowParallelLinear, VocabParallelEmbedding.

Behavioral parity: reference mpu/layers.py (ColumnParallelLinear :261,
RowParallelLinear :363, VocabParallelEmbedding :55) — re-designed for MI355X:
plain bf16 GEMMs route through hipBLASLt via F.linear; all TP collectives
go through fengshen_amd.parallel.mappings (RCCL over xGMI).
"""
from __future__ import annotations

from typing import Callable, Optional

import os
import torch
import torch.nn as nn
import torch.nn.functional as F
import torch.nn.init as init

from fengshen_amd.ops import functional as F_ops
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    scatter_to_tensor_model_parallel_region,
)


def ensure_divisibility(numerator: int, denominator: int) -> None:
    assert numerator % denominator == 0, f"{numerator} not divisible by {denominator}"


def divide(numerator: int, denominator: int) -> int:
    ensure_divisibility(numerator, denominator)
    return numerator // denominator


class VocabUtility:
    """Vocab-range helpers (reference: mpu/utils.py:56)."""

    @staticmethod
    def vocab_range_from_per_partition_vocab_size(per_partition_vocab_size, rank):
        first = rank * per_partition_vocab_size
        return first, first + per_partition_vocab_size

    @staticmethod
    def vocab_range_from_global_vocab_size(global_vocab_size, rank, world_size):
        per = divide(global_vocab_size, world_size)
        return VocabUtility.vocab_range_from_per_partition_vocab_size(per, rank)


def _init_partition(weight: torch.Tensor, init_method: Callable, full_shape=None,
                    partition_dim: int = 0, stride: int = 1):
    """Initialize a TP-partitioned weight so the sharded init matches the
    unsharded init sliced (master-weight init, reference mpu/layers.py:180-220):
    materialize the full fp32 master with the TP-IDENTICAL default RNG
    stream, then slice.  (The tracker's model-parallel stream is
    tp-distinct by design — right for dropout, wrong for master init:
    each rank would slice a different master.)"""
    tp = groups.get_tensor_model_parallel_world_size()
    if tp == 1 or full_shape is None:
        init_method(weight)
        return
    master = torch.empty(full_shape, dtype=torch.float32, device=weight.device)
    init_method(master)
    rank = groups.get_tensor_model_parallel_rank()
    per = weight.size(partition_dim)
    shard = master.narrow(partition_dim, rank * per, per)
    with torch.no_grad():
        weight.copy_(shard.to(weight.dtype))


class VocabParallelEmbedding(nn.Module):
    """Embedding sharded along the vocab dim; fwd = local lookup + masked
    zero-fill + TP all-reduce (reference mpu/layers.py:55-130)."""

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 init_method: Callable = init.xavier_normal_,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        tp = groups.get_tensor_model_parallel_world_size()
        self.vocab_start_index, self.vocab_end_index = (
            VocabUtility.vocab_range_from_global_vocab_size(
                num_embeddings, groups.get_tensor_model_parallel_rank(), tp))
        self.num_embeddings_per_partition = self.vocab_end_index - self.vocab_start_index
        self.weight = nn.Parameter(torch.empty(
            self.num_embeddings_per_partition, embedding_dim,
            dtype=dtype or torch.get_default_dtype()))
        setattr(self.weight, "tensor_model_parallel", True)
        setattr(self.weight, "partition_dim", 0)
        _init_partition(self.weight, init_method,
                        full_shape=(num_embeddings, embedding_dim), partition_dim=0)

    def forward(self, input_: torch.Tensor) -> torch.Tensor:
        if os.environ.get("FENGSHEN_CHECK_IDS") == "1":
            # opt-in host sync: an out-of-range id otherwise surfaces as
            # an opaque HSA_STATUS_ERROR_EXCEPTION device fault on ROCm
            mx, mn = int(input_.max()), int(input_.min())
            if mx >= self.num_embeddings or mn < 0:
                raise IndexError(
                    f"token id range [{mn}, {mx}] outside vocab "
                    f"[0, {self.num_embeddings})")
        tp = groups.get_tensor_model_parallel_world_size()
        if tp > 1:
            input_mask = (input_ < self.vocab_start_index) | (input_ >= self.vocab_end_index)
            masked_input = input_.clone() - self.vocab_start_index
            masked_input[input_mask] = 0
        else:
            masked_input = input_
        output = F.embedding(masked_input, self.weight)
        if tp > 1:
            output[input_mask, :] = 0.0
        return reduce_from_tensor_model_parallel_region(output)


class ColumnParallelLinear(nn.Module):
    """Y = XA + b with A sharded along columns (output dim).

    gather_output=True all-gathers Y across TP ranks (reference
    mpu/layers.py:261-360)."""

    def __init__(self, input_size: int, output_size: int, bias: bool = True,
                 gather_output: bool = True,
                 init_method: Callable = init.xavier_normal_,
                 stride: int = 1, skip_bias_add: bool = False,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.gather_output = gather_output
        self.skip_bias_add = skip_bias_add
        tp = groups.get_tensor_model_parallel_world_size()
        self.output_size_per_partition = divide(output_size, tp)
        dtype = dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(torch.empty(
            self.output_size_per_partition, input_size, dtype=dtype))
        setattr(self.weight, "tensor_model_parallel", True)
        setattr(self.weight, "partition_dim", 0)
        _init_partition(self.weight, init_method,
                        full_shape=(output_size, input_size), partition_dim=0)
        if bias:
            self.bias = nn.Parameter(torch.zeros(
                self.output_size_per_partition, dtype=dtype))
            setattr(self.bias, "tensor_model_parallel", True)
            setattr(self.bias, "partition_dim", 0)
        else:
            self.register_parameter("bias", None)

    def forward(self, input_: torch.Tensor):
        input_parallel = copy_to_tensor_model_parallel_region(input_)
        bias = self.bias if not self.skip_bias_add else None
        output_parallel = F_ops.linear(input_parallel, self.weight, bias)
        output = (gather_from_tensor_model_parallel_region(output_parallel)
                  if self.gather_output else output_parallel)
        if self.skip_bias_add:
            return output, self.bias
        return output


class RowParallelLinear(nn.Module):
    """Y = XA + b with A sharded along rows (input dim); fwd ends in a TP
    all-reduce (reference mpu/layers.py:363-470)."""

    def __init__(self, input_size: int, output_size: int, bias: bool = True,
                 input_is_parallel: bool = False,
                 init_method: Callable = init.xavier_normal_,
                 stride: int = 1, skip_bias_add: bool = False,
                 dtype: Optional[torch.dtype] = None,
                 reduce_output: bool = True):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.input_is_parallel = input_is_parallel
        self.skip_bias_add = skip_bias_add
        # reduce_output=False returns the rank-local partial sum so a
        # caller can DEFER the TP all-reduce and reduce several branches
        # once (the reference's GPT-J parallel-residual single deferred
        # all-reduce, transformer.py:710-752)
        self.reduce_output = reduce_output
        tp = groups.get_tensor_model_parallel_world_size()
        self.input_size_per_partition = divide(input_size, tp)
        dtype = dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(torch.empty(
            output_size, self.input_size_per_partition, dtype=dtype))
        setattr(self.weight, "tensor_model_parallel", True)
        setattr(self.weight, "partition_dim", 1)
        _init_partition(self.weight, init_method,
                        full_shape=(output_size, input_size), partition_dim=1)
        if bias:
            # bias added AFTER the all-reduce, only once (not per-shard)
            self.bias = nn.Parameter(torch.zeros(output_size, dtype=dtype))
        else:
            self.register_parameter("bias", None)

    def forward(self, input_: torch.Tensor):
        input_parallel = (input_ if self.input_is_parallel
                          else scatter_to_tensor_model_parallel_region(input_))
        output_parallel = F_ops.linear(input_parallel, self.weight)
        if not self.reduce_output:
            # partial sum: the caller owns the (deferred) all-reduce;
            # bias must be added after that reduce, so hand it back
            if self.skip_bias_add or self.bias is not None:
                return output_parallel, self.bias
            return output_parallel
        output_ = reduce_from_tensor_model_parallel_region(output_parallel)
        if self.skip_bias_add:
            return output_, self.bias
        if self.bias is not None:
            output_ = output_ + self.bias
        return output_


class ParallelRelativePositionBias(nn.Module):
    """T5-style bucketed relative attention bias, sharded over heads
    (reference mpu/layers.py:133-258: per-TP-rank head slice of the
    [num_buckets, heads] embedding; bias added to attention scores).

    forward(q_len, k_len) -> [1, heads/tp, q_len, k_len].
    """

    def __init__(self, num_buckets: int = 32, max_distance: int = 128,
                 num_heads: int = 12, causal: bool = True,
                 init_method: Callable = init.xavier_normal_):
        super().__init__()
        tp = groups.get_tensor_model_parallel_world_size()
        self.num_buckets = num_buckets
        self.max_distance = max_distance
        self.causal = causal
        self.heads_per_partition = divide(num_heads, tp)
        self.weight = nn.Parameter(
            torch.empty(num_buckets, self.heads_per_partition,
                        dtype=torch.float32))
        setattr(self.weight, "tensor_model_parallel", True)
        setattr(self.weight, "partition_dim", 1)
        _init_partition(self.weight, init_method,
                        full_shape=(num_buckets, num_heads),
                        partition_dim=1)
        self._cache = {}

    def _bucket(self, relative_position: torch.Tensor) -> torch.Tensor:
        """T5 relative_position_bucket: half the buckets exact, half
        log-spaced out to max_distance."""
        num_buckets = self.num_buckets
        ret = torch.zeros_like(relative_position)
        n = -relative_position
        if not self.causal:
            num_buckets //= 2
            ret = ret + (n < 0).long() * num_buckets
            n = n.abs()
        else:
            n = torch.clamp(n, min=0)
        max_exact = num_buckets // 2
        is_small = n < max_exact
        import math as _m
        val_large = max_exact + (
            torch.log(n.float() / max_exact + 1e-6)
            / _m.log(self.max_distance / max_exact)
            * (num_buckets - max_exact)).long()
        val_large = torch.clamp(val_large, max=num_buckets - 1)
        return ret + torch.where(is_small, n, val_large)

    def forward(self, q_len: int, k_len: int) -> torch.Tensor:
        key = (q_len, k_len, str(self.weight.device))
        if key not in self._cache:
            ctx = torch.arange(q_len, device=self.weight.device)[:, None]
            mem = torch.arange(k_len, device=self.weight.device)[None, :]
            self._cache[key] = self._bucket(mem - ctx)
        buckets = self._cache[key]
        bias = F.embedding(buckets, self.weight)        # [q, k, h/tp]
        return bias.permute(2, 0, 1).unsqueeze(0).to(self.weight.dtype)
