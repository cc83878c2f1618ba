"""Vocab-parallel cross entropy.

Behavioral parity: reference mpu/cross_entropy.py:115 — softmax over a
vocab-sharded logit tensor without gathering the full vocab: one MAX
all-reduce + two SUM all-reduces over the TP group.
Used in our LLaMA/GPT heads instead of the reference's gather-then-CE
(saves the V-wide all-gather, SURVEY.md §3.3 note).
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from fengshen_amd.parallel import groups
from fengshen_amd.parallel.layers import VocabUtility


def _hip_ext(t):
    if t.is_cuda and t.dtype == torch.bfloat16:
        from fengshen_amd.ops import get_ext, use_hip
        if use_hip(t):
            ext = get_ext()
            if hasattr(ext, "vocab_ce_fwd"):
                return ext
    return None


class _FusedVocabParallelCrossEntropy(torch.autograd.Function):
    """HIP fused path: saves only per-row (M, Z) fp32 stats and recomputes
    the softmax from the bf16 logits in backward — the composite path
    materializes and SAVES a full fp32 softmax ([N, V/tp], 5.2 GB at the
    13B bench shape).  Cross-shard max/sum/pred reductions stay here so
    TP>1 keeps the reference all-reduce structure."""

    @staticmethod
    def forward(ctx, vocab_parallel_logits: torch.Tensor,
                target: torch.Tensor, ext):
        tp = groups.get_tensor_model_parallel_world_size()
        group = groups.get_tensor_model_parallel_group()
        rank = groups.get_tensor_model_parallel_rank()
        w = vocab_parallel_logits.size(-1)
        start, end = VocabUtility.vocab_range_from_per_partition_vocab_size(
            w, rank)
        logits_2d = vocab_parallel_logits.reshape(-1, w).contiguous()
        target_1d = target.reshape(-1).contiguous()
        m, z, pred = ext.vocab_ce_fwd(logits_2d, target_1d, start, end)
        if tp > 1:
            gm = m.clone()
            dist.all_reduce(gm, op=dist.ReduceOp.MAX, group=group)
            z = z * torch.exp(m - gm)
            dist.all_reduce(z, group=group)
            dist.all_reduce(pred, group=group)
            m = gm
        loss = torch.log(z) + m - pred
        ctx.save_for_backward(logits_2d, target_1d, m, z)
        ctx.vrange = (start, end)
        ctx.ext = ext
        ctx.shape = target.shape
        return loss.view(target.shape)

    @staticmethod
    def backward(ctx, grad_output):
        logits_2d, target_1d, m, z = ctx.saved_tensors
        start, end = ctx.vrange
        dl = ctx.ext.vocab_ce_bwd(
            logits_2d, target_1d, m, z,
            grad_output.reshape(-1).float(), start, end)
        return dl.view(*ctx.shape, -1), None, None


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, vocab_parallel_logits: torch.Tensor, target: torch.Tensor):
        tp = groups.get_tensor_model_parallel_world_size()
        group = groups.get_tensor_model_parallel_group()

        # stable softmax: global max over vocab shards
        logits_max = torch.max(vocab_parallel_logits, dim=-1)[0]
        if tp > 1:
            dist.all_reduce(logits_max, op=dist.ReduceOp.MAX, group=group)
        logits = vocab_parallel_logits - logits_max.unsqueeze(-1)

        # local target mask
        partition_vocab_size = vocab_parallel_logits.size(-1)
        rank = groups.get_tensor_model_parallel_rank()
        start, end = VocabUtility.vocab_range_from_per_partition_vocab_size(
            partition_vocab_size, rank)
        target_mask = (target < start) | (target >= end)
        masked_target = target.clone() - start
        masked_target[target_mask] = 0

        logits_2d = logits.view(-1, partition_vocab_size)
        masked_target_1d = masked_target.view(-1)
        arange_1d = torch.arange(logits_2d.size(0), device=logits_2d.device)
        predicted_logits_1d = logits_2d[arange_1d, masked_target_1d].clone()
        predicted_logits = predicted_logits_1d.view_as(target)
        predicted_logits[target_mask] = 0.0
        if tp > 1:
            dist.all_reduce(predicted_logits, group=group)

        exp_logits = torch.exp(logits.float())
        sum_exp_logits = exp_logits.sum(dim=-1)
        if tp > 1:
            dist.all_reduce(sum_exp_logits, group=group)

        loss = torch.log(sum_exp_logits) - predicted_logits.float()

        exp_logits.div_(sum_exp_logits.unsqueeze(-1))
        ctx.save_for_backward(exp_logits, target_mask, masked_target_1d)
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        softmax, target_mask, masked_target_1d = ctx.saved_tensors
        grad_input = softmax  # reuse buffer: d loss / d logits = softmax - 1[target]
        partition_vocab_size = softmax.size(-1)
        grad_2d = grad_input.view(-1, partition_vocab_size)
        arange_1d = torch.arange(grad_2d.size(0), device=grad_2d.device)
        grad_2d[arange_1d, masked_target_1d] -= (~target_mask).view(-1).float()
        grad_input.mul_(grad_output.unsqueeze(-1).float())
        return grad_input, None


def vocab_parallel_cross_entropy(vocab_parallel_logits, target):
    """Per-token CE loss [*, s] from vocab-sharded logits [*, s, V/tp]."""
    ext = _hip_ext(vocab_parallel_logits)
    if ext is not None and vocab_parallel_logits.size(-1) % 8 == 0:
        return _FusedVocabParallelCrossEntropy.apply(
            vocab_parallel_logits, target, ext)
    return _VocabParallelCrossEntropy.apply(vocab_parallel_logits, target)
