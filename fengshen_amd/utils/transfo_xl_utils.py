"""Transfo-XL sampling helpers (ref fengshen/utils/transfo_xl_utils.py).

Device-agnostic re-implementation against the MI355X-native
TransfoXLDenoiseModel API (forward(input_ids, mems, ...) -> .logits/.mems,
memory handled inside the model) instead of the reference's CUDA-pinned
logits/mems tuple interface.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn.functional as F


def top_k_logits(logits: torch.Tensor, top_k: int = 0, top_p: float = 0.0,
                 filter_value: float = -float("inf")) -> torch.Tensor:
    """top-k / nucleus filtering over the last dim (ref :6-27)."""
    if top_k > 0:
        kth = torch.topk(logits, top_k)[0][..., -1, None]
        logits = logits.masked_fill(logits < kth, filter_value)
    if top_p > 0.0:
        sorted_logits, sorted_idx = torch.sort(logits, dim=-1,
                                               descending=True)
        cum = torch.cumsum(F.softmax(sorted_logits, dim=-1), dim=-1)
        remove = cum > top_p
        remove[..., 1:] = remove[..., :-1].clone()
        remove[..., 0] = False
        mask = torch.zeros_like(logits, dtype=torch.bool).scatter(
            -1, sorted_idx, remove)
        logits = logits.masked_fill(mask, filter_value)
    return logits


def enforce_repetition_penalty(lprobs: torch.Tensor, prev_output_tokens,
                               repetition_penalty: float = 1.5):
    """CTRL-style penalty, in place on a 1D prob/logit row (ref :30-37)."""
    for previous_token in set(int(t) for t in prev_output_tokens):
        if lprobs[previous_token] < 0:
            lprobs[previous_token] *= repetition_penalty
        else:
            lprobs[previous_token] /= repetition_penalty


def switch(next_value: torch.Tensor, init: torch.Tensor,
           is_update: torch.Tensor) -> torch.Tensor:
    """Replace sampled tokens with real prompt tokens where the prompt is
    still being consumed (ref :40-42)."""
    is_update = is_update.type_as(next_value)
    return (1 - is_update) * init + is_update * next_value


def get_atten_mask(batch_size: int, seq_length: int,
                   memory_length: int = 0) -> torch.Tensor:
    """[b,1,s,s+M] band mask: causal over the segment, memory fully
    visible (ref :45-51)."""
    m = torch.ones((batch_size, 1, seq_length, seq_length + memory_length),
                   dtype=torch.int16)
    return torch.tril(torch.triu(m, 1 - seq_length + memory_length),
                      memory_length)


def get_masks_and_position_ids(data: torch.Tensor, mem_length: int = 0):
    """(mask, position_ids) pair for a [b,s] batch (ref :54-66)."""
    batch_size, seq_length = data.size()
    attention_mask = torch.ones(
        (1, seq_length, seq_length + mem_length), device=data.device)
    attention_mask = torch.tril(
        torch.triu(attention_mask, 1 - seq_length + mem_length),
        mem_length).unsqueeze(1)
    position_ids = torch.arange(
        seq_length, dtype=torch.long,
        device=data.device).unsqueeze(0).expand_as(data)
    return attention_mask, position_ids


@torch.no_grad()
def sample_sequence(model, tokens: torch.Tensor,
                    do_sampling: bool = True,
                    repetition_penalty: float = 1.0,
                    max_out_seq: Optional[int] = None,
                    mems: Optional[List[torch.Tensor]] = None,
                    end_token_id: Optional[int] = None,
                    temperature: float = 1.0, top_k: int = 0,
                    top_p: float = 0.0):
    """Single-sequence incremental sampling with XL memory (ref :185-245).

    tokens: [1, s] prompt.  Returns (token id list, mems).
    """
    counter = 0
    if end_token_id is None:
        end_token_id = 50000
    if max_out_seq is None:
        max_out_seq = 512
    org_context_length = tokens.size(1)
    while counter < max_out_seq:
        if counter == 0:
            out = model(input_ids=tokens, mems=mems)
        else:
            index = org_context_length + counter
            out = model(input_ids=tokens[:, index - 1:index], mems=mems)
        logits, mems = out.logits, out.mems
        logits = logits[:, -1].float() / temperature
        if do_sampling:
            logits = top_k_logits(logits, top_k=top_k, top_p=top_p)
        log_probs = F.softmax(logits, dim=-1)
        if repetition_penalty != 1.0:
            enforce_repetition_penalty(log_probs[0, :], tokens[0, :].tolist(),
                                       repetition_penalty)
        prev = torch.multinomial(log_probs.clamp(min=0), num_samples=1)[0]
        if int(prev) == end_token_id:
            break
        tokens = torch.cat((tokens, prev.view(1, 1)), dim=1)
        counter += 1
    out_list = tokens[0].detach().cpu().tolist()
    if end_token_id in out_list:
        out_list = out_list[:out_list.index(end_token_id)]
    return out_list, mems


@torch.no_grad()
def sample_sequence_batch(model, context_tokens_tensor: torch.Tensor,
                          context_length_tensor: torch.Tensor,
                          max_out_seq: Optional[int] = None,
                          mems: Optional[List[torch.Tensor]] = None,
                          end_token_id: Optional[int] = None,
                          repetition_penalty: float = 1.0,
                          temperature: float = 1.0, top_k: int = 0,
                          top_p: float = 0.0):
    """Batched sampling with ragged prompts (ref :69-182): sequences join
    generation as their prompts end (switch), leave as they emit the end
    token, and outputs are restored to input order.  Returns
    (list of token id lists, list of log probs)."""
    org_context_length = int(torch.min(context_length_tensor).item())
    batch_size = context_tokens_tensor.shape[0]
    tokens = context_tokens_tensor[:, :org_context_length]

    counter = 0
    if end_token_id is None:
        end_token_id = 50000
    if max_out_seq is None:
        max_out_seq = 512

    output_tokens_lists: List[List[int]] = []
    origin_order = torch.arange(batch_size, device=tokens.device)
    output_order: List[int] = []
    log_probs_tensor = torch.zeros(batch_size, device=tokens.device)
    log_probs_list: List[float] = []

    while counter < max_out_seq:
        index = org_context_length + counter
        if counter == 0:
            out = model(input_ids=tokens, mems=mems)
        else:
            out = model(input_ids=tokens[:, index - 1:index], mems=mems)
        logits, mems = out.logits, out.mems
        logits = logits[:, -1].float() / temperature
        logits = top_k_logits(logits, top_k=top_k, top_p=top_p)
        if repetition_penalty != 1.0:
            for bz in range(tokens.shape[0]):
                enforce_repetition_penalty(
                    logits[bz, :], tokens[bz, :].tolist(),
                    repetition_penalty)
        log_probs = F.softmax(logits, dim=-1)
        prev = torch.multinomial(log_probs, num_samples=1).view(-1)

        if index < int(torch.max(context_length_tensor).item()):
            prev = switch(prev, context_tokens_tensor[:, index],
                          context_length_tensor <= index)

        for i in range(tokens.shape[0]):
            if index > context_length_tensor[i] and prev[i] != end_token_id:
                log_probs_tensor[i] += math.log(
                    float(log_probs[i][prev[i]]) + 1e-6)
            if prev[i] == end_token_id:
                denom = float(context_length_tensor[i]) - index
                log_probs_tensor[i] /= denom if denom != 0 else 1.0

        stop_idx = prev == end_token_id
        if bool(torch.all(stop_idx)):
            output_order.extend(origin_order[stop_idx].tolist())
            break

        finished = tokens[stop_idx]
        output_tokens_lists.extend(finished.detach().cpu().tolist())
        log_probs_list.extend(log_probs_tensor[stop_idx].tolist())
        output_order.extend(origin_order[stop_idx].tolist())

        conti_idx = prev != end_token_id
        origin_order = origin_order[conti_idx]
        tokens, prev = tokens[conti_idx], prev[conti_idx]
        context_tokens_tensor = context_tokens_tensor[conti_idx]
        context_length_tensor = context_length_tensor[conti_idx]
        log_probs_tensor = log_probs_tensor[conti_idx]
        if mems:
            mems = [m[conti_idx] for m in mems]
        tokens = torch.cat((tokens, prev.view(-1, 1)), dim=-1)
        counter += 1
    else:
        # loop exhausted without all sequences ending
        pass

    if tokens.shape[0]:
        output_tokens_lists.extend(tokens.detach().cpu().tolist())
        log_probs_list.extend(log_probs_tensor.tolist())
        output_order.extend(origin_order.tolist())

    output_tokens_lists = [
        t[:t.index(end_token_id)] if end_token_id in t else t
        for t in output_tokens_lists]
    output_tokens_lists = [
        t for _, t in sorted(zip(output_order, output_tokens_lists))]
    output_log_probs = [
        p for _, p in sorted(zip(output_order, log_probs_list))]
    return output_tokens_lists, output_log_probs
