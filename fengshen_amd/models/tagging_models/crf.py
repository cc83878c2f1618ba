"""Linear-chain CRF (forward log-likelihood + Viterbi decode).

Behavioral parity: reference models/tagging_models/layers/crf.py:212-368
(pure-python CRF with batched forward/viterbi, mask support).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


class CRF(nn.Module):
    def __init__(self, num_tags: int, batch_first: bool = True):
        super().__init__()
        assert num_tags > 0
        self.num_tags = num_tags
        self.batch_first = batch_first
        self.start_transitions = nn.Parameter(torch.empty(num_tags))
        self.end_transitions = nn.Parameter(torch.empty(num_tags))
        self.transitions = nn.Parameter(torch.empty(num_tags, num_tags))
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.uniform_(self.start_transitions, -0.1, 0.1)
        nn.init.uniform_(self.end_transitions, -0.1, 0.1)
        nn.init.uniform_(self.transitions, -0.1, 0.1)

    def forward(self, emissions: torch.Tensor, tags: torch.Tensor,
                mask: Optional[torch.Tensor] = None,
                reduction: str = "mean") -> torch.Tensor:
        """Negative log likelihood (lower = better fit)."""
        if self.batch_first:
            emissions = emissions.transpose(0, 1)
            tags = tags.transpose(0, 1)
            if mask is not None:
                mask = mask.transpose(0, 1)
        if mask is None:
            mask = torch.ones_like(tags, dtype=torch.bool)
        mask = mask.bool()
        numerator = self._score(emissions.float(), tags, mask)
        denominator = self._normalizer(emissions.float(), mask)
        llh = numerator - denominator
        nll = -llh
        if reduction == "none":
            return nll
        if reduction == "sum":
            return nll.sum()
        if reduction == "mean":
            return nll.mean()
        return nll.sum() / mask.float().sum()  # token_mean

    def _score(self, emissions, tags, mask):
        seq_len, batch = tags.shape
        mask = mask.float()
        score = self.start_transitions[tags[0]]
        score += emissions[0, torch.arange(batch), tags[0]]
        for i in range(1, seq_len):
            score += self.transitions[tags[i - 1], tags[i]] * mask[i]
            score += emissions[i, torch.arange(batch), tags[i]] * mask[i]
        seq_ends = mask.long().sum(dim=0) - 1
        last_tags = tags[seq_ends, torch.arange(batch)]
        score += self.end_transitions[last_tags]
        return score

    def _normalizer(self, emissions, mask):
        seq_len = emissions.shape[0]
        score = self.start_transitions + emissions[0]
        for i in range(1, seq_len):
            broadcast = score.unsqueeze(2)
            emit = emissions[i].unsqueeze(1)
            next_score = broadcast + self.transitions + emit
            next_score = torch.logsumexp(next_score, dim=1)
            score = torch.where(mask[i].unsqueeze(1), next_score, score)
        score += self.end_transitions
        return torch.logsumexp(score, dim=1)

    @torch.no_grad()
    def decode(self, emissions: torch.Tensor,
               mask: Optional[torch.Tensor] = None) -> List[List[int]]:
        """Viterbi best paths."""
        if self.batch_first:
            emissions = emissions.transpose(0, 1)
            if mask is not None:
                mask = mask.transpose(0, 1)
        if mask is None:
            mask = torch.ones(emissions.shape[:2], dtype=torch.bool,
                              device=emissions.device)
        mask = mask.bool()
        emissions = emissions.float()
        seq_len, batch = mask.shape
        score = self.start_transitions + emissions[0]
        history = []
        for i in range(1, seq_len):
            broadcast = score.unsqueeze(2)
            emit = emissions[i].unsqueeze(1)
            next_score = broadcast + self.transitions + emit
            next_score, indices = next_score.max(dim=1)
            score = torch.where(mask[i].unsqueeze(1), next_score, score)
            history.append(indices)
        score += self.end_transitions
        seq_ends = mask.long().sum(dim=0) - 1
        best_tags_list = []
        for b in range(batch):
            _, best_last = score[b].max(dim=0)
            best_tags = [best_last.item()]
            for hist in reversed(history[:seq_ends[b]]):
                best_tags.append(hist[b][best_tags[-1]].item())
            best_tags.reverse()
            best_tags_list.append(best_tags)
        return best_tags_list
