"""Sequence-tagging heads over our MegatronBERT encoder.

Behavioral parity: reference models/tagging_models/bert_for_tagging.py:21-136
(BertLinear / BertCrf / BertSpan / BertBiaffine) + losses (focal, label
smoothing).  All use MegatronBertModel as the encoder.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
    MegatronBertPreTrainedModel,
)
from fengshen_amd.models.tagging_models.crf import CRF


class FocalLoss(nn.Module):
    """ref models/tagging_models/losses/focal_loss.py behavior."""

    def __init__(self, gamma: float = 2.0, reduction: str = "mean"):
        super().__init__()
        self.gamma = gamma
        self.reduction = reduction

    def forward(self, logits, target, ignore_index: int = -100):
        valid = target != ignore_index
        logits = logits[valid].float()
        target = target[valid]
        logp = F.log_softmax(logits, dim=-1)
        p = logp.exp()
        picked_logp = logp.gather(-1, target.unsqueeze(-1)).squeeze(-1)
        picked_p = p.gather(-1, target.unsqueeze(-1)).squeeze(-1)
        loss = -((1 - picked_p) ** self.gamma) * picked_logp
        return loss.mean() if self.reduction == "mean" else loss.sum()


class LabelSmoothingCE(nn.Module):
    def __init__(self, eps: float = 0.1):
        super().__init__()
        self.eps = eps

    def forward(self, logits, target, ignore_index: int = -100):
        valid = target != ignore_index
        logits = logits[valid].float()
        target = target[valid]
        n = logits.shape[-1]
        logp = F.log_softmax(logits, dim=-1)
        nll = -logp.gather(-1, target.unsqueeze(-1)).squeeze(-1)
        smooth = -logp.mean(dim=-1)
        return ((1 - self.eps) * nll + self.eps * smooth).mean()


def _loss_fn(loss_type: str):
    if loss_type == "focal":
        return FocalLoss()
    if loss_type == "lsce":
        return LabelSmoothingCE()
    return None  # plain CE


@dataclass
class TaggingOutput:
    loss: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None
    predictions: Optional[list] = None


class BertLinear(MegatronBertPreTrainedModel):
    """token classification w/ optional focal / label-smoothing loss."""

    def __init__(self, config, num_labels: int = 9, loss_type: str = "ce"):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.classifier = nn.Linear(config.hidden_size, num_labels)
        self.loss_fn = _loss_fn(loss_type)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, **_kw):
        h = self.bert(input_ids, attention_mask, token_type_ids).last_hidden_state
        logits = self.classifier(self.dropout(h))
        loss = None
        if labels is not None:
            if self.loss_fn is None:
                loss = F.cross_entropy(
                    logits.float().view(-1, self.num_labels), labels.view(-1),
                    ignore_index=-100)
            else:
                loss = self.loss_fn(logits.view(-1, self.num_labels),
                                    labels.view(-1))
        return TaggingOutput(loss=loss, logits=logits)


class BertCrf(MegatronBertPreTrainedModel):
    def __init__(self, config, num_labels: int = 9):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.classifier = nn.Linear(config.hidden_size, num_labels)
        self.crf = CRF(num_labels, batch_first=True)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, decode: bool = False, **_kw):
        h = self.bert(input_ids, attention_mask, token_type_ids).last_hidden_state
        logits = self.classifier(self.dropout(h)).float()
        loss = None
        preds = None
        if labels is not None:
            mask = attention_mask.bool() if attention_mask is not None else None
            safe = labels.clamp(min=0)
            loss = self.crf(logits, safe, mask=mask)
        if decode:
            mask = attention_mask.bool() if attention_mask is not None else None
            preds = self.crf.decode(logits, mask=mask)
        return TaggingOutput(loss=loss, logits=logits, predictions=preds)


class BertSpan(MegatronBertPreTrainedModel):
    """start/end span pointers (ref BertSpan)."""

    def __init__(self, config, num_labels: int = 9):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.start_fc = nn.Linear(config.hidden_size, num_labels)
        self.end_fc = nn.Linear(config.hidden_size, num_labels)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                start_positions=None, end_positions=None, **_kw):
        h = self.bert(input_ids, attention_mask, token_type_ids).last_hidden_state
        start_logits = self.start_fc(h)
        end_logits = self.end_fc(h)
        loss = None
        if start_positions is not None and end_positions is not None:
            ls = F.cross_entropy(start_logits.float().view(-1, self.num_labels),
                                 start_positions.view(-1), ignore_index=-100)
            le = F.cross_entropy(end_logits.float().view(-1, self.num_labels),
                                 end_positions.view(-1), ignore_index=-100)
            loss = (ls + le) / 2
        return TaggingOutput(loss=loss, logits=torch.stack(
            [start_logits, end_logits], dim=1))


class Biaffine(nn.Module):
    """biaffine scorer (ref models/ubert/modeling_ubert.py:219)."""

    def __init__(self, in_size: int, out_size: int):
        super().__init__()
        self.U = nn.Parameter(torch.randn(in_size + 1, out_size, in_size + 1)
                              * 0.02)

    def forward(self, x, y):
        x = torch.cat([x, torch.ones_like(x[..., :1])], dim=-1)
        y = torch.cat([y, torch.ones_like(y[..., :1])], dim=-1)
        return torch.einsum("bxi,ioj,byj->bxyo", x, self.U.to(x.dtype), y)


class BertBiaffine(MegatronBertPreTrainedModel):
    def __init__(self, config, num_labels: int = 9, biaffine_size: int = 128):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.start_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.end_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.biaffine = Biaffine(biaffine_size, num_labels)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                span_labels=None, **_kw):
        h = self.bert(input_ids, attention_mask, token_type_ids).last_hidden_state
        logits = self.biaffine(self.start_proj(h), self.end_proj(h))
        loss = None
        if span_labels is not None:
            loss = F.cross_entropy(
                logits.float().reshape(-1, self.num_labels),
                span_labels.view(-1), ignore_index=-100)
        return TaggingOutput(loss=loss, logits=logits)
