"""UBERT: unified multi-task span extraction.

Behavioral parity: reference models/ubert/modeling_ubert.py:256-310 —
[b, num_label, seq] stacked inputs, BERT encoder, query/key projections,
biaffine span scorer [b, num_label, s, s], (softmax x2 + BCE) x10 loss.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UbertConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
    MegatronBertPreTrainedModel,
)
from fengshen_amd.models.tagging_models.bert_for_tagging import Biaffine


@dataclass
class UbertOutput:
    loss: Optional[torch.Tensor] = None
    span_logits: Optional[torch.Tensor] = None


class UbertModel(MegatronBertPreTrainedModel):
    config_class = UbertConfig

    def __init__(self, config: UbertConfig, biaffine_size: int = 128):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.query_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.key_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.biaffine = Biaffine(biaffine_size, 1)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                span_labels=None, span_mask=None, **_kw):
        """input_ids [b, num_label, s] (one row per candidate label prompt);
        span_labels/[mask] [b, num_label, s, s]."""
        b, nl, s = input_ids.shape
        flat = lambda t: t.reshape(b * nl, s) if t is not None else None
        h = self.bert(flat(input_ids), flat(attention_mask),
                      flat(token_type_ids)).last_hidden_state
        q = self.query_proj(h)
        k = self.key_proj(h)
        logits = self.biaffine(q, k).squeeze(-1).view(b, nl, s, s)
        loss = None
        if span_labels is not None:
            bce = nn.functional.binary_cross_entropy_with_logits(
                logits.float(), span_labels.float(), reduction="none")
            if span_mask is not None:
                bce = bce * span_mask.float()
                loss = 10.0 * bce.sum() / span_mask.float().sum().clamp(min=1)
            else:
                loss = 10.0 * bce.mean()
        return UbertOutput(loss=loss, span_logits=logits)

    @torch.no_grad()
    def extract(self, input_ids, attention_mask=None, token_type_ids=None,
                threshold: float = 0.5):
        """decode spans (start, end) per (batch, label) above threshold."""
        out = self.forward(input_ids, attention_mask, token_type_ids)
        probs = out.span_logits.sigmoid()
        b, nl, s, _ = probs.shape
        results = []
        for bi in range(b):
            per_label = []
            for li in range(nl):
                spans = (probs[bi, li] > threshold).nonzero(as_tuple=False)
                per_label.append([(int(st), int(en), float(probs[bi, li, st, en]))
                                  for st, en in spans if en >= st])
            results.append(per_label)
        return results
