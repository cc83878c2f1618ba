"""UniEX: unified information extraction (span + type matching).

Behavioral parity: reference models/uniex/modeling_uniex.py (fast/full
extract modes) — text and task-label prompts encoded together; span scorer
[b, s, s] picks entity spans, type scorer matches each span against the
label prompts' CLS representations.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from transformers.utils import ModelOutput

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UniEXConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
    MegatronBertPreTrainedModel,
)
from fengshen_amd.models.tagging_models.bert_for_tagging import Biaffine


@dataclass
class UniEXOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    span_logits: Optional[torch.Tensor] = None
    type_logits: Optional[torch.Tensor] = None


class UniEXModel(MegatronBertPreTrainedModel):
    config_class = UniEXConfig

    def __init__(self, config: UniEXConfig, biaffine_size: int = 128):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.start_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.end_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.span_biaffine = Biaffine(biaffine_size, 1)
        self.span_rep = nn.Linear(2 * config.hidden_size, config.hidden_size)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                label_prompt_ids=None, span_labels=None, type_labels=None,
                candidate_spans=None, **_kw):
        """label_prompt_ids [n_types, prompt_len]: one prompt per type;
        span_labels [b, s, s] binary; candidate_spans [b, n_cand, 2] with
        type_labels [b, n_cand] for type matching."""
        h = self.bert(input_ids, attention_mask,
                      token_type_ids).last_hidden_state
        span_logits = self.span_biaffine(
            self.start_proj(h), self.end_proj(h)).squeeze(-1)
        loss = None
        type_logits = None
        if label_prompt_ids is not None:
            th = self.bert(label_prompt_ids).last_hidden_state[:, 0]  # [T, h]
            if candidate_spans is not None:
                b, n_cand, _ = candidate_spans.shape
                starts = candidate_spans[..., 0]
                ends = candidate_spans[..., 1]
                hs = torch.gather(
                    h, 1, starts[:, :, None].expand(-1, -1, h.shape[-1]))
                he = torch.gather(
                    h, 1, ends[:, :, None].expand(-1, -1, h.shape[-1]))
                rep = self.span_rep(torch.cat([hs, he], dim=-1))
                type_logits = rep.float() @ th.float().t()  # [b, n_cand, T]
        if span_labels is not None:
            loss = nn.functional.binary_cross_entropy_with_logits(
                span_logits.float(), span_labels.float())
            if type_logits is not None and type_labels is not None:
                loss = loss + nn.functional.cross_entropy(
                    type_logits.view(-1, type_logits.shape[-1]),
                    type_labels.view(-1), ignore_index=-100)
        return UniEXOutput(loss=loss, span_logits=span_logits,
                           type_logits=type_logits)

    @torch.no_grad()
    def extract(self, input_ids, label_prompt_ids, attention_mask=None,
                threshold: float = 0.5, max_spans: int = 32):
        """fast extract: top spans then type match."""
        out = self.forward(input_ids, attention_mask)
        probs = out.span_logits.sigmoid()
        b, s, _ = probs.shape
        results = []
        th = self.bert(label_prompt_ids).last_hidden_state[:, 0]
        h = self.bert(input_ids, attention_mask).last_hidden_state
        for bi in range(b):
            upper = probs[bi].triu()
            flat = upper.flatten()
            vals, idxs = flat.topk(min(max_spans, flat.numel()))
            spans = []
            for v, ix in zip(vals, idxs):
                if v < threshold:
                    break
                st, en = int(ix // s), int(ix % s)
                rep = self.span_rep(torch.cat([h[bi, st], h[bi, en]], dim=-1))
                t = int((rep.float() @ th.float().t()).argmax())
                spans.append({"span": (st, en), "type": t, "score": float(v)})
            results.append(spans)
        return results
