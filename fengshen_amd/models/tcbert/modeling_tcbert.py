"""TCBert: topic-classification prompt BERT.

Behavioral parity: reference models/tcbert (366 LoC) — a prompt like
"这是一条关于[MASK][MASK]的新闻：" is prepended; classification reads the MLM
logits at the mask positions for each candidate label's tokens.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from transformers.utils import ModelOutput

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as TCBertConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForMaskedLM,
    MegatronBertPreTrainedModel,
)


@dataclass
class TCBertOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    label_logits: Optional[torch.Tensor] = None


class TCBertModel(MegatronBertPreTrainedModel):
    config_class = TCBertConfig

    def __init__(self, config: TCBertConfig):
        super().__init__(config)
        self.mlm = MegatronBertForMaskedLM(config)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                mask_positions=None, label_token_ids=None, labels=None, **_kw):
        """mask_positions [b, n_mask]; label_token_ids [n_labels, n_mask]:
        each label's verbalizer token ids; labels [b]."""
        out = self.mlm(input_ids, attention_mask, token_type_ids)
        logits = out.logits  # [b, s, V]
        b, n_mask = mask_positions.shape
        mask_logits = torch.gather(
            logits, 1,
            mask_positions[:, :, None].expand(-1, -1, logits.shape[-1]))
        logp = torch.log_softmax(mask_logits.float(), dim=-1)  # [b, n_mask, V]
        # score(label) = sum over mask slots of logP(verbalizer token);
        # label_token_ids: [L, n_mask]
        L = label_token_ids.shape[0]
        label_logits = torch.zeros(b, L, device=logits.device)
        for li in range(L):
            tok = label_token_ids[li]  # [n_mask]
            label_logits[:, li] = logp[
                :, torch.arange(n_mask, device=logits.device), tok].sum(-1)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(label_logits, labels)
        return TCBertOutput(loss=loss, label_logits=label_logits)
