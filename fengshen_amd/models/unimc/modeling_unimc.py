"""UniMC: zero-shot label-as-option multiple choice.

Behavioral parity: reference models/unimc/modeling_unimc.py:297-332 —
options + question + text in one sequence; the classification logit is read
from the MLM head's yes-token column at each option position.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UniMCConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForMaskedLM,
    MegatronBertPreTrainedModel,
)


@dataclass
class UniMCOutput:
    loss: Optional[torch.Tensor] = None
    option_logits: Optional[torch.Tensor] = None


class UniMCModel(MegatronBertPreTrainedModel):
    config_class = UniMCConfig

    def __init__(self, config: UniMCConfig, yes_token_id: int = 1):
        super().__init__(config)
        self.mlm = MegatronBertForMaskedLM(config)
        self.yes_token_id = yes_token_id
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                option_positions=None, labels=None, **_kw):
        """option_positions [b, num_options] — indices of each option's
        [MASK]-style anchor token; labels [b] — correct option index."""
        out = self.mlm(input_ids, attention_mask, token_type_ids)
        logits = out.logits  # [b, s, V]
        yes = logits[..., self.yes_token_id]  # [b, s]
        option_logits = torch.gather(yes, 1, option_positions)  # [b, n_opt]
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(option_logits.float(), labels)
        return UniMCOutput(loss=loss, option_logits=option_logits)

    @torch.no_grad()
    def predict(self, input_ids, attention_mask, token_type_ids,
                option_positions):
        out = self.forward(input_ids, attention_mask, token_type_ids,
                           option_positions)
        return out.option_logits.argmax(dim=-1)
