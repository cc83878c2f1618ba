"""Erlangshen-MegatronBERT: pre-LN BERT with MLM + SOP heads, MI355X-native.

Behavioral parity: the reference trains Erlangshen with HF
MegatronBertForPreTraining (pretrain_erlangshen.py:138-141, losses: MLM CE
with -100 ignore + sentence-order binary CE).  Megatron-style = pre-LN blocks
+ final LayerNorm (vs original BERT post-LN).  Hot ops (LayerNorm, fused
scaled-masked softmax, bias-GELU) are our HIP kernels.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
from transformers import PreTrainedModel
from transformers.modeling_outputs import (
    BaseModelOutput,
    MaskedLMOutput,
    SequenceClassifierOutput,
    TokenClassifierOutput,
)
from transformers.utils import ModelOutput

from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelTransformerLayer,
    init_normal,
    parallel_lm_logits,
    scaled_init_normal,
)
from fengshen_amd.ops import functional as F_ops
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from fengshen_amd.parallel.layers import VocabParallelEmbedding
from fengshen_amd.parallel.mappings import gather_from_tensor_model_parallel_region
from fengshen_amd.parallel.random import checkpoint as activation_checkpoint
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig,
)


class MegatronBertPreTrainedModel(PreTrainedModel):
    config_class = MegatronBertConfig
    base_model_prefix = "bert"
    supports_gradient_checkpointing = True

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=self.config.initializer_range)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=self.config.initializer_range)


class MegatronBertEmbeddings(nn.Module):
    def __init__(self, config: MegatronBertConfig):
        super().__init__()
        im = init_normal(config.initializer_range)
        self.word_embeddings = VocabParallelEmbedding(
            config.vocab_size, config.hidden_size, init_method=im)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.hidden_size)
        self.token_type_embeddings = nn.Embedding(
            config.type_vocab_size, config.hidden_size)
        # megatron places no LN here (final + per-layer pre-LN instead)
        self.dropout = nn.Dropout(config.hidden_dropout)

    def forward(self, input_ids, token_type_ids=None, position_ids=None):
        b, s = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(s, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        emb = (self.word_embeddings(input_ids)
               + self.position_embeddings(position_ids)
               + self.token_type_embeddings(token_type_ids))
        return self.dropout(emb)


class MegatronBertModel(MegatronBertPreTrainedModel):
    def __init__(self, config: MegatronBertConfig, add_pooling_layer: bool = True):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range, config.num_hidden_layers)
        self.embeddings = MegatronBertEmbeddings(config)
        self.encoder = nn.ModuleList([
            ParallelTransformerLayer(
                config.hidden_size, config.num_attention_heads, causal=False,
                norm="layernorm", norm_eps=config.layer_norm_eps,
                mlp_type="gelu", ffn_hidden_size=config.intermediate_size,
                rotary=False, max_positions=config.max_position_embeddings,
                attention_dropout=config.attention_dropout,
                hidden_dropout=config.hidden_dropout,
                bias=True, init_method=im, output_init_method=om, layer_idx=i)
            for i in range(config.num_hidden_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size) \
            if add_pooling_layer else None
        self.gradient_checkpointing = False
        # selective activation ckpt (cf. LlamaModel): every k-th layer
        # keeps activations instead of recomputing
        self.gradient_checkpointing_skip_interval = 0
        self.post_init()

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def set_input_embeddings(self, v):
        self.embeddings.word_embeddings = v

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                position_ids=None, output_hidden_states: bool = False,
                **_kw):
        h = self.embeddings(input_ids, token_type_ids, position_ids)
        mask = None
        if attention_mask is not None:
            if attention_mask.dim() == 3:
                # per-sample 2D mask [b, sq, sk] (e.g. UniMC option
                # isolation): 1=keep -> internal True=masked [b,1,sq,sk]
                mask = (attention_mask == 0)[:, None, :, :]
            else:
                # HF 1=keep [b,s] -> internal True=masked [b,1,1,s]
                mask = (attention_mask == 0)[:, None, None, :]
        all_h = (h,) if output_hidden_states else None
        skip = self.gradient_checkpointing_skip_interval
        for i, layer in enumerate(self.encoder):
            ckpt = self.gradient_checkpointing and self.training
            if ckpt and skip and (i % skip == skip - 1):
                ckpt = False
            if ckpt:
                h = activation_checkpoint(
                    lambda x, m, lyr=layer: lyr(x, attention_mask=m), h, mask)
            else:
                h = layer(h, attention_mask=mask)
            if output_hidden_states:
                all_h = all_h + (h,)
        h = self.ln_f(h)
        pooled = None
        if self.pooler is not None:
            pooled = torch.tanh(self.pooler(h[:, 0]))
        if output_hidden_states:
            # HF convention: embeddings output + every layer output
            return BaseModelOutput(last_hidden_state=h, hidden_states=all_h)
        return BaseModelOutput(last_hidden_state=h,
                               hidden_states=(pooled,) if pooled is not None else None)


class MegatronBertLMHead(nn.Module):
    """MLM transform: dense + gelu + LN, then tied vocab-parallel decode."""

    def __init__(self, config):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.layer_norm = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.bias = nn.Parameter(torch.zeros(config.vocab_size))

    def forward(self, hidden, word_embeddings_weight):
        h = self.dense(hidden)
        h = F_ops.eager_gelu(h.float()).to(h.dtype)
        h = self.layer_norm(h)
        logits = parallel_lm_logits(h, word_embeddings_weight,
                                    parallel_output=True)
        tp = groups.get_tensor_model_parallel_world_size()
        if tp > 1:
            rank = groups.get_tensor_model_parallel_rank()
            per = self.bias.shape[0] // tp
            logits = logits + self.bias[rank * per:(rank + 1) * per]
        else:
            logits = logits + self.bias
        return logits


from dataclasses import dataclass


@dataclass
class MegatronBertPreTrainingOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    prediction_logits: Optional[torch.Tensor] = None
    seq_relationship_logits: Optional[torch.Tensor] = None


class MegatronBertForPreTraining(MegatronBertPreTrainedModel):
    """MLM + sentence-order prediction (the Erlangshen pretrain objective)."""

    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=True)
        self.cls = MegatronBertLMHead(config)
        self.seq_relationship = nn.Linear(config.hidden_size, 2)
        self.post_init()

    def get_input_embeddings(self):
        return self.bert.embeddings.word_embeddings

    def gradient_checkpointing_enable(self, skip_interval: int = 0, **_kw):
        self.bert.gradient_checkpointing = True
        self.bert.gradient_checkpointing_skip_interval = max(skip_interval, 0)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, next_sentence_label=None, position_ids=None,
                return_dict=True, **_kw):
        out = self.bert(input_ids, attention_mask, token_type_ids, position_ids)
        h = out.last_hidden_state
        pooled = torch.tanh(self.bert.pooler(h[:, 0]))
        logits_parallel = self.cls(h, self.bert.embeddings.word_embeddings.weight)
        seq_logits = self.seq_relationship(pooled)

        loss = None
        if labels is not None:
            per_token = vocab_parallel_cross_entropy(logits_parallel, labels.clamp(min=0))
            valid = (labels != -100)
            mlm_loss = (per_token * valid).sum() / valid.sum().clamp(min=1)
            loss = mlm_loss
            if next_sentence_label is not None:
                sop_loss = nn.functional.cross_entropy(
                    seq_logits.float().view(-1, 2), next_sentence_label.view(-1))
                loss = loss + sop_loss
        logits = logits_parallel
        if groups.get_tensor_model_parallel_world_size() > 1 and labels is None:
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return MegatronBertPreTrainingOutput(
            loss=loss, prediction_logits=logits,
            seq_relationship_logits=seq_logits)


class MegatronBertForMaskedLM(MegatronBertPreTrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.cls = MegatronBertLMHead(config)
        self.post_init()

    def gradient_checkpointing_enable(self, skip_interval: int = 0, **_kw):
        self.bert.gradient_checkpointing = True
        self.bert.gradient_checkpointing_skip_interval = max(skip_interval, 0)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, position_ids=None, return_dict=True, **_kw):
        out = self.bert(input_ids, attention_mask, token_type_ids,
                        position_ids)
        logits_parallel = self.cls(out.last_hidden_state,
                                   self.bert.embeddings.word_embeddings.weight)
        loss = None
        if labels is not None:
            per_token = vocab_parallel_cross_entropy(logits_parallel,
                                                     labels.clamp(min=0))
            valid = (labels != -100)
            loss = (per_token * valid).sum() / valid.sum().clamp(min=1)
        logits = logits_parallel
        if groups.get_tensor_model_parallel_world_size() > 1 and labels is None:
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return MaskedLMOutput(loss=loss, logits=logits)


class MegatronBertForSequenceClassification(MegatronBertPreTrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.num_labels = getattr(config, "num_labels", 2)
        self.bert = MegatronBertModel(config, add_pooling_layer=True)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.classifier = nn.Linear(config.hidden_size, self.num_labels)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, return_dict=True, **_kw):
        out = self.bert(input_ids, attention_mask, token_type_ids)
        pooled = torch.tanh(self.bert.pooler(out.last_hidden_state[:, 0]))
        logits = self.classifier(self.dropout(pooled))
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, self.num_labels), labels.view(-1))
        return SequenceClassifierOutput(loss=loss, logits=logits)


class MegatronBertForTokenClassification(MegatronBertPreTrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.num_labels = getattr(config, "num_labels", 2)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.classifier = nn.Linear(config.hidden_size, self.num_labels)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, return_dict=True, **_kw):
        out = self.bert(input_ids, attention_mask, token_type_ids)
        logits = self.classifier(self.dropout(out.last_hidden_state))
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, self.num_labels), labels.view(-1),
                ignore_index=-100)
        return TokenClassifierOutput(loss=loss, logits=logits)
