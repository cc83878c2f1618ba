"""Zhouwenwang-RoFormer: BERT encoder with rotary position embeddings.

Behavioral parity: reference models/roformer/modeling_roformer.py
(RoPEmbedding applied to q/k inside self-attention :206-257) — here rotary
is native to our ParallelAttention (the same RoPE HIP kernel LLaMA uses).
"""
from __future__ import annotations

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import (
    BaseModelOutput,
    MaskedLMOutput,
    SequenceClassifierOutput,
)

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


class RoFormerConfig(PretrainedConfig):
    model_type = "fengshen_roformer"

    def __init__(self, vocab_size: int = 21128, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 max_position_embeddings: int = 1024,
                 type_vocab_size: int = 2, layer_norm_eps: float = 1e-12,
                 initializer_range: float = 0.02, hidden_dropout: float = 0.1,
                 attention_dropout: float = 0.1, rotary_emb_base: float = 10000.0,
                 pad_token_id: int = 0, torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        self.rotary_emb_base = rotary_emb_base
        super().__init__(pad_token_id=pad_token_id, torch_dtype=torch_dtype, **kw)


def roformer_tiny_config(**over):
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=128)
    cfg.update(over)
    return RoFormerConfig(**cfg)


class RoFormerPreTrainedModel(PreTrainedModel):
    config_class = RoFormerConfig
    base_model_prefix = "roformer"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()


class RoFormerModel(RoFormerPreTrainedModel):
    def __init__(self, config: RoFormerConfig, add_pooling_layer: bool = True):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range,
                                config.num_hidden_layers)
        self.word_embeddings = VocabParallelEmbedding(
            config.vocab_size, config.hidden_size, init_method=im)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.hidden_size)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.encoder = nn.ModuleList([
            ParallelTransformerLayer(
                config.hidden_size, config.num_attention_heads, causal=False,
                norm="layernorm", norm_eps=config.layer_norm_eps,
                mlp_type="gelu", ffn_hidden_size=config.intermediate_size,
                rotary=True, rope_base=config.rotary_emb_base,
                max_positions=config.max_position_embeddings,
                attention_dropout=config.attention_dropout,
                hidden_dropout=config.hidden_dropout,
                bias=True, init_method=im, output_init_method=om, layer_idx=i)
            for i in range(config.num_hidden_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size) \
            if add_pooling_layer else None
        self.post_init()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                **_kw):
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        h = self.word_embeddings(input_ids) \
            + self.token_type_embeddings(token_type_ids)
        h = self.dropout(h)
        mask = (attention_mask == 0)[:, None, None, :] \
            if attention_mask is not None else None
        for layer in self.encoder:
            h = layer(h, attention_mask=mask)
        h = self.ln_f(h)
        return BaseModelOutput(last_hidden_state=h)


class RoFormerForMaskedLM(RoFormerPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.roformer = RoFormerModel(config, add_pooling_layer=False)
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.ln = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.bias = nn.Parameter(torch.zeros(config.vocab_size))
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, **_kw):
        h = self.roformer(input_ids, attention_mask,
                          token_type_ids).last_hidden_state
        h = self.ln(F_ops.eager_gelu(self.dense(h).float()).to(h.dtype))
        logits_parallel = parallel_lm_logits(
            h, self.roformer.word_embeddings.weight, parallel_output=True)
        tp = groups.get_tensor_model_parallel_world_size()
        if tp > 1:
            rank = groups.get_tensor_model_parallel_rank()
            per = self.bias.shape[0] // tp
            logits_parallel = logits_parallel + self.bias[rank * per:(rank + 1) * per]
        else:
            logits_parallel = logits_parallel + self.bias
        loss = None
        if labels is not None:
            per_token = vocab_parallel_cross_entropy(logits_parallel,
                                                     labels.clamp(min=0))
            valid = (labels != -100)
            loss = (per_token * valid).sum() / valid.sum().clamp(min=1)
        logits = logits_parallel
        if tp > 1 and labels is None:
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return MaskedLMOutput(loss=loss, logits=logits)


class RoFormerForSequenceClassification(RoFormerPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.num_labels = getattr(config, "num_labels", 2)
        self.roformer = RoFormerModel(config, add_pooling_layer=True)
        self.classifier = nn.Linear(config.hidden_size, self.num_labels)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, **_kw):
        h = self.roformer(input_ids, attention_mask,
                          token_type_ids).last_hidden_state
        pooled = torch.tanh(self.roformer.pooler(h[:, 0]))
        logits = self.classifier(pooled)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, self.num_labels), labels.view(-1))
        return SequenceClassifierOutput(loss=loss, logits=logits)
