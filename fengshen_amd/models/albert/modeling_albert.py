"""ALBERT: factorized embeddings + cross-layer parameter sharing.

Behavioral parity: reference models/albert (1,363 LoC HF-style ALBERT) —
embedding_size < hidden_size with a projection, ONE shared transformer layer
applied num_hidden_layers times, MLM + SOP heads.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import BaseModelOutput, MaskedLMOutput

from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelTransformerLayer,
    init_normal,
)
from fengshen_amd.ops import functional as F_ops


class AlbertConfig(PretrainedConfig):
    model_type = "fengshen_albert"

    def __init__(self, vocab_size: int = 21128, embedding_size: int = 128,
                 hidden_size: int = 768, num_hidden_layers: int = 12,
                 num_attention_heads: int = 12, intermediate_size: int = 3072,
                 max_position_embeddings: int = 512, type_vocab_size: int = 2,
                 layer_norm_eps: float = 1e-12, initializer_range: float = 0.02,
                 hidden_dropout: float = 0.0, attention_dropout: float = 0.0,
                 pad_token_id: int = 0, torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.embedding_size = embedding_size
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
        super().__init__(pad_token_id=pad_token_id, torch_dtype=torch_dtype, **kw)


def albert_tiny_config(**over):
    cfg = dict(vocab_size=256, embedding_size=32, hidden_size=64,
               num_hidden_layers=3, num_attention_heads=4,
               intermediate_size=128, max_position_embeddings=128)
    cfg.update(over)
    return AlbertConfig(**cfg)


class AlbertPreTrainedModel(PreTrainedModel):
    config_class = AlbertConfig
    base_model_prefix = "albert"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()


class AlbertModel(AlbertPreTrainedModel):
    def __init__(self, config: AlbertConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.embedding_size)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.embedding_size)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.embedding_size)
        self.emb_ln = LayerNorm(config.embedding_size,
                                eps=config.layer_norm_eps)
        self.embedding_projection = nn.Linear(config.embedding_size,
                                              config.hidden_size)
        im = init_normal(config.initializer_range)
        # ONE shared layer (the ALBERT trick)
        self.shared_layer = ParallelTransformerLayer(
            config.hidden_size, config.num_attention_heads, causal=False,
            norm="layernorm", norm_eps=config.layer_norm_eps, mlp_type="gelu",
            ffn_hidden_size=config.intermediate_size,
            attention_dropout=config.attention_dropout,
            hidden_dropout=config.hidden_dropout, bias=True, init_method=im)
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size)
        self.post_init()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                **_kw):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        e = (self.word_embeddings(input_ids) + self.position_embeddings(pos)
             + self.token_type_embeddings(token_type_ids))
        h = self.embedding_projection(self.emb_ln(e))
        mask = (attention_mask == 0)[:, None, None, :] \
            if attention_mask is not None else None
        for _ in range(self.config.num_hidden_layers):
            h = self.shared_layer(h, attention_mask=mask)
        return BaseModelOutput(last_hidden_state=self.ln_f(h))


class AlbertForMaskedLM(AlbertPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.albert = AlbertModel(config)
        self.dense = nn.Linear(config.hidden_size, config.embedding_size)
        self.ln = LayerNorm(config.embedding_size, eps=config.layer_norm_eps)
        self.decoder = nn.Linear(config.embedding_size, config.vocab_size)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, **_kw):
        h = self.albert(input_ids, attention_mask,
                        token_type_ids).last_hidden_state
        h = self.ln(F_ops.eager_gelu(self.dense(h).float()).to(h.dtype))
        logits = self.decoder(h)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, logits.shape[-1]), labels.view(-1),
                ignore_index=-100)
        return MaskedLMOutput(loss=loss, logits=logits)
