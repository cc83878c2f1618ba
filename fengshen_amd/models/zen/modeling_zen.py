"""ZEN: n-gram-enhanced Chinese BERT (dual-stream encoder).

Behavioral parity: reference models/zen1/modeling.py:416-450 — token stream
through layer[i], n-gram stream through word_layers[i] for the first
num_hidden_word_layers, fused each layer via
hidden += bmm(ngram_position_matrix, ngram_hidden); n-gram vocab from
ngram.txt with per-sequence cap max_ngram_in_seq=128
(models/zen1/ngram_utils.py:28-54).  ZEN2 heads are the standard task heads
over the same encoder.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import (
    BaseModelOutput,
    QuestionAnsweringModelOutput,
    SequenceClassifierOutput,
    TokenClassifierOutput,
)

from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelTransformerLayer,
    init_normal,
    scaled_init_normal,
)


class ZenConfig(PretrainedConfig):
    model_type = "fengshen_zen"

    def __init__(self, vocab_size: int = 21128, word_vocab_size: int = 104089,
                 hidden_size: int = 768, num_hidden_layers: int = 12,
                 num_hidden_word_layers: int = 6, num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 max_position_embeddings: int = 512,
                 max_ngram_in_seq: int = 128, type_vocab_size: int = 2,
                 layer_norm_eps: float = 1e-12, initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1, attention_dropout: float = 0.1,
                 pad_token_id: int = 0, torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.word_vocab_size = word_vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_hidden_word_layers = num_hidden_word_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.max_ngram_in_seq = max_ngram_in_seq
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(pad_token_id=pad_token_id, torch_dtype=torch_dtype, **kw)


def zen_tiny_config(**over):
    cfg = dict(vocab_size=256, word_vocab_size=512, hidden_size=64,
               num_hidden_layers=2, num_hidden_word_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=128, max_ngram_in_seq=16)
    cfg.update(over)
    return ZenConfig(**cfg)


class ZenNgramDict:
    """n-gram lexicon + matcher (ref models/zen1/ngram_utils.py:28-54)."""

    def __init__(self, ngram_list: Optional[List[str]] = None,
                 path: Optional[str] = None, max_ngram_in_seq: int = 128):
        self.max_ngram_in_seq = max_ngram_in_seq
        if path is not None and os.path.exists(path):
            with open(path, encoding="utf-8") as f:
                ngram_list = [ln.strip() for ln in f if ln.strip()]
        self.id_to_ngram = ["[pad]"] + list(ngram_list or [])
        self.ngram_to_id = {g: i for i, g in enumerate(self.id_to_ngram)}
        self.max_ngram_len = max((len(g) for g in self.id_to_ngram), default=1)

    def match(self, chars: List[str]) -> List[Tuple[int, int, int]]:
        """-> [(ngram_id, start, length)] capped at max_ngram_in_seq."""
        out = []
        for i in range(len(chars)):
            for n in range(2, self.max_ngram_len + 1):
                if i + n > len(chars):
                    break
                g = "".join(chars[i:i + n])
                gid = self.ngram_to_id.get(g)
                if gid:
                    out.append((gid, i, n))
                    if len(out) >= self.max_ngram_in_seq:
                        return out
        return out


class ZenPreTrainedModel(PreTrainedModel):
    config_class = ZenConfig
    base_model_prefix = "zen"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()


class ZenModel(ZenPreTrainedModel):
    def __init__(self, config: ZenConfig):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range,
                                config.num_hidden_layers)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.hidden_size)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.hidden_size)
        self.ngram_embeddings = nn.Embedding(config.word_vocab_size,
                                             config.hidden_size)
        self.emb_ln = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout)

        def mk_layer(i):
            return ParallelTransformerLayer(
                config.hidden_size, config.num_attention_heads, causal=False,
                norm="layernorm", norm_eps=config.layer_norm_eps,
                mlp_type="gelu", ffn_hidden_size=config.intermediate_size,
                attention_dropout=config.attention_dropout,
                hidden_dropout=config.hidden_dropout,
                bias=True, init_method=im, output_init_method=om, layer_idx=i)

        self.layers = nn.ModuleList(
            [mk_layer(i) for i in range(config.num_hidden_layers)])
        self.word_layers = nn.ModuleList(
            [mk_layer(i) for i in range(config.num_hidden_word_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size)
        self.post_init()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, ngram_ids=None, ngram_position_matrix=None,
                attention_mask=None, token_type_ids=None,
                ngram_attention_mask=None, **_kw):
        """ngram_ids [b, max_ngram]; ngram_position_matrix [b, s, max_ngram]
        (1 where the ngram covers the token position)."""
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        h = (self.word_embeddings(input_ids) + self.position_embeddings(pos)
             + self.token_type_embeddings(token_type_ids))
        h = self.dropout(self.emb_ln(h))
        mask = (attention_mask == 0)[:, None, None, :] \
            if attention_mask is not None else None

        gh = None
        gmask = None
        if ngram_ids is not None:
            gh = self.ngram_embeddings(ngram_ids)
            if ngram_attention_mask is not None:
                gmask = (ngram_attention_mask == 0)[:, None, None, :]

        for i, layer in enumerate(self.layers):
            h = layer(h, attention_mask=mask)
            if gh is not None and i < len(self.word_layers):
                gh = self.word_layers[i](gh, attention_mask=gmask)
                # fuse: token i accumulates every covering ngram's state
                # (ref modeling.py:416-450)
                h = h + torch.bmm(
                    ngram_position_matrix.to(gh.dtype), gh)
        h = self.ln_f(h)
        return BaseModelOutput(last_hidden_state=h)


class ZenForSequenceClassification(ZenPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.num_labels = getattr(config, "num_labels", 2)
        self.zen = ZenModel(config)
        self.classifier = nn.Linear(config.hidden_size, self.num_labels)
        self.post_init()

    def forward(self, input_ids, ngram_ids=None, ngram_position_matrix=None,
                attention_mask=None, token_type_ids=None, labels=None, **_kw):
        h = self.zen(input_ids, ngram_ids, ngram_position_matrix,
                     attention_mask, token_type_ids).last_hidden_state
        pooled = torch.tanh(self.zen.pooler(h[:, 0]))
        logits = self.classifier(pooled)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, self.num_labels), labels.view(-1))
        return SequenceClassifierOutput(loss=loss, logits=logits)


class ZenForTokenClassification(ZenPreTrainedModel):
    """Token-level head (the zen2_finetune NER tasks; ref
    modeling.py token classification head)."""

    def __init__(self, config):
        super().__init__(config)
        self.num_labels = getattr(config, "num_labels", 2)
        self.zen = ZenModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.classifier = nn.Linear(config.hidden_size, self.num_labels)
        self.post_init()

    def forward(self, input_ids, ngram_ids=None, ngram_position_matrix=None,
                attention_mask=None, token_type_ids=None, labels=None, **_kw):
        h = self.zen(input_ids, ngram_ids, ngram_position_matrix,
                     attention_mask, token_type_ids).last_hidden_state
        logits = self.classifier(self.dropout(h))
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, self.num_labels), labels.view(-1),
                ignore_index=-100)
        return TokenClassifierOutput(loss=loss, logits=logits)


class ZenForQuestionAnswering(ZenPreTrainedModel):
    """Span QA head — ZEN2's addition (ref zen2/modeling.py:1291)."""

    def __init__(self, config):
        super().__init__(config)
        self.zen = ZenModel(config)
        self.qa_outputs = nn.Linear(config.hidden_size, 2)
        self.post_init()

    def forward(self, input_ids, ngram_ids=None, ngram_position_matrix=None,
                attention_mask=None, token_type_ids=None,
                start_positions=None, end_positions=None, **_kw):
        h = self.zen(input_ids, ngram_ids, ngram_position_matrix,
                     attention_mask, token_type_ids).last_hidden_state
        start_logits, end_logits = self.qa_outputs(h).split(1, dim=-1)
        start_logits = start_logits.squeeze(-1)
        end_logits = end_logits.squeeze(-1)
        loss = None
        if start_positions is not None and end_positions is not None:
            s_loss = nn.functional.cross_entropy(
                start_logits.float(), start_positions.clamp(
                    0, start_logits.shape[1] - 1))
            e_loss = nn.functional.cross_entropy(
                end_logits.float(), end_positions.clamp(
                    0, end_logits.shape[1] - 1))
            loss = (s_loss + e_loss) / 2
        return QuestionAnsweringModelOutput(
            loss=loss, start_logits=start_logits, end_logits=end_logits)
