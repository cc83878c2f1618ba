"""Longformer: sliding-window + global attention for 4k-context Chinese NLU.

Behavioral parity: reference models/longformer/modeling_longformer.py
(_sliding_chunks_query_key_matmul :696-741 + global attention machinery) —
O(s·w) chunked band attention, global tokens attend/attended everywhere.
"""
from __future__ import annotations

import math
import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.modeling_outputs import BaseModelOutput, MaskedLMOutput

from fengshen_amd.models.layers import LayerNorm, ParallelMLP, init_normal, scaled_init_normal
from fengshen_amd.ops import functional as F_ops


class LongformerConfig(PretrainedConfig):
    model_type = "fengshen_longformer"

    def __init__(self, vocab_size: int = 21128, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 max_position_embeddings: int = 4096,
                 attention_window: int = 256, type_vocab_size: int = 2,
                 layer_norm_eps: float = 1e-12, initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1, attention_dropout: float = 0.1,
                 pad_token_id: int = 0, torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.attention_window = attention_window
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(pad_token_id=pad_token_id, torch_dtype=torch_dtype, **kw)


def longformer_tiny_config(**over):
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=512, attention_window=16)
    cfg.update(over)
    return LongformerConfig(**cfg)


class SlidingWindowSelfAttention(nn.Module):
    """Band attention: each token attends within +-w/2; tokens flagged global
    attend to (and are attended by) everything."""

    def __init__(self, config):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        self.window = config.attention_window
        self.qkv = nn.Linear(config.hidden_size, 3 * config.hidden_size)
        self.out = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(config.attention_dropout)

    def forward(self, x, attention_mask=None, global_mask=None):
        b, s, hdim = x.shape
        np_, hn = self.num_heads, self.head_dim
        w = min(self.window, s)
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        q = q.view(b, s, np_, hn).transpose(1, 2)
        k = k.view(b, s, np_, hn).transpose(1, 2)
        v = v.view(b, s, np_, hn).transpose(1, 2)
        scale = 1.0 / math.sqrt(hn)

        # band mask (|i-j| <= w/2) as a sparse-equivalent dense mask for the
        # bf16 fused-softmax path; O(s*w) chunked kernels arrive with the
        # long-context kernel pass
        idx = torch.arange(s, device=x.device)
        band = (idx[None, :] - idx[:, None]).abs() > (w // 2)
        mask = band[None, None, :, :]
        if global_mask is not None:
            gm = global_mask.bool()
            # global tokens: row and column fully visible
            mask = mask & ~gm[:, None, None, :]
            mask = mask & ~gm[:, None, :, None]
        if attention_mask is not None:
            mask = mask | (attention_mask == 0)[:, None, None, :]

        ctx = F_ops.attention(q, k, v, causal=False, mask=mask,
                              dropout_p=self.dropout.p,
                              training=self.training, scale=scale)
        ctx = ctx.transpose(1, 2).reshape(b, s, hdim)
        return self.out(ctx)


class LongformerLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.ln1 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.attn = SlidingWindowSelfAttention(config)
        self.ln2 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range,
                                config.num_hidden_layers)
        self.mlp = ParallelMLP(config.hidden_size, config.intermediate_size,
                               init_method=im, output_init_method=om)
        self.dropout = nn.Dropout(config.hidden_dropout)

    def forward(self, x, attention_mask=None, global_mask=None):
        h = self.attn(self.ln1(x), attention_mask, global_mask)
        x = x + self.dropout(h)
        h = self.mlp(self.ln2(x))
        return x + self.dropout(h)


class LongformerPreTrainedModel(PreTrainedModel):
    config_class = LongformerConfig
    base_model_prefix = "longformer"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()


class LongformerModel(LongformerPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.hidden_size)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.hidden_size)
        self.emb_ln = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.layers = nn.ModuleList(
            [LongformerLayer(config) for _ in range(config.num_hidden_layers)])
        self.post_init()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                global_attention_mask=None, **_kw):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        h = (self.word_embeddings(input_ids) + self.position_embeddings(pos)
             + self.token_type_embeddings(token_type_ids))
        h = self.dropout(self.emb_ln(h))
        for layer in self.layers:
            h = layer(h, attention_mask, global_attention_mask)
        return BaseModelOutput(last_hidden_state=h)


class LongformerForMaskedLM(LongformerPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.longformer = LongformerModel(config)
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.ln = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.decoder = nn.Linear(config.hidden_size, config.vocab_size)
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                global_attention_mask=None, labels=None, **_kw):
        h = self.longformer(input_ids, attention_mask, token_type_ids,
                            global_attention_mask).last_hidden_state
        h = self.ln(F_ops.eager_gelu(self.dense(h).float()).to(h.dtype))
        logits = self.decoder(h)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.float().view(-1, logits.shape[-1]), labels.view(-1),
                ignore_index=-100)
        return MaskedLMOutput(loss=loss, logits=logits)
