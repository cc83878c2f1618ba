"""Bigan/Transfo-XL denoise family: GPT2-style blocks + segment recurrence.

Behavioral parity: reference models/transfo_xl_denoise/
modeling_transfo_xl_denoise.py:168-477 — causal transformer with cached
segment memories (mems) concatenated to K/V each layer; used for the
denoise / paraphrase / reasoning variants.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput

from fengshen_amd.models.layers import LayerNorm, ParallelMLP, init_normal, scaled_init_normal
from fengshen_amd.ops import functional as F_ops
from dataclasses import dataclass


class TransfoXLDenoiseConfig(PretrainedConfig):
    model_type = "fengshen_transfo_xl_denoise"

    def __init__(self, vocab_size: int = 50048, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072, mem_len: int = 256,
                 max_position_embeddings: int = 1024,
                 layer_norm_eps: float = 1e-5, initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1, attention_dropout: float = 0.1,
                 torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.mem_len = mem_len
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(torch_dtype=torch_dtype, **kw)


def transfo_xl_tiny_config(**over):
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128, mem_len=32,
               max_position_embeddings=128)
    cfg.update(over)
    return TransfoXLDenoiseConfig(**cfg)


class _XlAttention(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        self.qkv = nn.Linear(config.hidden_size, 3 * config.hidden_size)
        self.out = nn.Linear(config.hidden_size, config.hidden_size)
        self.attention_dropout = config.attention_dropout

    def forward(self, x, mem: Optional[torch.Tensor] = None):
        b, s, hdim = x.shape
        np_, hn = self.num_heads, self.head_dim
        src = x if mem is None else torch.cat([mem, x], dim=1)
        q = self.qkv(x).chunk(3, dim=-1)[0]
        _, k, v = self.qkv(src).chunk(3, dim=-1)
        m = src.shape[1] - s
        q = q.view(b, s, np_, hn).transpose(1, 2)
        k = k.view(b, s + m, np_, hn).transpose(1, 2)
        v = v.view(b, s + m, np_, hn).transpose(1, 2)
        # causal over the query segment; memory fully visible
        idx_q = torch.arange(s, device=x.device)[:, None]
        idx_k = torch.arange(s + m, device=x.device)[None, :]
        mask = (idx_k - m) > idx_q  # True = masked
        ctx = F_ops.attention(q, k, v, causal=False,
                              mask=mask[None, None, :, :],
                              dropout_p=self.attention_dropout,
                              training=self.training,
                              scale=1.0 / math.sqrt(hn))
        ctx = ctx.transpose(1, 2).reshape(b, s, hdim)
        return self.out(ctx)


class _XlLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.ln1 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.attn = _XlAttention(config)
        self.ln2 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range,
                                config.num_hidden_layers)
        self.mlp = ParallelMLP(config.hidden_size, config.intermediate_size,
                               init_method=im, output_init_method=om)

    def forward(self, x, mem=None):
        # memory stores pre-LN hiddens; normalize at use (ref :460)
        m = self.ln1(mem) if mem is not None else None
        x = x + self.attn(self.ln1(x), m)
        return x + self.mlp(self.ln2(x))


@dataclass
class TransfoXLOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None
    mems: Optional[List[torch.Tensor]] = None


class TransfoXLDenoiseModel(PreTrainedModel):
    config_class = TransfoXLDenoiseConfig

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(module, nn.Linear) and module.bias is not None:
                module.bias.data.zero_()

    def __init__(self, config):
        super().__init__(config)
        self.wte = nn.Embedding(config.vocab_size, config.hidden_size)
        self.wpe = nn.Embedding(config.max_position_embeddings,
                                config.hidden_size)
        self.layers = nn.ModuleList(
            [_XlLayer(config) for _ in range(config.num_hidden_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.mem_len = config.mem_len
        self.post_init()

    def get_input_embeddings(self):
        return self.wte

    def forward(self, input_ids, mems: Optional[List[torch.Tensor]] = None,
                labels=None, **_kw):
        b, s = input_ids.shape
        mem_len = mems[0].shape[1] if mems else 0
        pos = torch.arange(mem_len, mem_len + s,
                           device=input_ids.device).clamp(
            max=self.config.max_position_embeddings - 1).unsqueeze(0)
        h = self.wte(input_ids) + self.wpe(pos)
        new_mems = []
        for i, layer in enumerate(self.layers):
            old = mems[i] if mems else None
            # accumulate memory across segments (ref update_mems :649-662):
            # concat old memory with this segment, keep the last mem_len
            if old is not None:
                new_mems.append(
                    torch.cat([old, h.detach()], dim=1)[:, -self.mem_len:])
            else:
                new_mems.append(h.detach()[:, -self.mem_len:])
            h = layer(h, old)
        h = self.ln_f(h)
        logits = h @ self.wte.weight.t().to(h.dtype)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits[:, :-1].float().reshape(-1, logits.shape[-1]),
                labels[:, 1:].reshape(-1), ignore_index=-100)
        return TransfoXLOutput(loss=loss, logits=logits, mems=new_mems)
