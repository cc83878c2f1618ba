"""Shared encoder-decoder transformer blocks (T5/Randeng, BART/Randeng-BART).

Behavioral parity: reference models/megatron_t5 (HF T5 with @IDEA
modifications: bias=True dense, nn.LayerNorm instead of T5LayerNorm,
absolute positions instead of relative bias — modeling_megatron_t5.py:261-917)
and models/bart.  Built on our parallel layer library + fused ops.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelAttention,
    ParallelMLP,
)
from fengshen_amd.ops import functional as F_ops
from fengshen_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
    divide,
)
from fengshen_amd.parallel import groups
import math


class ParallelCrossAttention(nn.Module):
    """Q from decoder states, K/V from encoder states (column-parallel heads,
    row-parallel output)."""

    def __init__(self, hidden_size: int, num_heads: int, *,
                 attention_dropout: float = 0.0, bias: bool = True,
                 init_method=None, output_init_method=None, dtype=None):
        super().__init__()
        tp = groups.get_tensor_model_parallel_world_size()
        self.num_heads_per_partition = divide(num_heads, tp)
        self.head_dim = divide(hidden_size, num_heads)
        self.norm_factor = 1.0 / math.sqrt(self.head_dim)
        self.attention_dropout = attention_dropout
        im = init_method or nn.init.xavier_normal_
        om = output_init_method or im
        self.q_proj = ColumnParallelLinear(hidden_size, hidden_size, bias=bias,
                                           gather_output=False, init_method=im,
                                           dtype=dtype)
        from fengshen_amd.models.layers import MergedColumnParallelLinear
        self.kv_proj = MergedColumnParallelLinear(
            hidden_size, [hidden_size, hidden_size], bias=bias,
            init_method=im, dtype=dtype)
        self.out_proj = RowParallelLinear(hidden_size, hidden_size, bias=bias,
                                          input_is_parallel=True,
                                          init_method=om, dtype=dtype)

    def forward(self, x: torch.Tensor, encoder_states: torch.Tensor,
                encoder_mask: Optional[torch.Tensor] = None):
        b, sq, _ = x.shape
        sk = encoder_states.shape[1]
        np_ = self.num_heads_per_partition
        hn = self.head_dim
        q = self.q_proj(x)
        if isinstance(q, tuple):
            q = q[0]
        kv = self.kv_proj(encoder_states)
        k, v = kv.chunk(2, dim=-1)
        q = q.view(b, sq, np_, hn).transpose(1, 2)
        k = k.view(b, sk, np_, hn).transpose(1, 2)
        v = v.view(b, sk, np_, hn).transpose(1, 2)
        ctx = F_ops.attention(q, k, v, causal=False, mask=encoder_mask,
                              dropout_p=self.attention_dropout,
                              training=self.training, scale=self.norm_factor)
        ctx = ctx.transpose(1, 2).reshape(b, sq, np_ * hn)
        out = self.out_proj(ctx)
        return out[0] if isinstance(out, tuple) else out


class EncoderLayer(nn.Module):
    def __init__(self, hidden_size: int, num_heads: int, ffn_hidden_size: int,
                 *, norm_eps: float = 1e-5, prenorm: bool = True,
                 attention_dropout: float = 0.0, hidden_dropout: float = 0.0,
                 init_method=None, output_init_method=None, layer_idx: int = 0):
        super().__init__()
        self.prenorm = prenorm
        self.hidden_dropout = hidden_dropout
        self.ln1 = LayerNorm(hidden_size, eps=norm_eps)
        self.attn = ParallelAttention(
            hidden_size, num_heads, causal=False,
            attention_dropout=attention_dropout, hidden_dropout=hidden_dropout,
            init_method=init_method, output_init_method=output_init_method,
            layer_idx=layer_idx)
        self.ln2 = LayerNorm(hidden_size, eps=norm_eps)
        self.mlp = ParallelMLP(hidden_size, ffn_hidden_size,
                               init_method=init_method,
                               output_init_method=output_init_method)

    def forward(self, x, attention_mask=None):
        if self.prenorm:
            h = self.attn(self.ln1(x), attention_mask=attention_mask)
            x = F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                       self.training)
            h = self.mlp(self.ln2(x))
            return F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                          self.training)
        h = self.attn(x, attention_mask=attention_mask)
        x = self.ln1(F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                            self.training))
        h = self.mlp(x)
        return self.ln2(F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                               self.training))


class DecoderLayer(nn.Module):
    def __init__(self, hidden_size: int, num_heads: int, ffn_hidden_size: int,
                 *, norm_eps: float = 1e-5, prenorm: bool = True,
                 attention_dropout: float = 0.0, hidden_dropout: float = 0.0,
                 init_method=None, output_init_method=None, layer_idx: int = 0):
        super().__init__()
        self.prenorm = prenorm
        self.hidden_dropout = hidden_dropout
        self.ln1 = LayerNorm(hidden_size, eps=norm_eps)
        self.self_attn = ParallelAttention(
            hidden_size, num_heads, causal=True,
            attention_dropout=attention_dropout, hidden_dropout=hidden_dropout,
            init_method=init_method, output_init_method=output_init_method,
            layer_idx=layer_idx)
        self.ln2 = LayerNorm(hidden_size, eps=norm_eps)
        self.cross_attn = ParallelCrossAttention(
            hidden_size, num_heads, attention_dropout=attention_dropout,
            init_method=init_method, output_init_method=output_init_method)
        self.ln3 = LayerNorm(hidden_size, eps=norm_eps)
        self.mlp = ParallelMLP(hidden_size, ffn_hidden_size,
                               init_method=init_method,
                               output_init_method=output_init_method)

    def forward(self, x, encoder_states, self_mask=None, cross_mask=None,
                cache=None):
        if self.prenorm:
            h = self.self_attn(self.ln1(x), attention_mask=self_mask,
                               cache=cache)
            x = F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                       self.training)
            h = self.cross_attn(self.ln2(x), encoder_states,
                                encoder_mask=cross_mask)
            x = F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                       self.training)
            h = self.mlp(self.ln3(x))
            return F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                          self.training)
        h = self.self_attn(x, attention_mask=self_mask, cache=cache)
        x = self.ln1(F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                            self.training))
        h = self.cross_attn(x, encoder_states, encoder_mask=cross_mask)
        x = self.ln2(F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                            self.training))
        h = self.mlp(x)
        return self.ln3(F_ops.bias_dropout_add(h, None, x, self.hidden_dropout,
                                               self.training))
