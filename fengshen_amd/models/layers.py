"""Parallel transformer layer library (the reference's L3).

Behavioral parity: fengshen/models/megatron/layers/transformer.py
(ParallelSelfAttention :175, ParallelMLP :70, LLaMAParallelMLP :571,
ParallelTransformerLayer :626, parallel_lm_logits :800) — redesigned:
  * [b, s, h] layout end to end (HF-style), no [s,b,h] transposes
  * fused QKV / fused gate|up projections as merged column-parallel GEMMs
    (one hipBLASLt launch instead of 3)
  * all hot elementwise/softmax/norm ops route through fengshen_amd.ops
    (HIP kernels on GPU, eager oracles on CPU)
  * KV cache for generation
"""
from __future__ import annotations

import math
from typing import Callable, List, Optional

import torch
import torch.nn as nn
import torch.nn.init as init

from fengshen_amd.ops import functional as F_ops
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
    _init_partition,
    divide,
)
from fengshen_amd.parallel.mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
)


def init_normal(std: float) -> Callable:
    def fn(tensor):
        return init.normal_(tensor, mean=0.0, std=std)
    return fn


def scaled_init_normal(std: float, num_layers: int) -> Callable:
    """output-layer init scaled by 1/sqrt(2N) (megatron lineage,
    ref layers/init_functions.py)."""
    return init_normal(std / math.sqrt(2.0 * num_layers))


def small_init(dim: int) -> Callable:
    """Nguyen & Salazar transformers-without-tears init: N(0, sqrt(2/5d))
    (ref layers/init_functions.py small_init_init_method)."""
    return init_normal(math.sqrt(2.0 / (5.0 * dim)))


def wang_init(dim: int, num_layers: int) -> Callable:
    """Ben Wang's GPT-J output projection init: std = 2/(L*sqrt(d))
    (ref layers/init_functions.py wang_init_method)."""
    return init_normal(2.0 / (num_layers * math.sqrt(dim)))


class MergedColumnParallelLinear(ColumnParallelLinear):
    """Column-parallel GEMM whose output is a concat of logical segments
    (QKV or gate|up), arranged so each TP rank holds
    [seg0_shard; seg1_shard; ...] — local chunk() recovers the segments."""

    def __init__(self, input_size: int, segment_sizes: List[int], bias: bool = True,
                 init_method: Callable = init.xavier_normal_, dtype=None):
        self.segment_sizes = list(segment_sizes)
        super().__init__(input_size, sum(segment_sizes), bias=bias,
                         gather_output=False, init_method=init_method, dtype=dtype)
        # re-init weights segment-wise so each shard slices its own segment
        tp = groups.get_tensor_model_parallel_world_size()
        if tp > 1:
            off = 0
            for seg in self.segment_sizes:
                seg_per = divide(seg, tp)
                shard = self.weight[off:off + seg_per]
                _init_partition(shard, init_method,
                                full_shape=(seg, input_size), partition_dim=0)
                off += seg_per


class RMSNorm(nn.Module):
    """fp32-variance RMSNorm (ref norms.py:35-52) via HIP kernel."""

    def __init__(self, dim: int, eps: float = 1e-6, dtype=None):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return F_ops.rms_norm(x, self.weight, self.eps)


class LayerNorm(nn.Module):
    """Fused LayerNorm (the op the reference left vestigial in
    layer_norm_cuda.cpp)."""

    def __init__(self, dim: int, eps: float = 1e-5, dtype=None):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(dim, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return F_ops.layer_norm(x, self.weight, self.bias, self.eps)


class ScaleNorm(nn.Module):
    """Single learned scale over the l2-normalised vector
    (ref norms.py:55 ScaleNorm): y = g * x / max(||x||/sqrt(d), eps)."""

    def __init__(self, dim: int, eps: float = 1e-8, dtype=None):
        super().__init__()
        self.g = nn.Parameter(torch.ones(1, dtype=dtype))
        self.dim = dim
        self.eps = eps

    def forward(self, x):
        n = x.float().norm(dim=-1, keepdim=True) * self.dim ** -0.5
        return (x.float() / n.clamp(min=self.eps)).to(x.dtype) * self.g


def get_norm(kind: str, dim: int, eps: float, dtype=None) -> nn.Module:
    if kind == "rmsnorm":
        return RMSNorm(dim, eps, dtype)
    if kind == "layernorm":
        return LayerNorm(dim, eps, dtype)
    if kind == "scalenorm":
        return ScaleNorm(dim, dtype=dtype)
    raise ValueError(kind)


class ParallelAttention(nn.Module):
    """Multi-head attention with fused QKV column-parallel in, row-parallel
    out (ref ParallelSelfAttention, transformer.py:175-474)."""

    def __init__(self, hidden_size: int, num_heads: int, *, causal: bool,
                 rotary: bool = False, rope_base: float = 10000.0,
                 max_positions: int = 4096,
                 attention_dropout: float = 0.0, hidden_dropout: float = 0.0,
                 bias: bool = True, init_method=None, output_init_method=None,
                 dtype=None, num_kv_heads: Optional[int] = None,
                 layer_idx: int = 0, reduce_output: bool = True):
        super().__init__()
        self.layer_idx = layer_idx
        self.reduce_output = reduce_output
        tp = groups.get_tensor_model_parallel_world_size()
        self.hidden_size = hidden_size
        self.num_heads = num_heads
        self.num_kv_heads = num_kv_heads or num_heads
        assert self.num_kv_heads == num_heads, "GQA arrives with kernel pass 2"
        self.head_dim = divide(hidden_size, num_heads)
        self.num_heads_per_partition = divide(num_heads, tp)
        self.causal = causal
        self.rotary = rotary
        self.attention_dropout = attention_dropout
        self.hidden_dropout = hidden_dropout
        self.norm_factor = 1.0 / math.sqrt(self.head_dim)
        init_method = init_method or init.xavier_normal_
        output_init_method = output_init_method or init_method

        self.qkv_proj = MergedColumnParallelLinear(
            hidden_size, [hidden_size, hidden_size, hidden_size], bias=bias,
            init_method=init_method, dtype=dtype)
        self.out_proj = RowParallelLinear(
            hidden_size, hidden_size, bias=bias, input_is_parallel=True,
            init_method=output_init_method, dtype=dtype,
            reduce_output=reduce_output)

        if rotary:
            # lazy host-precomputed cos/sin tables (guide App. B: no on-device
            # trig); plain attrs, NOT buffers — HF meta-device from_pretrained
            # would materialize non-persistent buffers as garbage
            self.rope_base = rope_base
            self.max_positions = max_positions
            self._rope_cache = None

    def _rope_tables(self, device, seq_needed: int):
        cache = self._rope_cache
        if (cache is None or cache[0].device != device
                or cache[0].shape[0] < seq_needed):
            n = max(self.max_positions, seq_needed)
            cos, sin = F_ops.build_rope_cache(n, self.head_dim,
                                              base=self.rope_base)
            cache = (cos.to(device), sin.to(device))
            self._rope_cache = cache
        return cache

    def forward(self, x: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None,
                cache=None):
        """cache: any object with .get_seq_length(layer_idx) and
        .update(k, v, layer_idx) -> (k_full, v_full) — HF DynamicCache works."""
        b, sq, _ = x.shape
        np_ = self.num_heads_per_partition
        hn = self.head_dim

        qkv = self.qkv_proj(x)  # [b, sq, 3h/tp]
        q, k, v = qkv.chunk(3, dim=-1)
        # [b, s, np, hn] -> [b, np, s, hn]
        q = q.view(b, sq, np_, hn).transpose(1, 2)
        k = k.view(b, sq, np_, hn).transpose(1, 2)
        v = v.view(b, sq, np_, hn).transpose(1, 2)

        offset = 0
        if cache is not None:
            offset = cache.get_seq_length(self.layer_idx)
        if self.rotary:
            cos, sin = self._rope_tables(q.device, offset + sq)
            q, k = F_ops.apply_rotary(q, k, cos, sin, offset=offset)
        if cache is not None:
            k, v = cache.update(k, v, self.layer_idx)

        causal = self.causal
        if causal and offset > 0 and q.shape[-2] == 1:
            causal = False  # single-token decode: all past is visible
        # chunked prefill (offset>0, sq>1) stays causal: functional.attention
        # builds the offset-aware rectangular mask (triu(1+sk-sq)).
        ctx = F_ops.attention(q, k, v, causal=causal, mask=attention_mask,
                              dropout_p=self.attention_dropout,
                              training=self.training, scale=self.norm_factor)
        ctx = ctx.transpose(1, 2).reshape(b, sq, np_ * hn)
        out = self.out_proj(ctx)
        if isinstance(out, tuple) and self.reduce_output:
            out = out[0]
        return out  # (partial, bias) when reduce_output=False


class ParallelMLP(nn.Module):
    """h -> 4h -> h with fused bias-GELU (ref ParallelMLP, transformer.py:70)."""

    def __init__(self, hidden_size: int, ffn_hidden_size: Optional[int] = None,
                 bias: bool = True, init_method=None, output_init_method=None,
                 dtype=None, reduce_output: bool = True):
        super().__init__()
        ffn = ffn_hidden_size or 4 * hidden_size
        init_method = init_method or init.xavier_normal_
        self.reduce_output = reduce_output
        self.fc_in = ColumnParallelLinear(
            hidden_size, ffn, bias=bias, gather_output=False,
            init_method=init_method, skip_bias_add=bias, dtype=dtype)
        self.fc_out = RowParallelLinear(
            ffn, hidden_size, bias=bias, input_is_parallel=True,
            init_method=output_init_method or init_method, dtype=dtype,
            reduce_output=reduce_output)

    def forward(self, x):
        h = self.fc_in(x)
        if isinstance(h, tuple):
            h, bias = h
            h = F_ops.bias_gelu(h, bias)
        else:
            h = F_ops.eager_gelu(h.float()).to(h.dtype)
        out = self.fc_out(h)
        if isinstance(out, tuple) and self.reduce_output:
            out = out[0]
        return out  # (partial, bias) when reduce_output=False


class LLaMAParallelMLP(nn.Module):
    """SwiGLU MLP: fused [gate|up] column GEMM -> swiglu kernel -> row GEMM
    (ref LLaMAParallelMLP, transformer.py:571-623: w1/w3 column, w2 row)."""

    def __init__(self, hidden_size: int, intermediate_size: Optional[int] = None,
                 multiple_of: int = 256, init_method=None,
                 output_init_method=None, dtype=None,
                 reduce_output: bool = True):
        super().__init__()
        self.reduce_output = reduce_output
        if intermediate_size is None:
            # ref :589-590 rounding
            intermediate_size = int(2 * (4 * hidden_size) / 3)
            intermediate_size = multiple_of * (
                (intermediate_size + multiple_of - 1) // multiple_of)
        self.intermediate_size = intermediate_size
        init_method = init_method or init.xavier_normal_
        self.gate_up_proj = MergedColumnParallelLinear(
            hidden_size, [intermediate_size, intermediate_size], bias=False,
            init_method=init_method, dtype=dtype)
        self.down_proj = RowParallelLinear(
            intermediate_size, hidden_size, bias=False, input_is_parallel=True,
            init_method=output_init_method or init_method, dtype=dtype,
            reduce_output=reduce_output)

    def forward(self, x):
        packed = self.gate_up_proj(x)
        h = F_ops.swiglu(packed)
        return self.down_proj(h)  # (partial, bias) when reduce_output=False


class ParallelTransformerLayer(nn.Module):
    """Pre-LN transformer block (ref ParallelTransformerLayer,
    transformer.py:626-798)."""

    def __init__(self, hidden_size: int, num_heads: int, *, causal: bool,
                 norm: str = "layernorm", norm_eps: float = 1e-5,
                 mlp_type: str = "gelu", ffn_hidden_size=None,
                 rotary: bool = False, rope_base: float = 10000.0,
                 max_positions: int = 4096,
                 attention_dropout: float = 0.0, hidden_dropout: float = 0.0,
                 bias: bool = True, init_method=None, output_init_method=None,
                 dtype=None, layer_idx: int = 0,
                 parallel_residual: bool = False):
        super().__init__()
        # parallel_residual: GPT-J composition x + attn(ln1 x) + mlp(ln2 x)
        # with ONE deferred TP all-reduce over the summed partials
        # (ref transformer.py:710-752) — halves the per-layer TP comms
        self.parallel_residual = parallel_residual
        self.input_norm = get_norm(norm, hidden_size, norm_eps, dtype)
        self.attention = ParallelAttention(
            hidden_size, num_heads, causal=causal, rotary=rotary,
            rope_base=rope_base, max_positions=max_positions,
            attention_dropout=attention_dropout, hidden_dropout=hidden_dropout,
            bias=bias, init_method=init_method,
            output_init_method=output_init_method, dtype=dtype,
            layer_idx=layer_idx, reduce_output=not parallel_residual)
        self.post_attention_norm = get_norm(norm, hidden_size, norm_eps, dtype)
        if mlp_type == "swiglu":
            self.mlp = LLaMAParallelMLP(
                hidden_size, ffn_hidden_size, init_method=init_method,
                output_init_method=output_init_method, dtype=dtype,
                reduce_output=not parallel_residual)
        else:
            self.mlp = ParallelMLP(
                hidden_size, ffn_hidden_size, bias=bias, init_method=init_method,
                output_init_method=output_init_method, dtype=dtype,
                reduce_output=not parallel_residual)
        self.hidden_dropout = hidden_dropout

    @staticmethod
    def _split_partial(out):
        if isinstance(out, tuple):
            return out[0], out[1]
        return out, None

    def forward(self, x, attention_mask=None, cache=None):
        if self.parallel_residual:
            from fengshen_amd.parallel.mappings import (
                reduce_from_tensor_model_parallel_region)
            ap, ab = self._split_partial(self.attention(
                self.input_norm(x), attention_mask=attention_mask,
                cache=cache))
            mp, mb = self._split_partial(
                self.mlp(self.post_attention_norm(x)))
            total = reduce_from_tensor_model_parallel_region(ap + mp)
            bias = None
            if ab is not None or mb is not None:
                bias = (ab if ab is not None else 0)                     + (mb if mb is not None else 0)
            return F_ops.bias_dropout_add(total, bias, x,
                                          self.hidden_dropout,
                                          self.training)
        residual = x
        h = self.input_norm(x)
        attn_out = self.attention(h, attention_mask=attention_mask, cache=cache)
        x = F_ops.bias_dropout_add(attn_out, None, residual,
                                   self.hidden_dropout, self.training)
        residual = x
        h = self.post_attention_norm(x)
        mlp_out = self.mlp(h)
        x = F_ops.bias_dropout_add(mlp_out, None, residual,
                                   self.hidden_dropout, self.training)
        return x


def parallel_lm_logits(hidden: torch.Tensor, word_embeddings_weight: torch.Tensor,
                       parallel_output: bool = True):
    """LM head over (possibly TP-sharded) embedding weight
    (ref transformer.py:800-815)."""
    input_parallel = copy_to_tensor_model_parallel_region(hidden)
    logits_parallel = torch.nn.functional.linear(
        input_parallel, word_embeddings_weight)
    if parallel_output:
        return logits_parallel
    return gather_from_tensor_model_parallel_region(logits_parallel)


class SinusoidalPositionalEmbedding(nn.Module):
    """ref layers/positional_embeddings.py:19."""

    def __init__(self, dim: int, base: float = 10000.0):
        super().__init__()
        self.dim = dim
        self.base = base

    def forward(self, positions: torch.Tensor) -> torch.Tensor:
        half = self.dim // 2
        inv = 1.0 / (self.base ** (torch.arange(
            half, device=positions.device, dtype=torch.float32) / half))
        ang = positions.float()[..., None] * inv
        return torch.cat([ang.sin(), ang.cos()], dim=-1)


class AliBi(nn.Module):
    """Cached ALiBi bias matrix (ref positional_embeddings.py:90-173);
    TP-aware: each rank holds its heads' slopes."""

    def __init__(self, num_heads: int, mp_size: int = 1, mp_rank: int = 0):
        super().__init__()
        self.num_heads = num_heads
        slopes = torch.tensor(self._slopes(num_heads))
        per = num_heads // mp_size
        self.register_buffer("slopes", slopes[mp_rank * per:(mp_rank + 1) * per],
                             persistent=False)
        self._cache = None

    @staticmethod
    def _slopes(n: int):
        def pow2(n):
            start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
            return [start * (start ** i) for i in range(n)]
        if math.log2(n).is_integer():
            return pow2(n)
        closest = 2 ** math.floor(math.log2(n))
        return pow2(closest) + AliBi._slopes(2 * closest)[0::2][: n - closest]

    def forward(self, sq: int, sk: int, device, dtype):
        if (self._cache is None or self._cache.shape[-1] < sk
                or self._cache.device != device):
            pos = torch.arange(sk, device=device, dtype=torch.float32)
            self._cache = self.slopes.to(device)[:, None, None] * \
                (pos[None, None, :] - pos[None, :, None])
        bias = self._cache[:, :sq, :sk] if self._cache.shape[1] >= sq else \
            self._cache[:, -sq:, :sk]
        return bias.to(dtype)


class SoftEmbedding(nn.Module):
    """Prompt-tuning soft tokens prepended to the input embedding
    (ref word_embeddings.py:157)."""

    def __init__(self, wte: nn.Module, n_tokens: int = 10,
                 init_range: float = 0.5):
        super().__init__()
        self.wte = wte
        self.n_tokens = n_tokens
        dim = wte.weight.shape[1]
        self.soft_prompt = nn.Parameter(
            torch.empty(n_tokens, dim).uniform_(-init_range, init_range))

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        emb = self.wte(input_ids)
        b = emb.shape[0]
        prompt = self.soft_prompt.unsqueeze(0).expand(b, -1, -1).to(emb.dtype)
        return torch.cat([prompt, emb], dim=1)


class SpatialGatingUnit(nn.Module):
    """gMLP SGU (ref layers/gmlp.py:28-141)."""

    def __init__(self, dim_ff: int, seq_len: int):
        super().__init__()
        self.norm = LayerNorm(dim_ff // 2)
        self.proj = nn.Linear(seq_len, seq_len)
        nn.init.zeros_(self.proj.weight)
        nn.init.ones_(self.proj.bias)

    def forward(self, x):
        res, gate = x.chunk(2, dim=-1)
        gate = self.norm(gate)
        gate = self.proj(gate.transpose(1, 2)).transpose(1, 2)
        return res * gate


class GMLPBlock(nn.Module):
    def __init__(self, dim: int, dim_ff: int, seq_len: int):
        super().__init__()
        self.norm = LayerNorm(dim)
        self.fc_in = nn.Linear(dim, dim_ff)
        self.sgu = SpatialGatingUnit(dim_ff, seq_len)
        self.fc_out = nn.Linear(dim_ff // 2, dim)

    def forward(self, x):
        h = self.norm(x)
        h = F_ops.eager_gelu(self.fc_in(h).float()).to(h.dtype)
        h = self.sgu(h)
        return x + self.fc_out(h)


def expand_attention_types(attention_config, num_layers: int):
    """Expand a [[types, repeat], ...] spec into a per-layer type list
    (ref models/llama/modeling_llama.py:37-64 semantics): each entry
    contributes `repeat` layers cycling through its `types`; "all" in a
    spec means the same types tile the whole depth.

    >>> expand_attention_types([[["global"], 2], [["sparse_fixed"], 2]], 4)
    ['global', 'global', 'sparse_fixed', 'sparse_fixed']
    """
    if attention_config is None:
        return ["global"] * num_layers
    out = []
    for types, repeat in attention_config:
        if repeat == "all":
            for i in range(num_layers):
                out.append(types[i % len(types)])
            break
        for i in range(int(repeat)):
            out.append(types[i % len(types)])
    assert len(out) >= num_layers, (
        f"attention_config covers {len(out)} layers, model has {num_layers}")
    return out[:num_layers]


def get_ltor_masks_and_position_ids(data: torch.Tensor, eod_token: int,
                                    reset_position_ids: bool = False,
                                    reset_attention_mask: bool = False,
                                    eod_mask_loss: bool = False):
    """Left-to-right (causal) masks + position ids for packed GPT batches
    (ref layers/utils.py:38 get_ltor_masks_and_position_ids).

    Returns (attention_mask [b,1,s,s] bool — True where MASKED, the
    megatron convention — loss_mask [b,s] float, position_ids [b,s]).
    reset_attention_mask blocks attention across EOD boundaries;
    reset_position_ids restarts positions after each EOD; eod_mask_loss
    zeroes the loss at EOD tokens.
    """
    b, s = data.shape
    att = torch.tril(torch.ones(s, s, device=data.device)) \
        .unsqueeze(0).repeat(b, 1, 1)
    loss_mask = torch.ones(b, s, dtype=torch.float, device=data.device)
    if eod_mask_loss:
        loss_mask[data == eod_token] = 0.0
    position_ids = torch.arange(s, device=data.device) \
        .unsqueeze(0).repeat(b, 1)
    if reset_position_ids or reset_attention_mask:
        position_ids = position_ids.clone()
        for bi in range(b):
            eod_idx = position_ids[bi, data[bi] == eod_token]
            prev = 0
            for j in eod_idx.tolist():
                if reset_attention_mask:
                    att[bi, j + 1:, : j + 1] = 0
                if reset_position_ids:
                    position_ids[bi, j + 1:] -= j + 1 - prev
                    prev = j + 1
    return (att < 0.5).unsqueeze(1), loss_mask, position_ids
