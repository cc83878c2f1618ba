"""UNet2DConditionModel — the SD denoiser, MI355X-native, in diffusers'
weight layout.

Behavioral parity: the reference finetunes diffusers' UNet inside
StableDiffusionPipeline (finetune_taiyi_stable_diffusion/finetune.py:81-158;
hot op = UNet cross/self attention).  This implementation mirrors the
Stable-Diffusion-1 architecture AND parameter naming (conv_in,
time_embedding.linear_1/2, down_blocks.N.resnets/attentions/downsamplers,
mid_block, up_blocks, conv_norm_out/conv_out; BasicTransformerBlock with
attn1/attn2 to_q/to_k/to_v/to_out.0 and GEGLU ff.net) so real
Taiyi-SD / SD-1.x checkpoints map 1:1 onto state_dict keys.  Attention runs
through fengshen_amd.ops (fused softmax / flash HIP kernels); no xformers.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel

from fengshen_amd.ops import functional as F_ops
from fengshen_amd.parallel.random import checkpoint as activation_checkpoint


def _gn(norm: nn.GroupNorm, x: torch.Tensor) -> torch.Tensor:
    """GroupNorm in fp32 (stats + affine) regardless of param dtype."""
    return nn.functional.group_norm(
        x.float(), norm.num_groups,
        norm.weight.float() if norm.weight is not None else None,
        norm.bias.float() if norm.bias is not None else None,
        norm.eps).to(x.dtype)


def _ln(norm: nn.LayerNorm, x: torch.Tensor) -> torch.Tensor:
    """LayerNorm in fp32 regardless of param dtype."""
    return nn.functional.layer_norm(
        x.float(), norm.normalized_shape,
        norm.weight.float() if norm.weight is not None else None,
        norm.bias.float() if norm.bias is not None else None,
        norm.eps).to(x.dtype)


class UNetConfig(PretrainedConfig):
    model_type = "fengshen_sd_unet"

    def __init__(self,
                 sample_size: int = 64,
                 in_channels: int = 4,
                 out_channels: int = 4,
                 down_block_types: Tuple[str, ...] = (
                     "CrossAttnDownBlock2D", "CrossAttnDownBlock2D",
                     "CrossAttnDownBlock2D", "DownBlock2D"),
                 up_block_types: Tuple[str, ...] = (
                     "UpBlock2D", "CrossAttnUpBlock2D",
                     "CrossAttnUpBlock2D", "CrossAttnUpBlock2D"),
                 block_out_channels: Tuple[int, ...] = (320, 640, 1280, 1280),
                 layers_per_block: int = 2,
                 attention_head_dim: int = 8,  # = NUM heads (diffusers SD-1)
                 cross_attention_dim: int = 768,
                 norm_num_groups: int = 32,
                 torch_dtype="bfloat16", **kw):
        self.sample_size = sample_size
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.down_block_types = list(down_block_types)
        self.up_block_types = list(up_block_types)
        self.block_out_channels = list(block_out_channels)
        self.layers_per_block = layers_per_block
        self.attention_head_dim = attention_head_dim
        self.cross_attention_dim = cross_attention_dim
        self.norm_num_groups = norm_num_groups
        super().__init__(torch_dtype=torch_dtype, **kw)


def taiyi_sd_1b_config(**over):
    """SD-1 / Taiyi-SD-1B UNet shape (~860M params): channels
    320/640/1280/1280, 2 layers/block, 8 heads, cross dim 768."""
    return UNetConfig(**over)


def unet_tiny_config(**over):
    cfg = dict(sample_size=16,
               down_block_types=("CrossAttnDownBlock2D", "DownBlock2D"),
               up_block_types=("UpBlock2D", "CrossAttnUpBlock2D"),
               block_out_channels=(32, 64), layers_per_block=1,
               attention_head_dim=2, cross_attention_dim=64,
               norm_num_groups=8)
    cfg.update(over)
    return UNetConfig(**cfg)


def timestep_embedding(t: torch.Tensor, dim: int,
                       flip_sin_to_cos: bool = True,
                       downscale_freq_shift: float = 0.0) -> torch.Tensor:
    """diffusers get_timestep_embedding semantics (SD: flip, shift 0)."""
    half = dim // 2
    exponent = -math.log(10000.0) * torch.arange(
        half, device=t.device).float()
    exponent = exponent / (half - downscale_freq_shift)
    emb = t.float()[:, None] * exponent.exp()[None, :]
    sin, cos = emb.sin(), emb.cos()
    if flip_sin_to_cos:
        return torch.cat([cos, sin], dim=-1)
    return torch.cat([sin, cos], dim=-1)


class TimestepEmbedding(nn.Module):
    """time_embedding.linear_1/linear_2 (diffusers naming)."""

    def __init__(self, in_dim: int, time_embed_dim: int):
        super().__init__()
        self.linear_1 = nn.Linear(in_dim, time_embed_dim)
        self.linear_2 = nn.Linear(time_embed_dim, time_embed_dim)

    def forward(self, x):
        return self.linear_2(nn.functional.silu(self.linear_1(x)))


class ResnetBlock2D(nn.Module):
    """diffusers ResnetBlock2D: norm1/conv1 + time_emb_proj + norm2/conv2
    (+ conv_shortcut), silu activation."""

    def __init__(self, in_ch: int, out_ch: int, temb_ch: int, groups: int,
                 eps: float = 1e-5):
        super().__init__()
        self.norm1 = nn.GroupNorm(groups, in_ch, eps=eps)
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, padding=1)
        self.time_emb_proj = nn.Linear(temb_ch, out_ch)
        self.norm2 = nn.GroupNorm(groups, out_ch, eps=eps)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, padding=1)
        self.conv_shortcut = (nn.Conv2d(in_ch, out_ch, 1)
                              if in_ch != out_ch else None)

    def forward(self, x, temb):
        h = self.conv1(nn.functional.silu(_gn(self.norm1, x).float())
                       .to(x.dtype))
        h = h + self.time_emb_proj(
            nn.functional.silu(temb))[:, :, None, None].to(h.dtype)
        h = self.conv2(nn.functional.silu(_gn(self.norm2, h).float())
                       .to(h.dtype))
        skip = x if self.conv_shortcut is None else self.conv_shortcut(x)
        return h + skip


class CrossAttention(nn.Module):
    """diffusers CrossAttention: to_q/to_k/to_v (no bias) + to_out.0;
    compute rides ops.functional.attention (flash / fused-softmax HIP)."""

    def __init__(self, query_dim: int, context_dim: Optional[int],
                 heads: int):
        super().__init__()
        ctx = context_dim if context_dim is not None else query_dim
        self.heads = heads
        self.head_dim = query_dim // heads
        self.to_q = nn.Linear(query_dim, query_dim, bias=False)
        self.to_k = nn.Linear(ctx, query_dim, bias=False)
        self.to_v = nn.Linear(ctx, query_dim, bias=False)
        self.to_out = nn.ModuleList([nn.Linear(query_dim, query_dim)])

    def forward(self, x, context=None):
        b, sq, _ = x.shape
        ctx_in = x if context is None else context.to(x.dtype)
        sk = ctx_in.shape[1]
        np_, hn = self.heads, self.head_dim
        q = self.to_q(x).view(b, sq, np_, hn).transpose(1, 2)
        k = self.to_k(ctx_in).view(b, sk, np_, hn).transpose(1, 2)
        v = self.to_v(ctx_in).view(b, sk, np_, hn).transpose(1, 2)
        out = F_ops.attention(q, k, v, causal=False,
                              scale=1.0 / math.sqrt(hn))
        out = out.transpose(1, 2).reshape(b, sq, np_ * hn)
        return self.to_out[0](out)


class GEGLU(nn.Module):
    """ff.net.0: proj to 2*inner then x * gelu(gate) (diffusers GEGLU)."""

    def __init__(self, dim_in: int, dim_out: int):
        super().__init__()
        self.proj = nn.Linear(dim_in, dim_out * 2)

    def forward(self, x):
        h, gate = self.proj(x).chunk(2, dim=-1)
        return h * nn.functional.gelu(gate.float()).to(gate.dtype)


class FeedForward(nn.Module):
    """ff.net = [GEGLU, Dropout, Linear] (diffusers indices 0/1/2)."""

    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = dim * mult
        self.net = nn.ModuleList(
            [GEGLU(dim, inner), nn.Dropout(0.0), nn.Linear(inner, dim)])

    def forward(self, x):
        for layer in self.net:
            x = layer(x)
        return x


class BasicTransformerBlock(nn.Module):
    def __init__(self, dim: int, heads: int, context_dim: int):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim)
        self.attn1 = CrossAttention(dim, None, heads)       # self
        self.norm2 = nn.LayerNorm(dim)
        self.attn2 = CrossAttention(dim, context_dim, heads)  # cross
        self.norm3 = nn.LayerNorm(dim)
        self.ff = FeedForward(dim)

    def forward(self, x, context):
        x = x + self.attn1(_ln(self.norm1, x))
        x = x + self.attn2(_ln(self.norm2, x), context)
        x = x + self.ff(_ln(self.norm3, x))
        return x


class Transformer2DModel(nn.Module):
    """diffusers Transformer2DModel (SD-1 flavor: conv proj_in/out)."""

    def __init__(self, ch: int, heads: int, context_dim: int, groups: int):
        super().__init__()
        self.norm = nn.GroupNorm(groups, ch, eps=1e-6)
        self.proj_in = nn.Conv2d(ch, ch, 1)
        self.transformer_blocks = nn.ModuleList(
            [BasicTransformerBlock(ch, heads, context_dim)])
        self.proj_out = nn.Conv2d(ch, ch, 1)

    def forward(self, x, context):
        b, c, hh, ww = x.shape
        res = x
        h = self.proj_in(_gn(self.norm, x))
        tokens = h.flatten(2).transpose(1, 2)  # [b, hw, c]
        for block in self.transformer_blocks:
            tokens = block(tokens, context)
        h = tokens.transpose(1, 2).reshape(b, c, hh, ww)
        return res + self.proj_out(h)


class Downsample2D(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv = nn.Conv2d(ch, ch, 3, stride=2, padding=1)

    def forward(self, x):
        return self.conv(x)


class Upsample2D(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        x = nn.functional.interpolate(x.float(), scale_factor=2,
                                      mode="nearest").to(x.dtype)
        return self.conv(x)


class DownBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, temb_ch, layers, groups,
                 heads=None, context_dim=None, cross_attn=False,
                 add_downsample=True):
        super().__init__()
        self.resnets = nn.ModuleList()
        self.attentions = nn.ModuleList() if cross_attn else None
        for i in range(layers):
            self.resnets.append(ResnetBlock2D(
                in_ch if i == 0 else out_ch, out_ch, temb_ch, groups))
            if cross_attn:
                self.attentions.append(Transformer2DModel(
                    out_ch, heads, context_dim, groups))
        self.downsamplers = (nn.ModuleList([Downsample2D(out_ch)])
                             if add_downsample else None)

    def forward(self, h, temb, context, ckpt=False):
        skips = []
        for i, res in enumerate(self.resnets):
            def run(h, temb, context, i=i, res=res):
                h = res(h, temb)
                if self.attentions is not None:
                    h = self.attentions[i](h, context)
                return h
            if ckpt and torch.is_grad_enabled():
                h = activation_checkpoint(run, h, temb, context)
            else:
                h = run(h, temb, context)
            skips.append(h)
        if self.downsamplers is not None:
            h = self.downsamplers[0](h)
            skips.append(h)
        return h, skips


class UpBlock2D(nn.Module):
    def __init__(self, in_ch, prev_out_ch, out_ch, temb_ch, layers, groups,
                 heads=None, context_dim=None, cross_attn=False,
                 add_upsample=True):
        """in_ch: channels of the skip at the DEEPEST position of this
        block's level; prev_out_ch: channels flowing in from below."""
        super().__init__()
        self.resnets = nn.ModuleList()
        self.attentions = nn.ModuleList() if cross_attn else None
        for i in range(layers):
            res_skip_ch = in_ch if (i == layers - 1) else out_ch
            res_in_ch = prev_out_ch if i == 0 else out_ch
            self.resnets.append(ResnetBlock2D(
                res_in_ch + res_skip_ch, out_ch, temb_ch, groups))
            if cross_attn:
                self.attentions.append(Transformer2DModel(
                    out_ch, heads, context_dim, groups))
        self.upsamplers = (nn.ModuleList([Upsample2D(out_ch)])
                           if add_upsample else None)

    def forward(self, h, skips: List[torch.Tensor], temb, context,
                ckpt=False):
        for i, res in enumerate(self.resnets):
            skip = skips.pop()
            def run(h, skip, temb, context, i=i, res=res):
                h = res(torch.cat([h, skip], dim=1), temb)
                if self.attentions is not None:
                    h = self.attentions[i](h, context)
                return h
            if ckpt and torch.is_grad_enabled():
                h = activation_checkpoint(run, h, skip, temb, context)
            else:
                h = run(h, skip, temb, context)
        if self.upsamplers is not None:
            h = self.upsamplers[0](h)
        return h


class UNetMidBlock2DCrossAttn(nn.Module):
    def __init__(self, ch, temb_ch, groups, heads, context_dim):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(ch, ch, temb_ch, groups),
            ResnetBlock2D(ch, ch, temb_ch, groups)])
        self.attentions = nn.ModuleList(
            [Transformer2DModel(ch, heads, context_dim, groups)])

    def forward(self, h, temb, context):
        h = self.resnets[0](h, temb)
        h = self.attentions[0](h, context)
        return self.resnets[1](h, temb)


class UNet2DConditionModel(PreTrainedModel):
    config_class = UNetConfig

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            nn.init.normal_(module.weight, std=0.02)
            if module.bias is not None:
                nn.init.zeros_(module.bias)

    def __init__(self, config: UNetConfig):
        super().__init__(config)
        chs = config.block_out_channels
        heads = config.attention_head_dim
        ctx = config.cross_attention_dim
        g = config.norm_num_groups
        temb_ch = chs[0] * 4

        self.conv_in = nn.Conv2d(config.in_channels, chs[0], 3, padding=1)
        self.time_embedding = TimestepEmbedding(chs[0], temb_ch)

        self.down_blocks = nn.ModuleList()
        out_ch = chs[0]
        for i, btype in enumerate(config.down_block_types):
            in_ch, out_ch = out_ch, chs[i]
            self.down_blocks.append(DownBlock2D(
                in_ch, out_ch, temb_ch, config.layers_per_block, g,
                heads=heads, context_dim=ctx,
                cross_attn=(btype == "CrossAttnDownBlock2D"),
                add_downsample=(i < len(chs) - 1)))

        self.mid_block = UNetMidBlock2DCrossAttn(
            chs[-1], temb_ch, g, heads, ctx)

        self.up_blocks = nn.ModuleList()
        rev = list(reversed(chs))
        prev_out = rev[0]
        for i, btype in enumerate(config.up_block_types):
            out_ch = rev[i]
            in_ch = rev[min(i + 1, len(chs) - 1)]
            self.up_blocks.append(UpBlock2D(
                in_ch, prev_out, out_ch, temb_ch,
                config.layers_per_block + 1, g,
                heads=heads, context_dim=ctx,
                cross_attn=(btype == "CrossAttnUpBlock2D"),
                add_upsample=(i < len(chs) - 1)))
            prev_out = out_ch

        self.conv_norm_out = nn.GroupNorm(g, chs[0], eps=1e-5)
        self.conv_out = nn.Conv2d(chs[0], config.out_channels, 3, padding=1)
        self._ckpt = False
        self.post_init()

    def gradient_checkpointing_enable(self, **_kw):
        self._ckpt = True

    @property
    def dtype(self):
        return self.conv_in.weight.dtype

    def forward(self, sample: torch.Tensor, timestep: torch.Tensor,
                encoder_hidden_states: torch.Tensor, **_kw):
        if timestep.dim() == 0:
            timestep = timestep[None].expand(sample.shape[0])
        temb = self.time_embedding(
            timestep_embedding(timestep,
                               self.config.block_out_channels[0])
            .to(sample.dtype))
        h = self.conv_in(sample)
        skips: List[torch.Tensor] = [h]
        for block in self.down_blocks:
            h, s = block(h, temb, encoder_hidden_states, ckpt=self._ckpt)
            skips.extend(s)
        h = self.mid_block(h, temb, encoder_hidden_states)
        for block in self.up_blocks:
            h = block(h, skips, temb, encoder_hidden_states,
                      ckpt=self._ckpt)
        h = nn.functional.silu(
            _gn(self.conv_norm_out, h).float()).to(h.dtype)
        return self.conv_out(h)


# back-compat aliases (round-1 API)
ResBlock = ResnetBlock2D
