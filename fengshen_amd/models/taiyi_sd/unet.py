"""UNet2DConditionModel — the SD denoiser, MI355X-native.

Behavioral parity: the reference finetunes diffusers' UNet
(finetune_taiyi_stable_diffusion/finetune.py; hot op = UNet cross/self
attention).  Ours: ResBlocks + transformer blocks whose self/cross
attention runs through fengshen_amd.ops (fused softmax HIP kernels on
spatial-token sequences).
"""
from __future__ import annotations

import math
from typing import List, Tuple

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel

from fengshen_amd.ops import functional as F_ops


class UNetConfig(PretrainedConfig):
    model_type = "fengshen_sd_unet"

    def __init__(self, in_channels: int = 4, out_channels: int = 4,
                 block_channels: Tuple[int, ...] = (64, 128, 256),
                 layers_per_block: int = 1, num_attention_heads: int = 4,
                 cross_attention_dim: int = 256, norm_groups: int = 16,
                 torch_dtype="bfloat16", **kw):
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.block_channels = list(block_channels)
        self.layers_per_block = layers_per_block
        self.num_attention_heads = num_attention_heads
        self.cross_attention_dim = cross_attention_dim
        self.norm_groups = norm_groups
        super().__init__(torch_dtype=torch_dtype, **kw)


def unet_tiny_config(**over):
    cfg = dict(block_channels=(32, 64), layers_per_block=1,
               num_attention_heads=2, cross_attention_dim=64, norm_groups=8)
    cfg.update(over)
    return UNetConfig(**cfg)


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    half = dim // 2
    freqs = torch.exp(-math.log(10000.0)
                      * torch.arange(half, device=t.device).float() / half)
    ang = t.float()[:, None] * freqs[None, :]
    return torch.cat([ang.cos(), ang.sin()], dim=-1)


class ResBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, temb_ch: int, groups: int):
        super().__init__()
        self.norm1 = nn.GroupNorm(min(groups, in_ch), in_ch)
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, padding=1)
        self.temb_proj = nn.Linear(temb_ch, out_ch)
        self.norm2 = nn.GroupNorm(min(groups, out_ch), out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, padding=1)
        self.skip = nn.Conv2d(in_ch, out_ch, 1) if in_ch != out_ch \
            else nn.Identity()

    def forward(self, x, temb):
        h = self.conv1(nn.functional.silu(self.norm1(x.float())).to(x.dtype))
        h = h + self.temb_proj(nn.functional.silu(temb))[:, :, None, None] \
            .to(h.dtype)
        h = self.conv2(nn.functional.silu(self.norm2(h.float())).to(h.dtype))
        return h + self.skip(x)


class SpatialTransformer(nn.Module):
    """self-attn + cross-attn + MLP over flattened spatial tokens; attention
    runs through ops.functional.attention (HIP fused softmax)."""

    def __init__(self, ch: int, heads: int, context_dim: int, groups: int):
        super().__init__()
        self.norm = nn.GroupNorm(min(groups, ch), ch)
        self.proj_in = nn.Conv2d(ch, ch, 1)
        self.heads = heads
        self.head_dim = ch // heads
        self.ln1 = nn.LayerNorm(ch)
        self.self_qkv = nn.Linear(ch, 3 * ch)
        self.self_out = nn.Linear(ch, ch)
        self.ln2 = nn.LayerNorm(ch)
        self.cross_q = nn.Linear(ch, ch)
        self.cross_kv = nn.Linear(context_dim, 2 * ch)
        self.cross_out = nn.Linear(ch, ch)
        self.ln3 = nn.LayerNorm(ch)
        self.mlp = nn.Sequential(nn.Linear(ch, 4 * ch), nn.GELU(),
                                 nn.Linear(4 * ch, ch))
        self.proj_out = nn.Conv2d(ch, ch, 1)

    def _attn(self, q, k, v, b, sq, sk):
        np_, hn = self.heads, self.head_dim
        q = q.view(b, sq, np_, hn).transpose(1, 2)
        k = k.view(b, sk, np_, hn).transpose(1, 2)
        v = v.view(b, sk, np_, hn).transpose(1, 2)
        ctx = F_ops.attention(q, k, v, causal=False,
                              scale=1.0 / math.sqrt(hn))
        return ctx.transpose(1, 2).reshape(b, sq, np_ * hn)

    def forward(self, x, context):
        b, c, hh, ww = x.shape
        res = x
        h = self.proj_in(nn.functional.group_norm(
            x.float(), self.norm.num_groups, self.norm.weight.float(),
            self.norm.bias.float()).to(x.dtype))
        tokens = h.flatten(2).transpose(1, 2)  # [b, hw, c]
        s = tokens.shape[1]
        t1 = self.ln1(tokens.float()).to(tokens.dtype)
        q, k, v = self.self_qkv(t1).chunk(3, dim=-1)
        tokens = tokens + self.self_out(self._attn(q, k, v, b, s, s))
        t2 = self.ln2(tokens.float()).to(tokens.dtype)
        q = self.cross_q(t2)
        k, v = self.cross_kv(context.to(t2.dtype)).chunk(2, dim=-1)
        tokens = tokens + self.cross_out(
            self._attn(q, k, v, b, s, context.shape[1]))
        t3 = self.ln3(tokens.float()).to(tokens.dtype)
        tokens = tokens + self.mlp(t3)
        h = tokens.transpose(1, 2).reshape(b, c, hh, ww)
        return res + self.proj_out(h)


class UNet2DConditionModel(PreTrainedModel):
    config_class = UNetConfig

    def _init_weights(self, module):
        pass

    def __init__(self, config: UNetConfig):
        super().__init__(config)
        chs = config.block_channels
        temb_ch = chs[0] * 4
        self.time_mlp = nn.Sequential(
            nn.Linear(chs[0], temb_ch), nn.SiLU(), nn.Linear(temb_ch, temb_ch))
        self.conv_in = nn.Conv2d(config.in_channels, chs[0], 3, padding=1)

        g = config.norm_groups
        self.down_blocks = nn.ModuleList()
        self.downsamplers = nn.ModuleList()
        in_ch = chs[0]
        for level, ch in enumerate(chs):
            blocks = nn.ModuleList()
            for _ in range(config.layers_per_block):
                blocks.append(nn.ModuleList([
                    ResBlock(in_ch, ch, temb_ch, g),
                    SpatialTransformer(ch, config.num_attention_heads,
                                       config.cross_attention_dim, g)]))
                in_ch = ch
            self.down_blocks.append(blocks)
            self.downsamplers.append(
                nn.Conv2d(ch, ch, 3, stride=2, padding=1)
                if level < len(chs) - 1 else nn.Identity())

        self.mid_res1 = ResBlock(chs[-1], chs[-1], temb_ch, g)
        self.mid_attn = SpatialTransformer(
            chs[-1], config.num_attention_heads, config.cross_attention_dim, g)
        self.mid_res2 = ResBlock(chs[-1], chs[-1], temb_ch, g)

        self.up_blocks = nn.ModuleList()
        self.upsamplers = nn.ModuleList()
        for level, ch in enumerate(reversed(chs)):
            blocks = nn.ModuleList()
            for bi in range(config.layers_per_block):
                block_in = in_ch + ch if bi == 0 else ch  # skip concat once
                blocks.append(nn.ModuleList([
                    ResBlock(block_in, ch, temb_ch, g),
                    SpatialTransformer(ch, config.num_attention_heads,
                                       config.cross_attention_dim, g)]))
                in_ch = ch
            self.up_blocks.append(blocks)
            self.upsamplers.append(
                nn.Upsample(scale_factor=2, mode="nearest")
                if level < len(chs) - 1 else nn.Identity())

        self.norm_out = nn.GroupNorm(min(g, chs[0]), chs[0])
        self.conv_out = nn.Conv2d(chs[0], config.out_channels, 3, padding=1)
        self.post_init()

    def forward(self, sample: torch.Tensor, timestep: torch.Tensor,
                encoder_hidden_states: torch.Tensor, **_kw):
        temb = self.time_mlp(
            timestep_embedding(timestep, self.config.block_channels[0])
            .to(sample.dtype))
        h = self.conv_in(sample)
        skips: List[torch.Tensor] = []
        for blocks, down in zip(self.down_blocks, self.downsamplers):
            for res, attn in blocks:
                h = res(h, temb)
                h = attn(h, encoder_hidden_states)
            skips.append(h)
            h = down(h)
        h = self.mid_res1(h, temb)
        h = self.mid_attn(h, encoder_hidden_states)
        h = self.mid_res2(h, temb)
        for blocks, up in zip(self.up_blocks, self.upsamplers):
            skip = skips.pop()
            if h.shape[-2:] != skip.shape[-2:]:
                h = nn.functional.interpolate(h.float(), size=skip.shape[-2:],
                                              mode="nearest").to(h.dtype)
            for bi, (res, attn) in enumerate(blocks):
                inp = torch.cat([h, skip], dim=1) if bi == 0 else h
                h = res(inp, temb)
                h = attn(h, encoder_hidden_states)
            h = up(h)
        h = nn.functional.silu(self.norm_out(h.float())).to(h.dtype)
        return self.conv_out(h)
