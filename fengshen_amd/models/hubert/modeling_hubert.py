"""HuBERT: masked-prediction audio SSL over k-means pseudo-labels.

Behavioral parity: reference examples/hubert/pretrain_hubert.py (HF
HubertModel + label-embedding NCE head, compute_nce :141-151 /
compute_pred :165-171) and data/hubert/hubert_dataset.py. Self-contained
here: conv waveform frontend + transformer encoder built from this
package's layers (flash attention eligible on MI355X), span time-masking,
and the cosine-similarity NCE loss over masked frames.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput

from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelAttention,
    ParallelMLP,
    init_normal,
    scaled_init_normal,
)


class HubertConfig(PretrainedConfig):
    model_type = "fengshen_hubert"

    def __init__(self, vocab_size: int = 504,  # k-means clusters
                 hidden_size: int = 768, num_hidden_layers: int = 12,
                 num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 conv_dim: Tuple[int, ...] = (512, 512, 512, 512, 512, 512, 512),
                 conv_stride: Tuple[int, ...] = (5, 2, 2, 2, 2, 2, 2),
                 conv_kernel: Tuple[int, ...] = (10, 3, 3, 3, 3, 2, 2),
                 num_conv_pos_embeddings: int = 128,
                 num_conv_pos_embedding_groups: int = 16,
                 mask_time_prob: float = 0.65, mask_time_length: int = 10,
                 final_dim: int = 256, logit_temp: float = 0.1,
                 layer_norm_eps: float = 1e-5,
                 initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1, attention_dropout: float = 0.1,
                 torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.conv_dim = list(conv_dim)
        self.conv_stride = list(conv_stride)
        self.conv_kernel = list(conv_kernel)
        self.num_conv_pos_embeddings = num_conv_pos_embeddings
        self.num_conv_pos_embedding_groups = num_conv_pos_embedding_groups
        self.mask_time_prob = mask_time_prob
        self.mask_time_length = mask_time_length
        self.final_dim = final_dim
        self.logit_temp = logit_temp
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(torch_dtype=torch_dtype, **kw)


def hubert_tiny_config(**over):
    cfg = dict(vocab_size=16, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               conv_dim=(32, 32, 32), conv_stride=(5, 2, 2),
               conv_kernel=(10, 3, 3), num_conv_pos_embeddings=16,
               num_conv_pos_embedding_groups=4, final_dim=32,
               mask_time_length=2)
    cfg.update(over)
    return HubertConfig(**cfg)


def compute_mask_indices(shape: Tuple[int, int], mask_prob: float,
                         mask_length: int,
                         padding_mask: Optional[torch.Tensor] = None,
                         min_masks: int = 2) -> torch.Tensor:
    """Span time-masking (HF _compute_mask_indices semantics)."""
    b, t = shape
    mask = torch.zeros(b, t, dtype=torch.bool)
    if mask_length >= t:
        mask[:, :] = True
        return mask
    for i in range(b):
        length = t
        if padding_mask is not None:
            length = int((~padding_mask[i]).long().sum())
        n_spans = max(min_masks,
                      int(mask_prob * max(length, 1) / mask_length + 0.5))
        hi = max(length - mask_length, 1)
        starts = torch.randint(0, hi, (n_spans,))
        for s in starts:
            mask[i, s:s + mask_length] = True
    return mask


class _ConvFeatureExtractor(nn.Module):
    """Waveform -> frame features (ref: HF Hubert conv frontend)."""

    def __init__(self, config: HubertConfig):
        super().__init__()
        layers = []
        in_d = 1
        for d, k, s in zip(config.conv_dim, config.conv_kernel,
                           config.conv_stride):
            layers.append(nn.Conv1d(in_d, d, k, stride=s, bias=False))
            layers.append(nn.GELU())
            in_d = d
        self.conv_layers = nn.Sequential(*layers)

    def forward(self, source: torch.Tensor) -> torch.Tensor:
        # [b, samples] fp32 audio -> [b, T, conv_dim[-1]] in model dtype
        w = self.conv_layers[0].weight
        return self.conv_layers(
            source.to(w.dtype)[:, None, :]).transpose(1, 2)


class _ConvPositionalEmbedding(nn.Module):
    def __init__(self, config: HubertConfig):
        super().__init__()
        self.conv = nn.Conv1d(
            config.hidden_size, config.hidden_size,
            kernel_size=config.num_conv_pos_embeddings,
            padding=config.num_conv_pos_embeddings // 2,
            groups=config.num_conv_pos_embedding_groups)

    def forward(self, x):
        h = self.conv(x.transpose(1, 2))
        if self.conv.kernel_size[0] % 2 == 0:
            h = h[:, :, :-1]
        return x + F.gelu(h).transpose(1, 2)


class _EncoderLayer(nn.Module):
    def __init__(self, config: HubertConfig):
        super().__init__()
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range,
                                config.num_hidden_layers)
        self.ln1 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.attn = ParallelAttention(
            config.hidden_size, config.num_attention_heads, causal=False,
            attention_dropout=config.attention_dropout,
            hidden_dropout=config.hidden_dropout,
            init_method=im, output_init_method=om)
        self.ln2 = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.mlp = ParallelMLP(config.hidden_size, config.intermediate_size,
                               init_method=im, output_init_method=om)

    def forward(self, x, attention_mask=None):
        x = x + self.attn(self.ln1(x), attention_mask=attention_mask)
        return x + self.mlp(self.ln2(x))


@dataclass
class HubertOutput(ModelOutput):
    last_hidden_state: Optional[torch.Tensor] = None
    mask_time_indices: Optional[torch.Tensor] = None
    frame_padding_mask: Optional[torch.Tensor] = None


@dataclass
class HubertPreTrainingOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None
    mask_time_indices: Optional[torch.Tensor] = None


class HubertPreTrainedModel(PreTrainedModel):
    config_class = HubertConfig
    base_model_prefix = "hubert"
    main_input_name = "source"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Conv1d)):
            module.weight.data.normal_(0.0, self.config.initializer_range)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, self.config.initializer_range)


class HubertModel(HubertPreTrainedModel):
    def __init__(self, config: HubertConfig):
        super().__init__(config)
        self.feature_extractor = _ConvFeatureExtractor(config)
        self.feature_projection = nn.Linear(config.conv_dim[-1],
                                            config.hidden_size)
        self.feature_ln = LayerNorm(config.conv_dim[-1],
                                    eps=config.layer_norm_eps)
        self.pos_conv = _ConvPositionalEmbedding(config)
        self.layers = nn.ModuleList(
            [_EncoderLayer(config) for _ in range(config.num_hidden_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.masked_spec_embed = nn.Parameter(
            torch.empty(config.hidden_size).uniform_())
        self.post_init()

    def frame_lengths(self, sample_lengths: torch.Tensor) -> torch.Tensor:
        out = sample_lengths
        for k, s in zip(self.config.conv_kernel, self.config.conv_stride):
            out = torch.div(out - k, s, rounding_mode="floor") + 1
        return out

    def forward(self, source, padding_mask=None, apply_mask: bool = True,
                mask_time_indices=None, **_kw):
        """source [b, samples] fp32; padding_mask [b, samples] True=pad."""
        feats = self.feature_extractor(source)          # [b, T, c]
        h = self.feature_projection(self.feature_ln(feats))
        b, t, _ = h.shape
        frame_pad = None
        if padding_mask is not None:
            lens = self.frame_lengths((~padding_mask).long().sum(-1))
            frame_pad = (torch.arange(t, device=h.device)[None, :]
                         >= lens[:, None])
        if apply_mask and mask_time_indices is None:
            mask_time_indices = compute_mask_indices(
                (b, t), self.config.mask_time_prob,
                self.config.mask_time_length, frame_pad).to(h.device)
        if mask_time_indices is not None:
            h = torch.where(mask_time_indices[..., None],
                            self.masked_spec_embed.to(h.dtype), h)
        h = self.pos_conv(h)
        attn_mask = None
        if frame_pad is not None:
            attn_mask = frame_pad[:, None, None, :]  # True = masked
        for layer in self.layers:
            h = layer(h, attention_mask=attn_mask)
        return HubertOutput(last_hidden_state=self.ln_f(h),
                            mask_time_indices=mask_time_indices,
                            frame_padding_mask=frame_pad)


class HubertForPreTraining(HubertPreTrainedModel):
    """NCE over label embeddings (ref pretrain_hubert.py:141-171):
    logit[i, c] = cos(proj(x_i), emb_c) / temp; target = true cluster."""

    def __init__(self, config: HubertConfig):
        super().__init__(config)
        self.hubert = HubertModel(config)
        self.final_proj = nn.Linear(config.hidden_size, config.final_dim)
        self.label_embs = nn.Parameter(
            torch.empty(config.vocab_size, config.final_dim).uniform_())
        self.post_init()

    def forward(self, source, padding_mask=None, labels=None, **_kw):
        out = self.hubert(source, padding_mask=padding_mask, apply_mask=True)
        x = out.last_hidden_state
        mi = out.mask_time_indices
        loss = None
        logits = None
        if labels is not None:
            t = x.shape[1]
            lab = labels
            if lab.shape[1] < t:   # label-rate vs frame-rate slack
                lab = F.pad(lab, (0, t - lab.shape[1]), value=-100)
            lab = lab[:, :t]
            masked = mi & (lab != -100)
            if out.frame_padding_mask is not None:
                masked = masked & ~out.frame_padding_mask
            proj = self.final_proj(x[masked])               # [M, D]
            logits = F.cosine_similarity(
                proj.float()[:, None, :],
                self.label_embs.float()[None, :, :], dim=-1)
            logits = logits / self.config.logit_temp
            loss = F.cross_entropy(logits, lab[masked])
        return HubertPreTrainingOutput(loss=loss, logits=logits,
                                       mask_time_indices=mi)
