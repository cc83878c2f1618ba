"""Ziya-LLaMA, TP-aware, MI355X-native.

Behavioral parity: reference models/llama/modeling_llama.py (LlamaModel :97,
LlamaForCausalLM :239: Embedding + N ParallelTransformerLayer(rotary) +
final RMSNorm + parallel head + shift-CE; generation via HF generate()).
Redesign notes:
  * [b,s,h] layout (no :201-style transposes)
  * loss uses vocab-parallel CE on the sharded logits — the reference
    gathers full [b,s,V] logits then dense CE (modeling_llama.py:332-339);
    our way avoids the V-wide all-gather (SURVEY.md §3.3 note)
  * hot ops (RMSNorm/RoPE/SwiGLU/attention softmax) are HIP kernels
  * activation checkpointing with RNG restore per layer
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
from transformers import PreTrainedModel
from transformers.generation import GenerationMixin
from transformers.modeling_outputs import (
    BaseModelOutputWithPast,
    CausalLMOutputWithPast,
)

from fengshen_amd.models.layers import (
    ParallelTransformerLayer,
    RMSNorm,
    init_normal,
    scaled_init_normal,
)
from fengshen_amd.models.llama.configuration_llama import LlamaConfig
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from fengshen_amd.parallel.layers import ColumnParallelLinear, VocabParallelEmbedding
from fengshen_amd.parallel.mappings import gather_from_tensor_model_parallel_region
from fengshen_amd.parallel.random import checkpoint as activation_checkpoint


class LlamaPreTrainedModel(PreTrainedModel):
    config_class = LlamaConfig
    base_model_prefix = "model"
    supports_gradient_checkpointing = True
    _no_split_modules = ["ParallelTransformerLayer"]

    def _init_weights(self, module):
        # parallel layers self-initialize with the TP-aware master init
        pass


class LlamaModel(LlamaPreTrainedModel):
    def __init__(self, config: LlamaConfig):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range, config.num_hidden_layers)
        self.embed_tokens = VocabParallelEmbedding(
            config.vocab_size, config.hidden_size, init_method=im)
        self.layers = nn.ModuleList([
            ParallelTransformerLayer(
                config.hidden_size, config.num_attention_heads, causal=True,
                norm="rmsnorm", norm_eps=config.rms_norm_epsilon,
                mlp_type="swiglu", ffn_hidden_size=config.intermediate_size,
                rotary=True, rope_base=config.rotary_emb_base,
                max_positions=config.max_position_embeddings,
                attention_dropout=config.attention_dropout,
                hidden_dropout=config.hidden_dropout,
                bias=False, init_method=im, output_init_method=om,
                layer_idx=i,
                parallel_residual=getattr(config, "parallel_residual",
                                          False))
            for i in range(config.num_hidden_layers)])
        self.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_epsilon)
        self.gradient_checkpointing = False
        # selective activation ckpt: every k-th layer keeps activations
        # (no recompute) — spends spare HBM3E to skip recompute FLOPs
        self.gradient_checkpointing_skip_interval = 0
        self.post_init()

    def get_input_embeddings(self):
        return self.embed_tokens

    def set_input_embeddings(self, value):
        self.embed_tokens = value

    def forward(self, input_ids: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None,
                past_key_values=None, use_cache: bool = False,
                inputs_embeds: Optional[torch.Tensor] = None, **_kw):
        h = inputs_embeds if inputs_embeds is not None else \
            self.embed_tokens(input_ids)
        # mask convention: internal "True = masked" [b,1,sq,sk]; HF passes
        # [b, sk] with 1 = keep (ref mask build modeling_llama.py:180-198)
        mask = None
        if attention_mask is not None and attention_mask.dim() == 2:
            if attention_mask.min() == 1:
                mask = None  # nothing padded: fused causal path
            else:
                mask = (attention_mask == 0)[:, None, None, :]
        elif attention_mask is not None:
            mask = attention_mask

        cache = past_key_values
        if use_cache and cache is None:
            from transformers.cache_utils import DynamicCache
            cache = DynamicCache()

        skip = self.gradient_checkpointing_skip_interval
        for i, layer in enumerate(self.layers):
            ckpt = self.gradient_checkpointing and self.training \
                and cache is None
            if ckpt and skip and (i % skip == skip - 1):
                ckpt = False  # selective: keep this layer's activations
            if ckpt:
                h = activation_checkpoint(
                    lambda x, m, lyr=layer: lyr(x, attention_mask=m), h, mask)
            else:
                h = layer(h, attention_mask=mask, cache=cache)
        h = self.norm(h)
        return BaseModelOutputWithPast(last_hidden_state=h,
                                       past_key_values=cache)


class LlamaForCausalLM(LlamaPreTrainedModel, GenerationMixin):
    def __init__(self, config: LlamaConfig):
        super().__init__(config)
        self.model = LlamaModel(config)
        self.lm_head = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, bias=False,
            gather_output=False, init_method=init_normal(config.initializer_range))
        self.post_init()

    def get_input_embeddings(self):
        return self.model.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def gradient_checkpointing_enable(self, skip_interval: int = 0, **_kw):
        self.model.gradient_checkpointing = True
        self.model.gradient_checkpointing_skip_interval = max(skip_interval, 0)

    def forward(self, input_ids: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None,
                labels: Optional[torch.Tensor] = None,
                past_key_values=None, use_cache: bool = False,
                inputs_embeds=None, return_dict: bool = True, **_kw):
        out = self.model(input_ids, attention_mask=attention_mask,
                         past_key_values=past_key_values, use_cache=use_cache,
                         inputs_embeds=inputs_embeds)
        h = out.last_hidden_state
        logits_parallel = self.lm_head(h)  # [b, s, V/tp]

        loss = None
        if labels is not None:
            shift_logits = logits_parallel[:, :-1, :].contiguous()
            shift_labels = labels[:, 1:].contiguous()
            per_token = vocab_parallel_cross_entropy(shift_logits, shift_labels)
            valid = (shift_labels != -100)
            # vocab-parallel CE has no ignore_index: mask manually
            per_token = per_token * valid
            loss = per_token.sum() / valid.sum().clamp(min=1)

        logits = logits_parallel
        if groups.get_tensor_model_parallel_world_size() > 1 and labels is None:
            # inference path gathers full logits for sampling
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return CausalLMOutputWithPast(
            loss=loss, logits=logits, past_key_values=out.past_key_values)

    def prepare_inputs_for_generation(self, input_ids, past_key_values=None,
                                      attention_mask=None, **kwargs):
        if past_key_values is not None and past_key_values.get_seq_length() > 0:
            input_ids = input_ids[:, past_key_values.get_seq_length():]
        return {"input_ids": input_ids, "past_key_values": past_key_values,
                "attention_mask": attention_mask,
                "use_cache": kwargs.get("use_cache", True)}
