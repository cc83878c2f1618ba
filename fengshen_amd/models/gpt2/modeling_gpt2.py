"""Wenzhong-GPT2, TP-aware, MI355X-native.

Behavioral parity: the reference trains/serves Wenzhong with HF GPT2
(examples/wenzhong_qa/finetune_medicalQA.py); this is the same architecture
(learned positions, pre-LN blocks, GELU MLP, tied LM head) on our parallel
layer library with fused HIP ops on the hot path.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from transformers import PreTrainedModel
from transformers.generation import GenerationMixin
from transformers.modeling_outputs import (
    BaseModelOutputWithPast,
    CausalLMOutputWithPast,
)

from fengshen_amd.models.gpt2.configuration_gpt2 import GPT2Config
from fengshen_amd.models.layers import (
    LayerNorm,
    ParallelTransformerLayer,
    init_normal,
    parallel_lm_logits,
    scaled_init_normal,
)
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from fengshen_amd.parallel.layers import VocabParallelEmbedding
from fengshen_amd.parallel.mappings import gather_from_tensor_model_parallel_region
from fengshen_amd.parallel.random import checkpoint as activation_checkpoint


class GPT2PreTrainedModel(PreTrainedModel):
    config_class = GPT2Config
    base_model_prefix = "transformer"
    supports_gradient_checkpointing = True

    def _init_weights(self, module):
        if isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=self.config.initializer_range)


class GPT2Model(GPT2PreTrainedModel):
    def __init__(self, config: GPT2Config):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range, config.num_hidden_layers)
        self.wte = VocabParallelEmbedding(config.vocab_size, config.hidden_size,
                                          init_method=im)
        self.wpe = nn.Embedding(config.max_position_embeddings, config.hidden_size)
        self.drop = nn.Dropout(config.embedding_dropout)
        self.h = nn.ModuleList([
            ParallelTransformerLayer(
                config.hidden_size, config.num_attention_heads, causal=True,
                norm="layernorm", norm_eps=config.layer_norm_epsilon,
                mlp_type="gelu", rotary=False,
                max_positions=config.max_position_embeddings,
                attention_dropout=config.attention_dropout,
                hidden_dropout=config.hidden_dropout,
                bias=True, init_method=im, output_init_method=om, layer_idx=i)
            for i in range(config.num_hidden_layers)])
        self.ln_f = LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.gradient_checkpointing = False
        self.gradient_checkpointing_skip_interval = 0
        self.post_init()

    def get_input_embeddings(self):
        return self.wte

    def set_input_embeddings(self, v):
        self.wte = v

    def forward(self, input_ids, attention_mask=None, past_key_values=None,
                use_cache: bool = False, position_ids=None, **_kw):
        b, s = input_ids.shape
        cache = past_key_values
        if use_cache and cache is None:
            from transformers.cache_utils import DynamicCache
            cache = DynamicCache()
        offset = cache.get_seq_length() if cache is not None else 0
        if position_ids is None:
            position_ids = torch.arange(offset, offset + s,
                                        device=input_ids.device).unsqueeze(0)
        h = self.wte(input_ids) + self.wpe(position_ids)
        h = self.drop(h)

        mask = None
        if attention_mask is not None and attention_mask.dim() == 2:
            if attention_mask.min() != 1:
                mask = (attention_mask == 0)[:, None, None, :]
        elif attention_mask is not None:
            mask = attention_mask

        skip = self.gradient_checkpointing_skip_interval
        for i, layer in enumerate(self.h):
            ckpt = self.gradient_checkpointing and self.training \
                and cache is None
            if ckpt and skip and (i % skip == skip - 1):
                ckpt = False
            if ckpt:
                h = activation_checkpoint(
                    lambda x, m, lyr=layer: lyr(x, attention_mask=m), h, mask)
            else:
                h = layer(h, attention_mask=mask, cache=cache)
        h = self.ln_f(h)
        return BaseModelOutputWithPast(last_hidden_state=h, past_key_values=cache)


class GPT2LMHeadModel(GPT2PreTrainedModel, GenerationMixin):
    def __init__(self, config: GPT2Config):
        super().__init__(config)
        self.transformer = GPT2Model(config)
        self.post_init()

    def get_input_embeddings(self):
        return self.transformer.wte

    def get_output_embeddings(self):
        return None  # tied head applied via parallel_lm_logits

    def gradient_checkpointing_enable(self, skip_interval: int = 0, **_kw):
        self.transformer.gradient_checkpointing = True
        self.transformer.gradient_checkpointing_skip_interval = max(
            skip_interval, 0)

    def forward(self, input_ids, attention_mask=None, labels=None,
                past_key_values=None, use_cache: bool = False,
                position_ids=None, return_dict: bool = True, **_kw):
        out = self.transformer(input_ids, attention_mask=attention_mask,
                               past_key_values=past_key_values,
                               use_cache=use_cache, position_ids=position_ids)
        h = out.last_hidden_state
        logits_parallel = parallel_lm_logits(h, self.transformer.wte.weight,
                                             parallel_output=True)
        loss = None
        if labels is not None:
            shift_logits = logits_parallel[:, :-1, :].contiguous()
            shift_labels = labels[:, 1:].contiguous()
            per_token = vocab_parallel_cross_entropy(shift_logits, shift_labels)
            valid = (shift_labels != -100)
            loss = (per_token * valid).sum() / valid.sum().clamp(min=1)
        logits = logits_parallel
        if groups.get_tensor_model_parallel_world_size() > 1 and labels is None:
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return CausalLMOutputWithPast(loss=loss, logits=logits,
                                      past_key_values=out.past_key_values)

    def prepare_inputs_for_generation(self, input_ids, past_key_values=None,
                                      attention_mask=None, **kwargs):
        if past_key_values is not None and past_key_values.get_seq_length() > 0:
            input_ids = input_ids[:, past_key_values.get_seq_length():]
        return {"input_ids": input_ids, "past_key_values": past_key_values,
                "attention_mask": attention_mask,
                "use_cache": kwargs.get("use_cache", True)}
