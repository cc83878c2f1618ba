"""Randeng-T5 (megatron-style T5): enc-dec with absolute positions, standard
LayerNorm, biased dense layers.

Behavioral parity: reference models/megatron_t5/modeling_megatron_t5.py —
the @IDEA-modified HF T5: dense bias=True (:261-347), T5LayerNorm ->
nn.LayerNorm (:312-315), relative position bias dropped for absolute
embeddings (:551, :917).  Ours is built from our parallel encoder/decoder
layer library instead of patched HF code.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.generation import GenerationMixin
from transformers.modeling_outputs import Seq2SeqLMOutput, Seq2SeqModelOutput

from fengshen_amd.models.encoder_decoder import DecoderLayer, EncoderLayer
from fengshen_amd.models.layers import (
    LayerNorm,
    init_normal,
    parallel_lm_logits,
    scaled_init_normal,
)
from fengshen_amd.parallel import groups
from fengshen_amd.parallel.cross_entropy import vocab_parallel_cross_entropy
from fengshen_amd.parallel.layers import VocabParallelEmbedding
from fengshen_amd.parallel.mappings import gather_from_tensor_model_parallel_region


class T5Config(PretrainedConfig):
    model_type = "fengshen_t5"

    def __init__(self, vocab_size: int = 32596, hidden_size: int = 512,
                 num_layers: int = 8, num_decoder_layers: Optional[int] = None,
                 num_attention_heads: int = 8, intermediate_size: int = 2048,
                 max_position_embeddings: int = 512,
                 layer_norm_epsilon: float = 1e-5,
                 initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1, attention_dropout: float = 0.1,
                 decoder_start_token_id: int = 0, pad_token_id: int = 0,
                 eos_token_id: int = 1, tie_word_embeddings: bool = True,
                 torch_dtype="bfloat16", **kw):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_layers = num_layers
        self.num_hidden_layers = num_layers  # HF generation utils expect this
        self.num_decoder_layers = num_decoder_layers or num_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(decoder_start_token_id=decoder_start_token_id,
                         pad_token_id=pad_token_id, eos_token_id=eos_token_id,
                         tie_word_embeddings=tie_word_embeddings,
                         torch_dtype=torch_dtype,
                         is_encoder_decoder=True, **kw)


def randeng_t5_77m_config(**over):
    cfg = dict(vocab_size=32596, hidden_size=512, num_layers=8,
               num_attention_heads=6, intermediate_size=1024)
    cfg.update(over)
    return T5Config(**cfg)


def randeng_t5_784m_config(**over):
    cfg = dict(vocab_size=32596, hidden_size=1024, num_layers=24,
               num_attention_heads=16, intermediate_size=2816)
    cfg.update(over)
    return T5Config(**cfg)


def t5_tiny_config(**over):
    cfg = dict(vocab_size=256, hidden_size=64, num_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=128)
    cfg.update(over)
    return T5Config(**cfg)


class T5PreTrainedModel(PreTrainedModel):
    config_class = T5Config
    base_model_prefix = "t5"

    def _init_weights(self, module):
        if isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, self.config.initializer_range)


class T5Model(T5PreTrainedModel):
    def __init__(self, config: T5Config):
        super().__init__(config)
        im = init_normal(config.initializer_range)
        om = scaled_init_normal(config.initializer_range, config.num_layers)
        self.shared = VocabParallelEmbedding(config.vocab_size,
                                             config.hidden_size, init_method=im)
        self.enc_pos = nn.Embedding(config.max_position_embeddings,
                                    config.hidden_size)
        self.dec_pos = nn.Embedding(config.max_position_embeddings,
                                    config.hidden_size)
        self.encoder = nn.ModuleList([
            EncoderLayer(config.hidden_size, config.num_attention_heads,
                         config.intermediate_size,
                         norm_eps=config.layer_norm_epsilon, prenorm=True,
                         attention_dropout=config.attention_dropout,
                         hidden_dropout=config.hidden_dropout,
                         init_method=im, output_init_method=om, layer_idx=i)
            for i in range(config.num_layers)])
        self.encoder_ln = LayerNorm(config.hidden_size,
                                    eps=config.layer_norm_epsilon)
        self.decoder = nn.ModuleList([
            DecoderLayer(config.hidden_size, config.num_attention_heads,
                         config.intermediate_size,
                         norm_eps=config.layer_norm_epsilon, prenorm=True,
                         attention_dropout=config.attention_dropout,
                         hidden_dropout=config.hidden_dropout,
                         init_method=im, output_init_method=om, layer_idx=i)
            for i in range(config.num_decoder_layers)])
        self.decoder_ln = LayerNorm(config.hidden_size,
                                    eps=config.layer_norm_epsilon)
        self.dropout = nn.Dropout(config.hidden_dropout)
        self.post_init()

    def get_input_embeddings(self):
        return self.shared

    def set_input_embeddings(self, v):
        self.shared = v

    def encode(self, input_ids, attention_mask=None):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        h = self.dropout(self.shared(input_ids) + self.enc_pos(pos))
        mask = (attention_mask == 0)[:, None, None, :] \
            if attention_mask is not None else None
        for layer in self.encoder:
            h = layer(h, attention_mask=mask)
        return self.encoder_ln(h)

    def decode(self, decoder_input_ids, encoder_states, attention_mask=None,
               cache=None):
        # HF generate hands enc-dec models an EncoderDecoderCache; our layers
        # only cache decoder self-attention (cross K/V are cheap to recompute)
        if cache is not None and hasattr(cache, "self_attention_cache"):
            cache = cache.self_attention_cache
        b, s = decoder_input_ids.shape
        offset = cache.get_seq_length() if cache is not None else 0
        pos = torch.arange(offset, offset + s,
                           device=decoder_input_ids.device).unsqueeze(0)
        h = self.dropout(self.shared(decoder_input_ids) + self.dec_pos(pos))
        cross_mask = (attention_mask == 0)[:, None, None, :] \
            if attention_mask is not None else None
        for layer in self.decoder:
            h = layer(h, encoder_states, self_mask=None, cross_mask=cross_mask,
                      cache=cache)
        return self.decoder_ln(h)

    def forward(self, input_ids, decoder_input_ids, attention_mask=None,
                **_kw):
        enc = self.encode(input_ids, attention_mask)
        dec = self.decode(decoder_input_ids, enc, attention_mask)
        return Seq2SeqModelOutput(last_hidden_state=dec,
                                  encoder_last_hidden_state=enc)


class T5ForConditionalGeneration(T5PreTrainedModel, GenerationMixin):
    main_input_name = "input_ids"

    def __init__(self, config: T5Config):
        super().__init__(config)
        self.t5 = T5Model(config)
        self.post_init()

    def get_input_embeddings(self):
        return self.t5.shared

    def get_encoder(self):
        outer = self

        class _Enc(nn.Module):
            main_input_name = "input_ids"

            def forward(self, input_ids=None, attention_mask=None, **kw):
                from transformers.modeling_outputs import BaseModelOutput
                return BaseModelOutput(
                    last_hidden_state=outer.t5.encode(input_ids, attention_mask))
        return _Enc()

    def prepare_inputs_for_generation(self, input_ids, past_key_values=None,
                                      attention_mask=None,
                                      encoder_outputs=None, **kw):
        decoder_input_ids = input_ids
        if past_key_values is not None and past_key_values.get_seq_length() > 0:
            decoder_input_ids = decoder_input_ids[
                :, past_key_values.get_seq_length():]
        return {"decoder_input_ids": decoder_input_ids,
                "encoder_outputs": encoder_outputs,
                "attention_mask": attention_mask,
                "past_key_values": past_key_values,
                "use_cache": kw.get("use_cache", True)}

    def _shift_right(self, labels):
        shifted = labels.new_zeros(labels.shape)
        shifted[:, 1:] = labels[:, :-1].clone()
        shifted[:, 0] = self.config.decoder_start_token_id
        shifted.masked_fill_(shifted == -100, self.config.pad_token_id)
        return shifted

    def forward(self, input_ids=None, attention_mask=None,
                decoder_input_ids=None, labels=None, encoder_outputs=None,
                past_key_values=None, use_cache=False, return_dict=True, **_kw):
        if decoder_input_ids is None and labels is not None:
            decoder_input_ids = self._shift_right(labels)
        if encoder_outputs is None:
            enc = self.t5.encode(input_ids, attention_mask)
        else:
            enc = encoder_outputs.last_hidden_state \
                if hasattr(encoder_outputs, "last_hidden_state") \
                else encoder_outputs[0]
        cache = past_key_values
        if use_cache and cache is None:
            from transformers.cache_utils import DynamicCache
            cache = DynamicCache()
        dec = self.t5.decode(decoder_input_ids, enc, attention_mask,
                             cache=cache)
        logits_parallel = parallel_lm_logits(dec, self.t5.shared.weight,
                                             parallel_output=True)
        loss = None
        if labels is not None:
            per_token = vocab_parallel_cross_entropy(
                logits_parallel, labels.clamp(min=0))
            valid = (labels != -100)
            loss = (per_token * valid).sum() / valid.sum().clamp(min=1)
        logits = logits_parallel
        if groups.get_tensor_model_parallel_world_size() > 1 and labels is None:
            logits = gather_from_tensor_model_parallel_region(logits_parallel)
        return Seq2SeqLMOutput(loss=loss, logits=logits,
                               past_key_values=cache,
                               encoder_last_hidden_state=enc)
