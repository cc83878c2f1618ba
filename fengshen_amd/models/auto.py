"""Auto* registration — load any fengshen_amd checkpoint via HF Auto classes.

Behavioral parity: reference models/auto/ (fengshen fork of HF Auto*
registering roformer+longformer, configuration_auto.py:30-35) — ours
registers every model family with the stock HF registries instead of
forking them.  Call register_fengshen_auto_classes() (idempotent), then
AutoConfig/AutoModel.from_pretrained work on our checkpoints.
"""
from __future__ import annotations

_REGISTERED = False


def register_fengshen_auto_classes() -> None:
    global _REGISTERED
    if _REGISTERED:
        return
    from transformers import (
        AutoConfig,
        AutoModel,
        AutoModelForCausalLM,
        AutoModelForMaskedLM,
        AutoModelForSeq2SeqLM,
        AutoModelForSequenceClassification,
    )

    from fengshen_amd.models.llama.configuration_llama import LlamaConfig
    from fengshen_amd.models.llama.modeling_llama import (
        LlamaForCausalLM, LlamaModel)
    from fengshen_amd.models.gpt2.configuration_gpt2 import GPT2Config
    from fengshen_amd.models.gpt2.modeling_gpt2 import (
        GPT2LMHeadModel, GPT2Model)
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        MegatronBertConfig)
    from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
        MegatronBertForMaskedLM, MegatronBertForSequenceClassification,
        MegatronBertModel)
    from fengshen_amd.models.t5.modeling_t5 import (
        T5Config, T5ForConditionalGeneration, T5Model)
    from fengshen_amd.models.bart.modeling_bart import (
        BartConfig, BartForConditionalGeneration)
    from fengshen_amd.models.roformer.modeling_roformer import (
        RoFormerConfig, RoFormerForMaskedLM, RoFormerModel,
        RoFormerForSequenceClassification)
    from fengshen_amd.models.longformer.modeling_longformer import (
        LongformerConfig, LongformerForMaskedLM, LongformerModel)
    from fengshen_amd.models.deberta_v2.modeling_deberta_v2 import (
        DebertaV2Config, DebertaV2ForMaskedLM, DebertaV2Model,
        DebertaV2ForSequenceClassification)
    from fengshen_amd.models.albert.modeling_albert import (
        AlbertConfig, AlbertForMaskedLM, AlbertModel)
    from fengshen_amd.models.zen.modeling_zen import (
        ZenConfig, ZenForSequenceClassification, ZenModel)

    table = [
        (LlamaConfig, LlamaModel, {AutoModelForCausalLM: LlamaForCausalLM}),
        (GPT2Config, GPT2Model, {AutoModelForCausalLM: GPT2LMHeadModel}),
        (MegatronBertConfig, MegatronBertModel,
         {AutoModelForMaskedLM: MegatronBertForMaskedLM,
          AutoModelForSequenceClassification:
              MegatronBertForSequenceClassification}),
        (T5Config, T5Model,
         {AutoModelForSeq2SeqLM: T5ForConditionalGeneration}),
        (BartConfig, None,
         {AutoModelForSeq2SeqLM: BartForConditionalGeneration}),
        (RoFormerConfig, RoFormerModel,
         {AutoModelForMaskedLM: RoFormerForMaskedLM,
          AutoModelForSequenceClassification:
              RoFormerForSequenceClassification}),
        (LongformerConfig, LongformerModel,
         {AutoModelForMaskedLM: LongformerForMaskedLM}),
        (DebertaV2Config, DebertaV2Model,
         {AutoModelForMaskedLM: DebertaV2ForMaskedLM,
          AutoModelForSequenceClassification:
              DebertaV2ForSequenceClassification}),
        (AlbertConfig, AlbertModel,
         {AutoModelForMaskedLM: AlbertForMaskedLM}),
        (ZenConfig, ZenModel,
         {AutoModelForSequenceClassification: ZenForSequenceClassification}),
    ]
    for config_cls, base_cls, heads in table:
        AutoConfig.register(config_cls.model_type, config_cls, exist_ok=True)
        if base_cls is not None:
            AutoModel.register(config_cls, base_cls, exist_ok=True)
        for auto_cls, model_cls in heads.items():
            auto_cls.register(config_cls, model_cls, exist_ok=True)
    _REGISTERED = True
