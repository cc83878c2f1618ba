"""Erlangshen-MegatronBERT config.

The reference trains Erlangshen with HF MegatronBertForPreTraining
(examples/pretrain_erlangshen_bert/pretrain_erlangshen.py:138-141).
Named sizes: Erlangshen-MegatronBert-1.3B (README.md:146).
"""
from transformers import PretrainedConfig


class MegatronBertConfig(PretrainedConfig):
    model_type = "fengshen_megatron_bert"

    def __init__(self,
                 vocab_size: int = 21248,
                 hidden_size: int = 768,
                 num_hidden_layers: int = 12,
                 num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 max_position_embeddings: int = 512,
                 type_vocab_size: int = 2,
                 layer_norm_eps: float = 1e-12,
                 initializer_range: float = 0.02,
                 hidden_dropout: float = 0.1,
                 attention_dropout: float = 0.1,
                 pad_token_id: int = 0,
                 torch_dtype="bfloat16",
                 **kwargs):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        super().__init__(pad_token_id=pad_token_id, torch_dtype=torch_dtype,
                         **kwargs)


def erlangshen_1b3_config(**over) -> MegatronBertConfig:
    """Erlangshen-MegatronBert-1.3B (BASELINE config 2)."""
    cfg = dict(vocab_size=21248, hidden_size=2048, num_hidden_layers=24,
               num_attention_heads=32, intermediate_size=8192,
               max_position_embeddings=512)
    cfg.update(over)
    return MegatronBertConfig(**cfg)


def erlangshen_base_config(**over) -> MegatronBertConfig:
    """Erlangshen-MegatronBert-110M-ish base (BASELINE config 1)."""
    cfg = dict(vocab_size=21248, hidden_size=768, num_hidden_layers=12,
               num_attention_heads=12, intermediate_size=3072,
               max_position_embeddings=512)
    cfg.update(over)
    return MegatronBertConfig(**cfg)


def bert_tiny_config(**over) -> MegatronBertConfig:
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=128)
    cfg.update(over)
    return MegatronBertConfig(**cfg)
