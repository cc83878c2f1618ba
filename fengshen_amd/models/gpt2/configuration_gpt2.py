"""Wenzhong-GPT2 config (the reference serves Wenzhong via HF GPT2,
examples/wenzhong_qa; sizes here are the published Wenzhong-GPT2-3.5B)."""
from transformers import PretrainedConfig


class GPT2Config(PretrainedConfig):
    model_type = "fengshen_gpt2"

    def __init__(self,
                 vocab_size: int = 50304,
                 hidden_size: int = 768,
                 num_hidden_layers: int = 12,
                 num_attention_heads: int = 12,
                 max_position_embeddings: int = 1024,
                 layer_norm_epsilon: float = 1e-5,
                 initializer_range: float = 0.02,
                 hidden_dropout: float = 0.0,
                 attention_dropout: float = 0.0,
                 embedding_dropout: float = 0.0,
                 use_cache: bool = True,
                 tie_word_embeddings: bool = True,
                 bos_token_id: int = 50256,
                 eos_token_id: int = 50256,
                 torch_dtype="bfloat16",
                 **kwargs):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        self.embedding_dropout = embedding_dropout
        self.use_cache = use_cache
        super().__init__(bos_token_id=bos_token_id, eos_token_id=eos_token_id,
                         tie_word_embeddings=tie_word_embeddings,
                         torch_dtype=torch_dtype, **kwargs)


def wenzhong_gpt2_3b5_config(**over) -> GPT2Config:
    """Wenzhong-GPT2-3.5B (BASELINE config 3)."""
    cfg = dict(vocab_size=50304, hidden_size=3072, num_hidden_layers=30,
               num_attention_heads=32, max_position_embeddings=1024)
    cfg.update(over)
    return GPT2Config(**cfg)


def gpt2_tiny_config(**over) -> GPT2Config:
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, max_position_embeddings=128)
    cfg.update(over)
    return GPT2Config(**cfg)
