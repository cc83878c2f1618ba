"""Ziya-LLaMA config.

Behavioral parity: reference models/llama/configuration_llama.py:24
(hidden_act silu, rotary_pct 1.0, rms_norm_epsilon, init from std).
Named sizes for the Fengshenbang releases: Ziya-LLaMA-13B below.
"""
from transformers import PretrainedConfig


class LlamaConfig(PretrainedConfig):
    model_type = "fengshen_llama"

    def __init__(self,
                 vocab_size: int = 39424,  # Ziya extended-zh vocab
                 hidden_size: int = 5120,
                 num_hidden_layers: int = 40,
                 num_attention_heads: int = 40,
                 intermediate_size: int = 13824,
                 max_position_embeddings: int = 2048,
                 rms_norm_epsilon: float = 1e-6,
                 initializer_range: float = 0.02,
                 rotary_emb_base: float = 10000.0,
                 hidden_dropout: float = 0.0,
                 attention_dropout: float = 0.0,
                 use_cache: bool = True,
                 pad_token_id: int = 0,
                 bos_token_id: int = 1,
                 eos_token_id: int = 2,
                 tie_word_embeddings: bool = False,
                 parallel_residual: bool = False,
                 torch_dtype="bfloat16",
                 **kwargs):
        # GPT-J composition with one deferred TP all-reduce per layer
        self.parallel_residual = parallel_residual
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.rms_norm_epsilon = rms_norm_epsilon
        self.initializer_range = initializer_range
        self.rotary_emb_base = rotary_emb_base
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        self.use_cache = use_cache
        super().__init__(pad_token_id=pad_token_id, bos_token_id=bos_token_id,
                         eos_token_id=eos_token_id,
                         tie_word_embeddings=tie_word_embeddings,
                         torch_dtype=torch_dtype, **kwargs)


def ziya_llama_13b_config(**over) -> LlamaConfig:
    """Ziya-LLaMA-13B (the BASELINE.json flagship)."""
    cfg = dict(vocab_size=39424, hidden_size=5120, num_hidden_layers=40,
               num_attention_heads=40, intermediate_size=13824,
               max_position_embeddings=2048)
    cfg.update(over)
    return LlamaConfig(**cfg)


def llama_tiny_config(**over) -> LlamaConfig:
    """For tests."""
    cfg = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=128)
    cfg.update(over)
    return LlamaConfig(**cfg)
