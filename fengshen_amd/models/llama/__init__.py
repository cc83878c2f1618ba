from fengshen_amd.models.llama.configuration_llama import LlamaConfig  # noqa: F401
from fengshen_amd.models.llama.modeling_llama import (  # noqa: F401
    LlamaModel,
    LlamaForCausalLM,
)
