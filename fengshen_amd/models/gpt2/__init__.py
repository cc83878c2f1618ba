from fengshen_amd.models.gpt2.configuration_gpt2 import GPT2Config  # noqa: F401
from fengshen_amd.models.gpt2.modeling_gpt2 import (  # noqa: F401
    GPT2Model,
    GPT2LMHeadModel,
)
