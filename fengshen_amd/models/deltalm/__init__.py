from fengshen_amd.models.deltalm.modeling_deltalm import (  # noqa: F401
    DeltaLMConfig,
    DeltaLMForConditionalGeneration,
)
