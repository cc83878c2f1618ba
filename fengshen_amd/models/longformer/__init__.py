from fengshen_amd.models.longformer.modeling_longformer import (  # noqa: F401
    LongformerConfig,
    LongformerModel,
    LongformerForMaskedLM,
)
