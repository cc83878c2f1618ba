from fengshen_amd.models.roformer.modeling_roformer import (  # noqa: F401
    RoFormerConfig,
    RoFormerModel,
    RoFormerForMaskedLM,
    RoFormerForSequenceClassification,
)
