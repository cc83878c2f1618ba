from fengshen_amd.models.deberta_v2.modeling_deberta_v2 import (  # noqa: F401
    DebertaV2Config,
    DebertaV2Model,
    DebertaV2ForMaskedLM,
    DebertaV2ForSequenceClassification,
)
