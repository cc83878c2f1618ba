from fengshen_amd.models.bart.modeling_bart import (  # noqa: F401
    BartConfig,
    BartForConditionalGeneration,
    randeng_bart_139m_config,
)
