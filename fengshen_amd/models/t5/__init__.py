from fengshen_amd.models.t5.modeling_t5 import (  # noqa: F401
    T5Config,
    T5Model,
    T5ForConditionalGeneration,
    randeng_t5_77m_config,
    randeng_t5_784m_config,
)
