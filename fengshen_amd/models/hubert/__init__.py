from fengshen_amd.models.hubert.modeling_hubert import (  # noqa: F401
    HubertConfig,
    HubertModel,
    HubertForPreTraining,
    hubert_tiny_config,
)
