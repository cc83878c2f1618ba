from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (  # noqa: F401
    MegatronBertConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (  # noqa: F401
    MegatronBertModel,
    MegatronBertForPreTraining,
    MegatronBertForMaskedLM,
    MegatronBertForSequenceClassification,
    MegatronBertForTokenClassification,
)
