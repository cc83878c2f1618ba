from fengshen_amd.models.tcbert.modeling_tcbert import (  # noqa: F401
    TCBertConfig,
    TCBertModel,
)
