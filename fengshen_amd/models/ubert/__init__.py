from fengshen_amd.models.ubert.modeling_ubert import (  # noqa: F401
    UbertConfig,
    UbertModel,
)
