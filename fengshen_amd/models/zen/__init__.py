from fengshen_amd.models.zen.modeling_zen import (  # noqa: F401
    ZenConfig,
    ZenModel,
    ZenForSequenceClassification,
    ZenNgramDict,
)
