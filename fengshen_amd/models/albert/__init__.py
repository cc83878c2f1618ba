from fengshen_amd.models.albert.modeling_albert import (  # noqa: F401
    AlbertConfig,
    AlbertModel,
    AlbertForMaskedLM,
)
