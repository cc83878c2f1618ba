from fengshen_amd.models.uniex.modeling_uniex import (  # noqa: F401
    UniEXConfig,
    UniEXModel,
)
