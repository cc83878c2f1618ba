from fengshen_amd.models.unimc.modeling_unimc import (  # noqa: F401
    UniMCConfig,
    UniMCModel,
)
