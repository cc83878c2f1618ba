from fengshen_amd.models.transfo_xl_denoise.modeling_transfo_xl_denoise import (  # noqa: F401
    TransfoXLDenoiseConfig,
    TransfoXLDenoiseModel,
)
