"""Transfo-XL reasoning variant (ref models/transfo_xl_reasoning)."""
from fengshen_amd.models.transfo_xl_denoise.modeling_transfo_xl_denoise import (  # noqa: F401
    TransfoXLDenoiseModel as TransfoXLModel,
)

from .generate import (  # noqa: F401
    abduction_generate,
    deduction_generate,
    en_to_zh,
)
