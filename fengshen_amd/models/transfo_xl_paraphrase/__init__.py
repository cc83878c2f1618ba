"""Transfo-XL paraphrase variant (ref models/transfo_xl_paraphrase)."""
from fengshen_amd.models.transfo_xl_denoise.modeling_transfo_xl_denoise import (  # noqa: F401
    TransfoXLDenoiseModel as TransfoXLModel,
)

from .generate import paraphrase_generate  # noqa: F401
