from fengshen_amd.models.davae.modeling_davae import (  # noqa: F401
    DAVAEConfig,
    DAVAEModel,
)
