from fengshen_amd.models.deep_vae.modeling_deep_vae import (  # noqa: F401
    DeepVAEConfig,
    DeepVAEModel,
)
