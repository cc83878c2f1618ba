from fengshen_amd.models.gavae.modeling_gavae import GAVAEModel, GAVAEConfig  # noqa: F401
