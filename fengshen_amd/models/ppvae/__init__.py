from fengshen_amd.models.ppvae.modeling_ppvae import PPVAEModel, PPVAEConfig  # noqa: F401
