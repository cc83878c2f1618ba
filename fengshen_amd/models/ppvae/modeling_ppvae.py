"""PPVAE: plug-in conditional VAE over a frozen pretrained VAE's latent.

Behavioral parity: reference models/PPVAE (232 LoC) — a small MLP
encoder/decoder ("plug-in") maps the big VAE's latent space to a compact
conditional space; only the plug-in trains.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput


class PPVAEConfig(PretrainedConfig):
    model_type = "fengshen_ppvae"

    def __init__(self, latent_dim: int = 32, bottleneck_dim: int = 8,
                 hidden_dim: int = 64, beta_kl: float = 1.0, **kw):
        self.latent_dim = latent_dim
        self.bottleneck_dim = bottleneck_dim
        self.hidden_dim = hidden_dim
        self.beta_kl = beta_kl
        super().__init__(**kw)


@dataclass
class PPVAEOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    rec_loss: Optional[torch.Tensor] = None
    kl_loss: Optional[torch.Tensor] = None
    z_rec: Optional[torch.Tensor] = None


class PPVAEModel(PreTrainedModel):
    config_class = PPVAEConfig

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, 0.02)
            if module.bias is not None:
                module.bias.data.zero_()

    def __init__(self, config: PPVAEConfig):
        super().__init__(config)
        d, b, h = config.latent_dim, config.bottleneck_dim, config.hidden_dim
        self.encoder = nn.Sequential(
            nn.Linear(d, h), nn.GELU(), nn.Linear(h, 2 * b))
        self.decoder = nn.Sequential(
            nn.Linear(b, h), nn.GELU(), nn.Linear(h, d))
        self.post_init()

    def forward(self, latent: torch.Tensor, **_kw):
        mu, logvar = self.encoder(latent).chunk(2, dim=-1)
        z = mu + torch.exp(0.5 * logvar) * torch.randn_like(mu) \
            if self.training else mu
        z_rec = self.decoder(z)
        rec = nn.functional.mse_loss(z_rec.float(), latent.float())
        kl = -0.5 * (1 + logvar - mu.pow(2) - logvar.exp()).sum(-1).mean()
        loss = rec + self.config.beta_kl * kl
        return PPVAEOutput(loss=loss, rec_loss=rec, kl_loss=kl, z_rec=z_rec)

    @torch.no_grad()
    def sample_latent(self, n: int, device=None):
        device = device or next(self.parameters()).device
        b = torch.randn(n, self.config.bottleneck_dim, device=device)
        return self.decoder(b)
