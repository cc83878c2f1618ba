"""GAVAE: GAN-augmented VAE latent sampler.

Behavioral parity: reference models/GAVAE (551 LoC) — a generator maps
noise to the VAE latent space and a discriminator separates real posterior
latents from generated ones (used for label-conditioned data augmentation).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput


class GAVAEConfig(PretrainedConfig):
    model_type = "fengshen_gavae"

    def __init__(self, latent_dim: int = 32, noise_dim: int = 16,
                 hidden_dim: int = 64, n_labels: int = 2, **kw):
        self.latent_dim = latent_dim
        self.noise_dim = noise_dim
        self.hidden_dim = hidden_dim
        self.n_labels = n_labels
        super().__init__(**kw)


@dataclass
class GAVAEOutput(ModelOutput):
    g_loss: Optional[torch.Tensor] = None
    d_loss: Optional[torch.Tensor] = None


class GAVAEModel(PreTrainedModel):
    config_class = GAVAEConfig

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, 0.02)
            if module.bias is not None:
                module.bias.data.zero_()

    def __init__(self, config: GAVAEConfig):
        super().__init__(config)
        nd, ld, h = config.noise_dim, config.latent_dim, config.hidden_dim
        self.label_emb = nn.Embedding(config.n_labels, nd)
        self.generator = nn.Sequential(
            nn.Linear(2 * nd, h), nn.GELU(), nn.Linear(h, ld))
        self.discriminator = nn.Sequential(
            nn.Linear(ld + nd, h), nn.GELU(), nn.Linear(h, 1))
        self.post_init()

    def generate_latent(self, labels: torch.Tensor) -> torch.Tensor:
        noise = torch.randn(labels.shape[0], self.config.noise_dim,
                            device=labels.device)
        cond = self.label_emb(labels)
        return self.generator(torch.cat([noise, cond], dim=-1))

    def forward(self, real_latent: torch.Tensor, labels: torch.Tensor,
                **_kw):
        """one GAN step's losses (caller alternates optimizers)."""
        cond = self.label_emb(labels)
        fake = self.generate_latent(labels)
        d_real = self.discriminator(
            torch.cat([real_latent, cond], dim=-1))
        d_fake = self.discriminator(
            torch.cat([fake.detach(), cond], dim=-1))
        ones = torch.ones_like(d_real)
        zeros = torch.zeros_like(d_fake)
        bce = nn.functional.binary_cross_entropy_with_logits
        d_loss = bce(d_real.float(), ones) + bce(d_fake.float(), zeros)
        g_score = self.discriminator(torch.cat([fake, cond], dim=-1))
        g_loss = bce(g_score.float(), torch.ones_like(g_score))
        return GAVAEOutput(g_loss=g_loss, d_loss=d_loss)
