"""AutoencoderKL — the SD latent VAE (8x spatial downsample).

Behavioral parity: the reference splits diffusers' pipeline into
tokenizer/text_encoder/vae/unet (finetune.py:81-87); this is our own
compact KL autoencoder with the standard 0.18215 latent scaling.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel


class VAEConfig(PretrainedConfig):
    model_type = "fengshen_sd_vae"

    def __init__(self, in_channels: int = 3, latent_channels: int = 4,
                 base_channels: int = 32, scaling_factor: float = 0.18215,
                 **kw):
        self.in_channels = in_channels
        self.latent_channels = latent_channels
        self.base_channels = base_channels
        self.scaling_factor = scaling_factor
        super().__init__(**kw)


def _block(cin, cout, down=False, up=False):
    layers = []
    if up:
        layers.append(nn.Upsample(scale_factor=2, mode="nearest"))
    layers += [nn.Conv2d(cin, cout, 3, stride=2 if down else 1, padding=1),
               nn.GroupNorm(min(8, cout), cout), nn.SiLU()]
    return nn.Sequential(*layers)


class AutoencoderKL(PreTrainedModel):
    config_class = VAEConfig

    def _init_weights(self, module):
        pass

    def __init__(self, config: VAEConfig = None):
        config = config or VAEConfig()
        super().__init__(config)
        c = config.base_channels
        self.encoder = nn.Sequential(
            _block(config.in_channels, c),
            _block(c, c, down=True),
            _block(c, 2 * c, down=True),
            _block(2 * c, 4 * c, down=True),
            nn.Conv2d(4 * c, 2 * config.latent_channels, 3, padding=1))
        self.decoder = nn.Sequential(
            nn.Conv2d(config.latent_channels, 4 * c, 3, padding=1),
            _block(4 * c, 2 * c, up=True),
            _block(2 * c, c, up=True),
            _block(c, c, up=True),
            nn.Conv2d(c, config.in_channels, 3, padding=1))
        self.post_init()

    def encode(self, pixels: torch.Tensor,
               sample: bool = True) -> torch.Tensor:
        mu, logvar = self.encoder(pixels).chunk(2, dim=1)
        z = mu + torch.exp(0.5 * logvar) * torch.randn_like(mu) \
            if sample else mu
        return z * self.config.scaling_factor

    def decode(self, latents: torch.Tensor) -> torch.Tensor:
        return self.decoder(latents / self.config.scaling_factor)

    def forward(self, pixels: torch.Tensor):
        mu, logvar = self.encoder(pixels).chunk(2, dim=1)
        z = mu + torch.exp(0.5 * logvar) * torch.randn_like(mu)
        rec = self.decoder(z)
        rec_loss = nn.functional.mse_loss(rec.float(), pixels.float())
        kl = -0.5 * (1 + logvar - mu.pow(2)
                     - logvar.exp()).mean()
        return rec, rec_loss + 1e-6 * kl
