"""PPVAE / GAVAE latent-plugin demo.

Behavioral parity: reference examples/PPVAE + examples/GAVAE generate
scripts — train a small plug-in over a frozen base VAE's latent space
(PPVAE: conditional bottleneck VAE; GAVAE: GAN + classifier over
latents), then sample new latents and decode with the frozen DAVAE.

Run:  python latent_plugins.py [--steps 50]
Random-weight tiny models (smoke mode; real checkpoints via
from_pretrained).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

from fengshen_amd.models.davae.modeling_davae import DAVAEModel, davae_tiny_config
from fengshen_amd.models.gavae.modeling_gavae import GAVAEConfig, GAVAEModel
from fengshen_amd.models.ppvae.modeling_ppvae import PPVAEConfig, PPVAEModel


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--steps", default=50, type=int)
    args = parser.parse_args()
    torch.manual_seed(0)

    base = DAVAEModel(davae_tiny_config()).float().eval()
    latent_dim = base.config.latent_dim

    # harvest "real" latents from the frozen base VAE
    with torch.no_grad():
        x = torch.randint(5, 100, (64, 16))
        mu, _ = base.encode(x)

    # --- PPVAE: plug-in conditional VAE over the latent space ---------
    ppvae = PPVAEModel(PPVAEConfig(latent_dim=latent_dim)).float()
    opt = torch.optim.AdamW(ppvae.parameters(), lr=1e-3)
    for step in range(args.steps):
        out = ppvae(mu)
        opt.zero_grad()
        out.loss.backward()
        opt.step()
    print(f"ppvae final loss {out.loss.item():.4f} "
          f"(kl {out.kl_loss.item():.4f})")
    # sample the bottleneck prior -> latent -> decode one step
    z = ppvae.gen_latent(2)
    bos = torch.full((2, 1), 5, dtype=torch.long)
    h = base.decode(z, bos)
    print("ppvae-decoded first-step hidden norm:",
          round(h[:, -1].float().norm().item(), 3))

    # --- GAVAE: GAN over latents with label conditioning --------------
    gavae = GAVAEModel(GAVAEConfig(latent_dim=latent_dim)).float()
    opt_g = torch.optim.AdamW(
        list(gavae.generator.parameters())
        + list(gavae.label_emb.parameters()), lr=1e-3)
    opt_d = torch.optim.AdamW(gavae.discriminator.parameters(), lr=1e-3)
    labels = torch.randint(0, 2, (mu.shape[0],))
    for step in range(args.steps):
        out = gavae(mu, labels)
        opt_d.zero_grad()
        out.d_loss.backward()
        opt_d.step()
        out = gavae(mu, labels)
        opt_g.zero_grad()
        out.g_loss.backward()
        opt_g.step()
    print(f"gavae final g_loss {out.g_loss.item():.4f} "
          f"d_loss {out.d_loss.item():.4f}")
    z = gavae.generate_latent(torch.tensor([0, 1]))
    h = base.decode(z, bos)
    print("gavae-decoded first-step hidden norm:",
          round(h[:, -1].float().norm().item(), 3))


if __name__ == "__main__":
    main()
