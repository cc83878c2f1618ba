"""DDPM/DDIM noise scheduler for the Taiyi-SD training/sampling path.

Behavioral parity: the reference uses diffusers' DDPMScheduler.add_noise in
finetune_taiyi_stable_diffusion/finetune.py:112-152; this is our own.
"""
from __future__ import annotations

import torch


class DDPMScheduler:
    def __init__(self, num_train_timesteps: int = 1000,
                 beta_start: float = 0.00085, beta_end: float = 0.012,
                 beta_schedule: str = "scaled_linear"):
        self.num_train_timesteps = num_train_timesteps
        if beta_schedule == "scaled_linear":
            betas = torch.linspace(beta_start ** 0.5, beta_end ** 0.5,
                                   num_train_timesteps) ** 2
        else:
            betas = torch.linspace(beta_start, beta_end, num_train_timesteps)
        self.betas = betas
        self.alphas = 1.0 - betas
        self.alphas_cumprod = torch.cumprod(self.alphas, dim=0)

    def add_noise(self, sample: torch.Tensor, noise: torch.Tensor,
                  timesteps: torch.Tensor) -> torch.Tensor:
        ac = self.alphas_cumprod.to(sample.device)[timesteps].float()
        while ac.dim() < sample.dim():
            ac = ac.unsqueeze(-1)
        return (ac.sqrt() * sample.float()
                + (1 - ac).sqrt() * noise.float()).to(sample.dtype)

    @torch.no_grad()
    def step(self, model_output: torch.Tensor, t: int,
             sample: torch.Tensor) -> torch.Tensor:
        """one reverse DDPM step (epsilon parameterization)."""
        beta_t = self.betas[t]
        alpha_t = self.alphas[t]
        ac_t = self.alphas_cumprod[t]
        x0 = (sample.float() - (1 - ac_t).sqrt() * model_output.float()) \
            / ac_t.sqrt()
        x0 = x0.clamp(-4, 4)
        if t == 0:
            return x0.to(sample.dtype)
        ac_prev = self.alphas_cumprod[t - 1]
        coef_x0 = ac_prev.sqrt() * beta_t / (1 - ac_t)
        coef_xt = alpha_t.sqrt() * (1 - ac_prev) / (1 - ac_t)
        mean = coef_x0 * x0 + coef_xt * sample.float()
        var = beta_t * (1 - ac_prev) / (1 - ac_t)
        return (mean + var.sqrt() * torch.randn_like(mean)).to(sample.dtype)
