"""PPVAE: plug-in conditional VAE over a frozen pretrained VAE's latent.

Behavioral parity with reference models/PPVAE/pluginVAE.py:
- Encoder (:13-44): latent -> d/2 -> d/4 leaky-relu funnel with separate
  mean / log_var heads; eval mode returns the mean;
- Decoder (:46-57): mirrored expansion;
- PluginVAE loss (:75-79): z-reconstruction MSE +
  kl_weight * |KL - beta|  (free-bits-style absolute constraint);
- train_plugin (:94-160): positive/negative conditional training —
  loss = pos_loss - gamma * neg_loss, with the negative term DETACHED
  when it exceeds neg_loss_threshold * pos_loss; dynamic beta warmup
  (get_beta_weight :178-180); early stopping on the average loss;
- generate (:162-165) / gen_latent (:172-176): sample the bottleneck,
  decode to the big-VAE latent, decode text through the frozen VAE.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput


class PPVAEConfig(PretrainedConfig):
    model_type = "fengshen_ppvae"

    def __init__(self, latent_dim: int = 128, bottle_dim: int = 20,
                 kl_weight: float = 1.0, beta: float = 1.0,
                 ppvae_lr: float = 5e-4, mu: float = 0.9, nu: float = 0.999,
                 gamma: float = 1.0, neg_loss_threshold: float = 1.5,
                 get_dymanic_beta: bool = False,
                 beta_total_step: int = 500, batch_size: int = 32,
                 total_epoch: int = 20, patience: int = 5, **kw):
        self.latent_dim = latent_dim
        self.bottle_dim = bottle_dim
        self.kl_weight = kl_weight
        self.beta = beta
        self.ppvae_lr = ppvae_lr
        self.mu = mu
        self.nu = nu
        self.gamma = gamma
        self.neg_loss_threshold = neg_loss_threshold
        self.get_dymanic_beta = get_dymanic_beta
        self.beta_total_step = beta_total_step
        self.batch_size = batch_size
        self.total_epoch = total_epoch
        self.patience = patience
        super().__init__(**kw)


@dataclass
class PPVAEOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    rec_loss: Optional[torch.Tensor] = None
    kl_loss: Optional[torch.Tensor] = None
    z_rec: Optional[torch.Tensor] = None


class Encoder(nn.Module):
    """Funnel encoder with separate mean/log_var heads (ref :13-44)."""

    def __init__(self, latent_dim: int = 128, bottle_dim: int = 20):
        super().__init__()
        self.fc1 = nn.Linear(latent_dim, latent_dim // 2)
        self.fc2 = nn.Linear(latent_dim // 2, latent_dim // 4)
        self.mean = nn.Linear(latent_dim // 4, bottle_dim)
        self.log_var = nn.Linear(latent_dim // 4, bottle_dim)

    @staticmethod
    def kl_loss(mean, log_var):
        return (-0.5 * (1 + log_var - mean ** 2
                        - log_var.exp()).sum(-1)).mean()

    @staticmethod
    def sampling(mean, log_var):
        return mean + (log_var / 2).exp() * torch.randn_like(mean)

    def forward(self, z):
        z = F.leaky_relu(self.fc1(z))
        z = F.leaky_relu(self.fc2(z))
        z_mean = self.mean(z)
        z_log_var = self.log_var(z)
        kl = self.kl_loss(z_mean, z_log_var)
        enc_z = self.sampling(z_mean, z_log_var) if self.training else z_mean
        return enc_z, kl


class Decoder(nn.Module):
    def __init__(self, latent_dim: int = 128, bottle_dim: int = 20):
        super().__init__()
        self.fc1 = nn.Linear(bottle_dim, latent_dim // 4)
        self.fc2 = nn.Linear(latent_dim // 4, latent_dim // 2)
        self.fc3 = nn.Linear(latent_dim // 2, latent_dim)

    def forward(self, enc_z):
        z = F.leaky_relu(self.fc1(enc_z))
        z = F.leaky_relu(self.fc2(z))
        return self.fc3(z)


class PluginVAE(nn.Module):
    """Plug-in inner VAE with the |KL - beta| constraint (ref :59-79)."""

    def __init__(self, config: PPVAEConfig):
        super().__init__()
        self.kl_weight = config.kl_weight
        self.beta = config.beta
        self.encoder = Encoder(config.latent_dim, config.bottle_dim)
        self.decoder = Decoder(config.latent_dim, config.bottle_dim)

    def set_beta(self, beta: float):
        self.beta = beta

    def forward(self, z):
        enc_z, kl = self.encoder(z)
        return self.decoder(enc_z), kl

    def loss(self, z):
        z_out, kl = self.forward(z)
        z_loss = ((z_out - z) ** 2).mean()
        return z_loss + self.kl_weight * (kl - self.beta).abs(), kl


class _EarlyStopping:
    def __init__(self, patience: int = 5):
        self.patience = patience
        self.counter = 0
        self.early_stop = False

    def __call__(self, loss: float, min_loss: float):
        if loss > min_loss:
            self.counter += 1
            if self.counter >= self.patience:
                self.early_stop = True


class PPVAEModel(PreTrainedModel):
    config_class = PPVAEConfig

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, 0.02)
            if module.bias is not None:
                module.bias.data.zero_()

    def __init__(self, config: PPVAEConfig, vae_model=None):
        super().__init__(config)
        self.pluginvae = PluginVAE(config)
        # frozen big VAE (DAVAEModel-compatible: encode/sample interfaces);
        # optional so the plug-in also works on raw latent tensors
        self.vae_model = vae_model
        self.post_init()

    # -- latent-tensor interface (also used by the trainer tests) -------
    def forward(self, latent: torch.Tensor, **_kw):
        loss, kl = self.pluginvae.loss(latent)
        z_rec, _ = self.pluginvae(latent)
        return PPVAEOutput(loss=loss, kl_loss=kl, z_rec=z_rec)

    @staticmethod
    def get_beta_weight(iter_num: int, beta: float,
                        total_step: int) -> float:
        return min((beta / total_step) * iter_num, beta)

    # -- reference conditional training loop (ref :94-160) --------------
    def train_plugin(self, pos_latents: torch.Tensor,
                     neg_latents: Optional[torch.Tensor] = None,
                     log: Optional[List[float]] = None):
        """Train the plug-in on positive (conditional) latents, pushed
        away from negatives: loss = pos - gamma * neg, neg detached when
        neg > threshold * pos."""
        cfg = self.config
        opt = torch.optim.Adam(self.pluginvae.parameters(),
                               lr=cfg.ppvae_lr, betas=(cfg.mu, cfg.nu))
        pos_loader = torch.utils.data.DataLoader(
            pos_latents, batch_size=cfg.batch_size, shuffle=True)
        neg_loader = None
        if neg_latents is not None:
            nb = max(int(cfg.batch_size
                         * (len(neg_latents) / max(len(pos_latents), 1))), 1)
            neg_loader = torch.utils.data.DataLoader(
                neg_latents, batch_size=nb, shuffle=True)
        stopper = _EarlyStopping(cfg.patience)
        min_loss = float("inf")
        it = 0
        for _epoch in range(cfg.total_epoch):
            self.pluginvae.train()
            total = 0.0
            for data in pos_loader:
                if cfg.get_dymanic_beta:
                    self.pluginvae.set_beta(self.get_beta_weight(
                        it, cfg.beta, cfg.beta_total_step))
                it += 1
                pos_loss, _pos_kl = self.pluginvae.loss(data)
                neg_loss = 0.0
                if neg_loader is not None:
                    neg_data = next(iter(neg_loader))
                    neg_loss, _ = self.pluginvae.loss(neg_data)
                    if neg_loss.item() > cfg.neg_loss_threshold \
                            * pos_loss.item():
                        neg_loss = neg_loss.detach()
                loss = pos_loss - cfg.gamma * neg_loss
                opt.zero_grad()
                loss.backward()
                opt.step()
                total += float(loss)
            avg = total / max(len(pos_loader), 1)
            if log is not None:
                log.append(avg)
            if avg < min_loss:
                min_loss = avg
                stopper.counter = 0
            stopper(avg, min_loss)
            if stopper.early_stop:
                break

    # -- generation ------------------------------------------------------
    @torch.no_grad()
    def gen_latent(self, gen_num: int = 5, device=None):
        device = device or next(self.parameters()).device
        rand = torch.randn(gen_num, self.config.bottle_dim, device=device)
        return self.pluginvae.decoder(rand)

    @torch.no_grad()
    def sample_latent(self, n: int, device=None):  # round-1 alias
        return self.gen_latent(n, device)

    @torch.no_grad()
    def generate(self, n: int, seq_len: int = 32, bos_id: int = 5):
        """Sample bottleneck -> big-VAE latent -> text ids through the
        frozen VAE (ref :162-165)."""
        assert self.vae_model is not None, "attach a DAVAE-style vae_model"
        z = self.gen_latent(n)
        return self.vae_model.sample_from_latent(z, seq_len, bos_id=bos_id)
