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


# ---------------------------------------------------------------------------
# Reference-parity GAN stack (ref models/GAVAE/gans_model.py)
# ---------------------------------------------------------------------------
import torch.nn.functional as F  # noqa: E402


class ClsNet(nn.Module):
    """Discriminator/classifier (ref CLS_Net :35-96): z -> 256 relu ->
    64 dropout relu -> cls_num, exposing the first hidden layer for the
    generator's feature matching; self_dis computes the pairwise
    distance matrix used as a diversity feature."""

    def __init__(self, cls_num: int, z_dim: int, cls_batch_size: int = 16):
        super().__init__()
        self.cls_batch_size = cls_batch_size
        self.jie = 1
        self.fc1 = nn.Linear(z_dim, 256)
        self.fc2 = nn.Linear(256, 64)
        self.out = nn.Linear(64, cls_num)
        for m in (self.fc1, self.fc2, self.out):
            m.weight.data.normal_(0, 0.1)

    def self_dis(self, a: torch.Tensor) -> torch.Tensor:
        """Pairwise p=jie distance matrix [n, n] (ref :58-84)."""
        return torch.cdist(a.unsqueeze(0), a.unsqueeze(0),
                           p=float(self.jie)).squeeze(0)

    def forward(self, x):
        x1 = F.relu(self.fc1(x))
        x2 = F.relu(F.dropout(self.fc2(x1), p=0.1,
                              training=self.training))
        return self.out(x2), x1


class GenNet(nn.Module):
    """Generator (ref Gen_Net :99-134): noise -> 60 -> 128 -> 256 ->
    128 -> latent."""

    def __init__(self, input_x2_dim: int, output_dim: int):
        super().__init__()
        self.x2_input = nn.Linear(input_x2_dim, 60)
        self.fc1 = nn.Linear(60, 128)
        self.fc2 = nn.Linear(128, 256)
        self.fc3 = nn.Linear(256, 128)
        self.out = nn.Linear(128, output_dim)
        for m in (self.x2_input, self.fc1, self.fc2, self.fc3, self.out):
            m.weight.data.normal_(0, 0.1)

    def forward(self, x2):
        x = self.x2_input(x2)
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        x = F.relu(self.fc3(x))
        return self.out(x)


class GansProcess:
    """Alternating cls/gen training (ref gans_process :136-520):
    the classifier separates real latents (class 0) from generated ones
    (class 1); the generator trains by FEATURE MATCHING — MSE between
    the classifier's hidden features of fake vs real batches, decayed by
    0.9^round — not by fooling the classifier head directly."""

    def __init__(self, z_dim: int, x2_dim: int = 80, cls_num: int = 2,
                 cls_lr: float = 1e-3, gen_lr: float = 1e-3,
                 cls_epoches: int = 1, gen_epoches: int = 1,
                 batch_size: int = 16, device="cpu"):
        self.z_dim = z_dim
        self.x2_dim = x2_dim
        self.device = device
        self.cls_net = ClsNet(cls_num, z_dim, batch_size).to(device)
        self.gen_net = GenNet(x2_dim, z_dim).to(device)
        self.cls_optimizer = torch.optim.Adam(
            self.cls_net.parameters(), lr=cls_lr)
        self.gen_optimizer = torch.optim.Adam(
            self.gen_net.parameters(), lr=gen_lr)
        self.cls_epoches = cls_epoches
        self.gen_epoches = gen_epoches
        self.batch_size = batch_size
        self.loss_fn = nn.CrossEntropyLoss()
        self.loss_mse = nn.MSELoss()

    def labels2genx(self, n: int) -> torch.Tensor:
        return torch.rand(n, self.x2_dim, device=self.device)

    def ready_cls(self, sent_output):
        """real latents class 0, generated class 1, shuffled (ref
        :198-223)."""
        n = len(sent_output)
        sent_output = sent_output.to(self.device)
        sent_noise = self.gen_test(n)
        x = torch.cat((sent_output, sent_noise), dim=0)
        y = torch.cat((torch.zeros(n, dtype=torch.long),
                       torch.ones(n, dtype=torch.long))).to(self.device)
        perm = torch.randperm(len(x))
        return x[perm], y[perm]

    def cls_train(self, x, y):
        self.cls_net.train()
        self.gen_net.eval()
        for _ in range(self.cls_epoches):
            for i in range(0, len(x), self.batch_size):
                xb = x[i:i + self.batch_size].float()
                yb = y[i:i + self.batch_size]
                logits, _ = self.cls_net(xb)
                loss = self.loss_fn(logits, yb)
                self.cls_optimizer.zero_grad()
                loss.backward()
                self.cls_optimizer.step()
        return float(loss)

    def gen_train(self, real_latents, times: int):
        """Feature matching with 0.9^times decay (ref :427-470)."""
        self.cls_net.eval()
        self.gen_net.train()
        real_latents = real_latents.to(self.device)
        for _ in range(self.gen_epoches):
            for i in range(0, len(real_latents), self.batch_size):
                s = real_latents[i:i + self.batch_size].float()
                x2 = self.labels2genx(len(s))
                fake = self.gen_net(x2)
                _out, hds = self.cls_net(fake)
                with torch.no_grad():
                    _out2, hds2 = self.cls_net(s)
                loss = self.loss_mse(hds, hds2) * pow(0.9, times)
                self.gen_optimizer.zero_grad()
                loss.backward()
                self.gen_optimizer.step()
        return float(loss)

    @torch.no_grad()
    def gen_test(self, n: int) -> torch.Tensor:
        self.gen_net.eval()
        return self.gen_net(self.labels2genx(n))


def gavae_train_gan(gan: GansProcess, latents: torch.Tensor,
                    gan_epoch: int = 4, max_retries: int = 3):
    """train_gan with the reference's NaN-retry loop (GAVAEModel.py
    :44-53): rerun the alternating schedule until generated latents are
    NaN-free."""
    for _retry in range(max_retries):
        for gt in range(gan_epoch):
            x, y = gan.ready_cls(latents)
            gan.cls_train(x, y)
            gan.gen_train(latents, gt)
        if not gan.gen_test(len(latents)).isnan().any():
            return gan
    raise RuntimeError("GAN training produced NaN latents repeatedly")
