"""DreamBooth finetune with prior preservation.

Behavioral parity: reference examples/stable_diffusion_dreambooth
(readme.md:11,38 — instance + class ("prior") images, combined loss
L = mse(instance) + prior_weight * mse(class)).  Runs on our Taiyi-SD
stack (UNet/VAE/DDPM scheduler + BERT text encoder).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    bert_tiny_config,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.taiyi_sd import (
    AutoencoderKL,
    DDPMScheduler,
    UNet2DConditionModel,
)
from fengshen_amd.models.taiyi_sd.unet import unet_tiny_config
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class DreamBooth(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.text_encoder = MegatronBertModel(bert_tiny_config(),
                                              add_pooling_layer=False)
        self.vae = AutoencoderKL()
        self.unet = UNet2DConditionModel(unet_tiny_config())
        self.noise_scheduler = DDPMScheduler()
        for p in self.vae.parameters():
            p.requires_grad = False
        self.prior_weight = args.prior_loss_weight

    def _mse(self, pixels, input_ids):
        with torch.no_grad():
            latents = self.vae.encode(pixels)
        noise = torch.randn_like(latents)
        t = torch.randint(0, self.noise_scheduler.num_train_timesteps,
                          (latents.shape[0],), device=latents.device)
        noisy = self.noise_scheduler.add_noise(latents, noise, t)
        ctx = self.text_encoder(input_ids).last_hidden_state
        pred = self.unet(noisy, t, ctx)
        return torch.nn.functional.mse_loss(pred.float(), noise.float())

    def training_step(self, batch, batch_idx):
        loss = self._mse(batch["pixel_values"], batch["input_ids"])
        if "class_pixel_values" in batch:
            prior = self._mse(batch["class_pixel_values"],
                              batch["class_input_ids"])
            self.log("prior_loss", prior)
            loss = loss + self.prior_weight * prior
        self.log("train_loss", loss)
        return loss

    def configure_optimizers(self):
        return configure_optimizers(self)


class _DreamBoothCollator:
    def __init__(self, tokenizer, with_prior: bool):
        self.tokenizer = tokenizer
        self.with_prior = with_prior

    def _enc(self, texts):
        ids = [self.tokenizer.encode(t)[:24] for t in texts]
        L = max(len(x) for x in ids)
        pad = self.tokenizer.pad_token_id
        return torch.tensor([x + [pad] * (L - len(x)) for x in ids])

    def __call__(self, samples):
        batch = {
            "input_ids": self._enc([s["instance_text"] for s in samples]),
            "pixel_values": torch.stack([
                torch.as_tensor(s["instance_pixels"]) for s in samples]),
        }
        if self.with_prior and "class_pixels" in samples[0]:
            batch["class_input_ids"] = self._enc(
                [s["class_text"] for s in samples])
            batch["class_pixel_values"] = torch.stack([
                torch.as_tensor(s["class_pixels"]) for s in samples])
        return batch


def synthetic_dreambooth(n=64, image_size=32):
    import numpy as np
    rng = np.random.RandomState(0)
    return [{
        "instance_text": "一张sks狗的照片",
        "instance_pixels": rng.randn(3, image_size, image_size).astype("float32"),
        "class_text": "一张狗的照片",
        "class_pixels": rng.randn(3, image_size, image_size).astype("float32"),
    } for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--with_prior_preservation", action="store_true",
                        default=True)
    parser.add_argument("--prior_loss_weight", type=float, default=1.0)
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(
        tokenizer,
        _DreamBoothCollator(tokenizer, args.with_prior_preservation), args,
        datasets={"train": synthetic_dreambooth()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(DreamBooth(args), datamodule=dm)


if __name__ == "__main__":
    main()
