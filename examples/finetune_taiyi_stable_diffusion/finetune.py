"""Taiyi Stable-Diffusion finetune app.

Behavioral parity: reference examples/finetune_taiyi_stable_diffusion/
finetune.py — pipeline split into tokenizer/text_encoder/vae/unet/scheduler
(:81-87), freeze flags (:91-100), training_step = VAE encode -> noise/
timestep -> text encode -> UNet -> mse (:112-152), report_memory probe at
step 100 (:147-150).  Runs on our own UNet/VAE/scheduler (no diffusers).
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
from fengshen_amd.models.taiyi_sd.unet import (
    taiyi_sd_1b_config,
    unet_tiny_config,
)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint
from fengshen_amd.utils.utils import report_memory


class TaiyiSD(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        if args.model_size == "tiny":
            text_cfg = bert_tiny_config()
            unet_cfg = unet_tiny_config()
        else:
            # Taiyi-SD-1B: BERT text tower (768 hidden) + the SD-1 UNet
            # (860M, diffusers weight layout)
            text_cfg = bert_tiny_config(hidden_size=768,
                                        num_hidden_layers=12,
                                        num_attention_heads=12,
                                        intermediate_size=3072)
            unet_cfg = taiyi_sd_1b_config()
        self.text_encoder = MegatronBertModel(text_cfg,
                                              add_pooling_layer=False)
        self.vae = AutoencoderKL()
        self.unet = UNet2DConditionModel(unet_cfg)
        self.noise_scheduler = DDPMScheduler()
        # freeze flags (ref :91-100): default trains text encoder only
        # (the Taiyi-SD-Chinese recipe) unless --train_unet
        for p in self.vae.parameters():
            p.requires_grad = False
        if not args.train_unet:
            for p in self.unet.parameters():
                p.requires_grad = False
        if not args.train_text_encoder:
            for p in self.text_encoder.parameters():
                p.requires_grad = False

    def training_step(self, batch, batch_idx):
        with torch.no_grad():
            latents = self.vae.encode(batch["pixel_values"])
        noise = torch.randn_like(latents)
        t = torch.randint(0, self.noise_scheduler.num_train_timesteps,
                          (latents.shape[0],), device=latents.device)
        noisy = self.noise_scheduler.add_noise(latents, noise, t)
        ctx = self.text_encoder(batch["input_ids"]).last_hidden_state
        pred = self.unet(noisy, t, ctx)
        loss = torch.nn.functional.mse_loss(pred.float(), noise.float())
        self.log("train_loss", loss)
        if batch_idx == 100 and self.global_rank == 0:
            report_memory("taiyi_sd step100")
        return loss

    def configure_optimizers(self):
        return configure_optimizers(self)

    def on_save_checkpoint(self, checkpoint):
        """Export HF pipeline layout (ref finetune.py:154-158): one subdir
        per component, loadable with from_pretrained."""
        if self.global_rank == 0:
            root = os.path.join(
                self.hparams.default_root_dir,
                f"hf_out_{self.trainer.current_epoch}"
                f"_{self.trainer.global_step}")
            self.unet.save_pretrained(os.path.join(root, "unet"))
            self.vae.save_pretrained(os.path.join(root, "vae"))
            self.text_encoder.save_pretrained(
                os.path.join(root, "text_encoder"))


class _SDCollator:
    def __init__(self, tokenizer, image_size=32, max_len=32):
        self.tokenizer = tokenizer
        self.image_size = image_size
        self.max_len = max_len

    def __call__(self, samples):
        ids = [self.tokenizer.encode(s["text"])[:self.max_len]
               for s in samples]
        L = max(len(x) for x in ids)
        pad = self.tokenizer.pad_token_id
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids]),
            "pixel_values": torch.stack([
                torch.as_tensor(s["pixels"], dtype=torch.float32)
                for s in samples]),
        }


def synthetic_pairs(n=128, image_size=32):
    import numpy as np
    rng = np.random.RandomState(0)
    return [{"text": f"一幅编号{i}的画",
             "pixels": rng.randn(3, image_size, image_size).astype("float32")}
            for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny",
                        choices=["tiny", "1b"])
    parser.add_argument("--train_unet", action="store_true", default=False)
    parser.add_argument("--train_text_encoder", action="store_true",
                        default=True)
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, _SDCollator(tokenizer), args,
                             datasets={"train": synthetic_pairs()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(TaiyiSD(args), datamodule=dm)


if __name__ == "__main__":
    main()
