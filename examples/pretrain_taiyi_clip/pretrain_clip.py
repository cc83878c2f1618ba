"""Taiyi-CLIP contrastive pretraining (reference examples/pretrain_taiyi_clip:
BERT text tower + ViT; open_clip-style cross-rank contrastive loss)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.clip.modeling_taiyi_clip import (
    TaiyiCLIPModel,
    taiyi_clip_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class TaiyiCLIP(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = TaiyiCLIPModel(taiyi_clip_tiny_config())
        if args.freeze_vision:
            for p in self.model.vision_model.parameters():
                p.requires_grad = False

    def training_step(self, batch, batch_idx):
        out = self.model(**batch, return_loss=True)
        self.log("train_loss", out.loss, sync_dist=True)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


class _ClipCollator:
    def __init__(self, tokenizer, image_size=32):
        self.tokenizer = tokenizer
        self.image_size = image_size

    def __call__(self, samples):
        ids = [self.tokenizer.encode(s["text"])[:32] for s in samples]
        L = max(len(x) for x in ids)
        pad = self.tokenizer.pad_token_id
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids]),
            "pixel_values": torch.stack(
                [torch.as_tensor(s["pixels"], dtype=torch.float32)
                 for s in samples]),
        }


def synthetic_pairs(n=128, image_size=32):
    import numpy as np
    rng = np.random.RandomState(0)
    return [{"text": f"一张编号{i}的图片",
             "pixels": rng.randn(3, image_size, image_size).astype("float32")}
            for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--freeze_vision", action="store_true", default=False)
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, _ClipCollator(tokenizer), args,
                             datasets={"train": synthetic_pairs()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(TaiyiCLIP(args), datamodule=dm)


if __name__ == "__main__":
    main()
