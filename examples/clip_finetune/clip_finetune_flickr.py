"""Taiyi-CLIP retrieval finetune (flickr-style).

Behavioral parity: reference examples/clip_finetune/clip_finetune_flickr.py
— image-text contrastive finetune + validation retrieval metrics computed
over ALL ranks' embeddings (val-embedding all_gather, ref :50-51).

Run:
  torchrun --standalone --nproc-per-node N clip_finetune_flickr.py \
    --strategy ddp --max_steps 500
With no --data a synthetic image-text set is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

from fengshen_amd import FengshenModule, Trainer
from fengshen_amd.models.clip.modeling_taiyi_clip import (
    TaiyiCLIPModel,
    taiyi_clip_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class ClipFinetune(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = TaiyiCLIPModel(taiyi_clip_tiny_config())
        self._val_img, self._val_txt = [], []

    def training_step(self, batch, batch_idx):
        out = self.model(input_ids=batch["input_ids"],
                         pixel_values=batch["pixel_values"],
                         return_loss=True)
        self.log("train_loss", out.loss)
        return out.loss

    def validation_step(self, batch, batch_idx):
        img = self.model.get_image_features(batch["pixel_values"])
        txt = self.model.get_text_features(batch["input_ids"])
        self._val_img.append(torch.nn.functional.normalize(img, dim=-1))
        self._val_txt.append(torch.nn.functional.normalize(txt, dim=-1))

    def on_validation_epoch_end(self):
        if not self._val_img:
            return
        img = torch.cat(self._val_img)
        txt = torch.cat(self._val_txt)
        # gather embeddings from every rank so retrieval runs over the
        # full val set (ref clip_finetune_flickr.py:50-51)
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            gi = [torch.zeros_like(img) for _ in range(dist.get_world_size())]
            gt = [torch.zeros_like(txt) for _ in range(dist.get_world_size())]
            dist.all_gather(gi, img)
            dist.all_gather(gt, txt)
            img, txt = torch.cat(gi), torch.cat(gt)
        sim = img.float() @ txt.float().t()
        r1_i2t = (sim.argmax(dim=1) == torch.arange(len(sim))).float().mean()
        r1_t2i = (sim.argmax(dim=0) == torch.arange(len(sim))).float().mean()
        self.log("val_r1_i2t", r1_i2t)
        self.log("val_r1_t2i", r1_t2i)
        self._val_img.clear()
        self._val_txt.clear()

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_pairs(n=64, image_size=32, seq=16, vocab=256):
    g = torch.Generator().manual_seed(0)
    return [{"pixel_values": torch.randn(3, image_size, image_size,
                                         generator=g),
             "input_ids": torch.randint(5, vocab, (seq,), generator=g)}
            for _ in range(n)]


def collate(batch):
    return {"pixel_values": torch.stack([s["pixel_values"] for s in batch]),
            "input_ids": torch.stack([s["input_ids"] for s in batch])}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--train_batchsize", default=16, type=int)
    add_module_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()

    data = synthetic_pairs()
    train = torch.utils.data.DataLoader(
        data, batch_size=args.train_batchsize, collate_fn=collate,
        shuffle=True)
    val = torch.utils.data.DataLoader(
        data[:32], batch_size=args.train_batchsize, collate_fn=collate)

    module = ClipFinetune(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, train_dataloaders=train, val_dataloaders=val)


if __name__ == "__main__":
    main()
