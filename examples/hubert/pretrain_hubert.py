"""HuBERT audio-SSL pretraining app.

Behavioral parity: reference examples/hubert/pretrain_hubert.py
(HubertLightning :107-230 — masked-frame NCE over k-means pseudo-labels;
dataset parity in fengshen_amd/data/hubert_dataset.py without the fairseq
Dictionary dependency: labels are int arrays per waveform).

Run:
  torchrun --standalone --nproc-per-node N pretrain_hubert.py \
    --strategy zero2 --max_steps 1000 --precision bf16
With no --manifest a synthetic waveform corpus is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import numpy as np
import torch

from fengshen_amd import FengshenModule, Trainer
from fengshen_amd.data.hubert_dataset import HubertDataset
from fengshen_amd.models.hubert import (
    HubertConfig,
    HubertForPreTraining,
    hubert_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class HubertPretrain(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = hubert_tiny_config() if args.model_size == "tiny" \
            else HubertConfig()
        self.model = HubertForPreTraining(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(batch["source"], padding_mask=batch["padding_mask"],
                         labels=batch["labels"])
        self.log("train_loss", out.loss)
        if batch_idx % 50 == 0 and out.logits is not None \
                and out.logits.numel():
            # masked-frame cluster accuracy
            lab = batch["labels"]
            t = min(out.mask_time_indices.shape[1], lab.shape[1])
            tgt = lab[:, :t][out.mask_time_indices[:, :t] & (lab[:, :t] != -100)]
            if tgt.numel() == out.logits.shape[0]:
                acc = (out.logits.argmax(-1) == tgt).float().mean()
                self.log("train_acc_m", acc)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_audio(n=64, sr=16000, label_rate=50, clusters=16):
    rng = np.random.RandomState(0)
    waves, labels = [], []
    for _ in range(n):
        dur = rng.randint(sr // 4, sr // 2)
        waves.append(rng.randn(dur).astype(np.float32) * 0.1)
        labels.append(rng.randint(0, clusters,
                                  size=(int(dur / sr * label_rate),)))
    return waves, labels


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny",
                        choices=["tiny", "base"])
    parser.add_argument("--manifest", default=None,
                        help=".npz with arrays wave_i / label_i")
    parser.add_argument("--train_batchsize", default=4, type=int)
    parser.add_argument("--max_sample_size", default=8000, type=int)
    add_module_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()

    if args.manifest:
        data = np.load(args.manifest, allow_pickle=True)
        waves = [data[k] for k in data.files if k.startswith("wave")]
        labels = [data[k] for k in data.files if k.startswith("label")]
    else:
        waves, labels = synthetic_audio()
    ds = HubertDataset(waves, labels, max_sample_size=args.max_sample_size)
    loader = torch.utils.data.DataLoader(
        ds, batch_size=args.train_batchsize, collate_fn=ds.collater,
        shuffle=True)

    module = HubertPretrain(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, train_dataloaders=loader)


if __name__ == "__main__":
    main()
