"""Della deep-VAE pretraining app.

Behavioral parity: reference examples/deepVAE/pretrain_deep_vae.py —
layer-wise latent VAE (BERT encoder + GPT2 decoder) with KL-annealed ELBO,
via our DeepVAEModel and the native trainer.

Run:
  torchrun --standalone --nproc-per-node N pretrain_deep_vae.py \
    --strategy zero2 --max_steps 2000 --train_file corpus.jsonl
With no --train_file a synthetic corpus is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import CausalCollator
from fengshen_amd.models.deep_vae.modeling_deep_vae import (
    DeepVAEModel,
    deep_vae_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class DeepVAEPretrain(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = DeepVAEModel(deep_vae_tiny_config())

    def training_step(self, batch, batch_idx):
        out = self.model(batch["input_ids"], batch["attention_mask"],
                         labels=batch["labels"])
        self.log("train_loss", out.loss)
        if out.kl_loss is not None:
            self.log("kl_loss", out.kl_loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_corpus(n=256):
    import random
    rng = random.Random(0)
    sents = ["今天天气真好。", "我们一起去公园散步吧！", "他正在学习新的技术。"]
    return [{"text": "".join(rng.sample(sents, 2))} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--tokenizer", default=None)
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()

    if args.tokenizer:
        from transformers import AutoTokenizer
        tokenizer = AutoTokenizer.from_pretrained(args.tokenizer)
    else:
        from fengshen_amd.tokenizer import SimpleCharTokenizer as FakeTokenizer
        tokenizer = FakeTokenizer()

    datasets = None
    if not args.train_file and not args.datasets_name:
        datasets = {"train": synthetic_corpus()}
    dm = UniversalDataModule(
        tokenizer, CausalCollator(tokenizer, max_seq_length=64), args,
        datasets=datasets)
    module = DeepVAEPretrain(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm)


if __name__ == "__main__":
    main()
