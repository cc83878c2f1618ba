"""UniMC zero-shot/few-shot label-as-option finetune
(reference examples/unimc)."""
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
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.unimc.modeling_unimc import (
    UniMCEncoder, UniMCModel, unimc_collate)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

CHOICES = ["好评", "差评"]


class UniMCCollator:
    """Reference-parity collation: option-isolation attention mask,
    per-option restarted position ids, yes/no MLM anchors + aux masking
    (ref UniMCDataset.encode + collate_fn)."""

    def __init__(self, tokenizer, max_len=64, used_mask=True):
        self.encoder = UniMCEncoder(
            tokenizer, yes_token=5, no_token=6, max_length=max_len,
            used_mask=used_mask)

    def __call__(self, samples):
        enc = [self.encoder.encode(
            {"texta": s["texta"], "choice": CHOICES, "label": s["label"]})
            for s in samples]
        batch = unimc_collate(enc)
        batch.pop("option_positions", None)
        batch.pop("mlmlabels_mask", None)
        return batch


class UniMCTask(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = UniMCModel(bert_tiny_config(), yes_token_id=5)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_sentiment(n=64):
    return [{"texta": "味道很棒下次还来" if i % 2 == 0 else "太难吃了不会再来",
             "label": i % 2} for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, UniMCCollator(tokenizer), args,
                             datasets={"train": synthetic_sentiment()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(UniMCTask(args), datamodule=dm)


if __name__ == "__main__":
    main()
