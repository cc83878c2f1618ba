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
from fengshen_amd.models.unimc.modeling_unimc import UniMCModel
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

CHOICES = ["好评", "差评"]


class UniMCCollator:
    def __init__(self, tokenizer, max_len=64):
        self.tokenizer = tokenizer
        self.max_len = max_len

    def __call__(self, samples):
        vocab = self.tokenizer.get_vocab()
        ids_b, pos_b, lab_b = [], [], []
        for s in samples:
            ids = [self.tokenizer.cls_token_id]
            pos = []
            for choice in CHOICES:
                pos.append(len(ids))
                ids += [vocab.get(c, 4) for c in choice]
                ids.append(self.tokenizer.sep_token_id)
            ids += [vocab.get(c, 4) for c in s["texta"]][:self.max_len]
            ids.append(self.tokenizer.sep_token_id)
            ids_b.append(ids)
            pos_b.append(pos)
            lab_b.append(int(s["label"]))
        L = max(len(x) for x in ids_b)
        pad = self.tokenizer.pad_token_id
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids_b]),
            "option_positions": torch.tensor(pos_b),
            "labels": torch.tensor(lab_b),
        }


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
