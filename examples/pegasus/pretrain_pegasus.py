"""Randeng-Pegasus gap-sentence pretraining (reference examples/pegasus)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.pegasus_utils import build_gap_sentence_sample
from fengshen_amd.models.bart.modeling_bart import (
    BartForConditionalGeneration,
    bart_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class GapSentenceCollator:
    def __init__(self, tokenizer, max_len=128):
        self.tokenizer = tokenizer
        self.max_len = max_len

    def __call__(self, samples):
        srcs, tgts = [], []
        for s in samples:
            inp, tgt = build_gap_sentence_sample(
                s["text"] if isinstance(s, dict) else s)
            src_ids = []
            for chunk in inp.split("[MASK]"):
                src_ids += self.tokenizer.encode(chunk,
                                                 add_special_tokens=False)
                src_ids.append(self.tokenizer.mask_token_id)
            src_ids.pop()
            srcs.append(src_ids[:self.max_len])
            tgts.append(self.tokenizer.encode(
                tgt, add_special_tokens=False)[:self.max_len])
        pad = self.tokenizer.pad_token_id
        Ls = max(len(x) for x in srcs)
        Lt = max(len(x) for x in tgts)
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (Ls - len(x)) for x in srcs]),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (Ls - len(x)) for x in srcs]),
            "labels": torch.tensor(
                [x + [-100] * (Lt - len(x)) for x in tgts]),
        }


class Pegasus(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = BartForConditionalGeneration(bart_tiny_config())

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_docs(n=128):
    doc = "城市的清晨充满活力。街道上人来人往。早餐店飘出香味。公园里有人晨练。新的一天开始了。"
    return [{"text": doc} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, GapSentenceCollator(tokenizer), args,
                             datasets={"train": synthetic_docs()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(Pegasus(args), datamodule=dm)


if __name__ == "__main__":
    main()
