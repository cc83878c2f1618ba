"""Summarization finetune + generate (reference examples/mt5_summary and
examples/summary) on Randeng-T5."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.t5.modeling_t5 import (
    T5ForConditionalGeneration,
    t5_tiny_config,
)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class SummaryCollator:
    def __init__(self, tokenizer, max_src=96, max_tgt=32):
        self.tokenizer = tokenizer
        self.max_src = max_src
        self.max_tgt = max_tgt

    def __call__(self, samples):
        pad = self.tokenizer.pad_token_id
        srcs = [self.tokenizer.encode(s["text"],
                                      add_special_tokens=False)[:self.max_src]
                for s in samples]
        tgts = [self.tokenizer.encode(s["summary"],
                                      add_special_tokens=False)[:self.max_tgt]
                for s in samples]
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


class SummaryTask(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = T5ForConditionalGeneration(t5_tiny_config())

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def predict_step(self, batch, batch_idx):
        return self.model.generate(batch["input_ids"],
                                   attention_mask=batch["attention_mask"],
                                   max_new_tokens=self.hparams.max_tgt_len)

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_news(n=64):
    body = "昨天本市举行了大型科技展览。很多公司展出了最新产品。现场观众反响热烈。"
    return [{"text": body, "summary": "本市举行科技展览。"} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--max_tgt_len", type=int, default=32)
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, SummaryCollator(tokenizer), args,
                             datasets={"train": synthetic_news()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(SummaryTask(args), datamodule=dm)


if __name__ == "__main__":
    main()
