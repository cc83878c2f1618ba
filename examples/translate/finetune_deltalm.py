"""DeltaLM translation finetune (zh<->en).

Behavioral parity: reference examples/translate/finetune_deltalm.py —
parallel-corpus seq2seq with DeltaLM (interleaved decoder), label-smoothed
CE handled inside the model.

Run:
  torchrun --standalone --nproc-per-node N finetune_deltalm.py \
    --strategy zero2 --max_steps 2000 --train_file parallel.jsonl
With no --train_file a synthetic corpus is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.deltalm.modeling_deltalm import (
    DeltaLMConfig,
    DeltaLMForConditionalGeneration,
    deltalm_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class TranslateCollator:
    def __init__(self, tokenizer, src_key="src", tgt_key="tgt",
                 max_len=256):
        self.tk = tokenizer
        self.src_key, self.tgt_key = src_key, tgt_key
        self.max_len = max_len

    def __call__(self, samples):
        src_ids = [self.tk.encode(s[self.src_key])[:self.max_len]
                   for s in samples]
        tgt_ids = [self.tk.encode(s[self.tgt_key])[:self.max_len]
                   for s in samples]
        pad = self.tk.pad_token_id or 0
        Ls = max(len(x) for x in src_ids)
        Lt = max(len(x) for x in tgt_ids)
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (Ls - len(x)) for x in src_ids]),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (Ls - len(x)) for x in src_ids]),
            "labels": torch.tensor(
                [x + [-100] * (Lt - len(x)) for x in tgt_ids]),
        }


class DeltaLMTranslate(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = deltalm_tiny_config() if args.model_size == "tiny" \
            else DeltaLMConfig()
        self.model = DeltaLMForConditionalGeneration(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_parallel(n=256):
    import random
    rng = random.Random(0)
    pairs = [("你好世界", "hello world"), ("今天天气很好", "the weather is nice"),
             ("我喜欢读书", "i like reading")]
    return [dict(zip(("src", "tgt"), rng.choice(pairs))) for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny",
                        choices=["tiny", "base"])
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
        datasets = {"train": synthetic_parallel()}
    dm = UniversalDataModule(tokenizer, TranslateCollator(tokenizer), args,
                             datasets=datasets)
    module = DeltaLMTranslate(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm)


if __name__ == "__main__":
    main()
