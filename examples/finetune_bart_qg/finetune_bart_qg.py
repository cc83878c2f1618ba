"""Randeng-BART question generation finetune.

Behavioral parity: reference examples/finetune_bart_qg/finetune_bart.py —
"context + answer -> question" seq2seq: encoder input
"answer:{a} context:{c}", decoder target = question, BART conditional
generation with our encoder-decoder stack.

Run:
  torchrun --standalone --nproc-per-node N finetune_bart_qg.py \
    --strategy zero2 --max_steps 2000 --train_file qg.jsonl
With no --train_file a synthetic dataset is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.bart.modeling_bart import (
    BartForConditionalGeneration,
    bart_tiny_config,
    randeng_bart_139m_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class QGCollator:
    def __init__(self, tokenizer, max_src=256, max_tgt=64):
        self.tk = tokenizer
        self.max_src, self.max_tgt = max_src, max_tgt

    def __call__(self, samples):
        src_ids, tgt_ids = [], []
        for s in samples:
            src = f"answer:{s['answer']} context:{s['context']}"
            src_ids.append(self.tk.encode(src)[:self.max_src])
            tgt_ids.append(self.tk.encode(s["question"])[:self.max_tgt])
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


class BartQG(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = bart_tiny_config() if args.model_size == "tiny" \
            else randeng_bart_139m_config()
        self.model = BartForConditionalGeneration(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_qg(n=256):
    import random
    rng = random.Random(0)
    ctxs = ["北京是中国的首都。", "长江是亚洲最长的河流。", "熊猫生活在四川。"]
    return [{"context": rng.choice(ctxs), "answer": "北京",
             "question": "中国的首都是哪里？"} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny",
                        choices=["tiny", "139m"])
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
        datasets = {"train": synthetic_qg()}
    dm = UniversalDataModule(tokenizer, QGCollator(tokenizer), args,
                             datasets=datasets)
    module = BartQG(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm)


if __name__ == "__main__":
    main()
