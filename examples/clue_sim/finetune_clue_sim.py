"""CLUE semantic similarity (QBQTC-style) finetune.

Behavioral parity: reference examples/clue_sim/main.py +
finetune_clue_sim.py — sentence-pair classification ([CLS] A [SEP] B)
with an Erlangshen BERT backbone.

Run:
  torchrun --standalone --nproc-per-node N finetune_clue_sim.py \
    --strategy ddp --max_steps 2000 --train_file qbqtc.jsonl
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
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    bert_tiny_config,
    erlangshen_base_config,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForSequenceClassification,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class PairCollator:
    def __init__(self, tokenizer, max_len=128):
        self.tk = tokenizer
        self.max_len = max_len

    def __call__(self, samples):
        cls_id = self.tk.cls_token_id
        sep_id = self.tk.sep_token_id
        pad = self.tk.pad_token_id or 0
        ids, types, labels = [], [], []
        for s in samples:
            a = self.tk.encode(s["query"], add_special_tokens=False)
            b = self.tk.encode(s["title"], add_special_tokens=False)
            seq = [cls_id] + a + [sep_id] + b + [sep_id]
            tt = [0] * (len(a) + 2) + [1] * (len(b) + 1)
            ids.append(seq[:self.max_len])
            types.append(tt[:self.max_len])
            labels.append(int(s["label"]))
        L = max(len(x) for x in ids)
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids]),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (L - len(x)) for x in ids]),
            "token_type_ids": torch.tensor(
                [x + [0] * (L - len(x)) for x in types]),
            "labels": torch.tensor(labels),
        }


class ClueSim(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        mk = bert_tiny_config if args.model_size == "tiny" \
            else erlangshen_base_config
        cfg = mk(num_labels=args.num_labels)
        self.model = MegatronBertForSequenceClassification(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        if batch_idx % 50 == 0:
            acc = (out.logits.argmax(-1) == batch["labels"]).float().mean()
            self.log("train_acc", acc)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("val_loss", out.loss, sync_dist=True)

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_pairs(n=256):
    import random
    rng = random.Random(0)
    qs = ["如何学习编程", "天气怎么样", "附近的餐厅"]
    ts = ["编程入门教程", "今日天气预报", "美食推荐"]
    return [{"query": rng.choice(qs), "title": rng.choice(ts),
             "label": rng.randint(0, 2)} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny",
                        choices=["tiny", "base"])
    parser.add_argument("--num_labels", default=3, type=int)
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
        datasets = {"train": synthetic_pairs()}
    dm = UniversalDataModule(tokenizer, PairCollator(tokenizer), args,
                             datasets=datasets)
    module = ClueSim(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm)


if __name__ == "__main__":
    main()
