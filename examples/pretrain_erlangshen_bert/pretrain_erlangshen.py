"""Erlangshen-MegatronBERT MLM+SOP pretraining app.

Behavioral parity: reference
examples/pretrain_erlangshen_bert/pretrain_erlangshen.py:36-237 —
wwm/n-gram MLM + sentence-order collator, MegatronBertForPreTraining,
UniversalDataModule, exact-resume checkpointing.

Run (1 node, N GPUs):
  torchrun --standalone --nproc-per-node N pretrain_erlangshen.py \
    --model_size base --strategy zero2 --max_steps 10000 \
    --train_file corpus.jsonl --precision bf16
With no --train_file a synthetic corpus is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))  # repo root (for direct runs)


import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import MlmSopCollator
from fengshen_amd.metric.metric import metrics_mlm_acc
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    bert_tiny_config,
    erlangshen_1b3_config,
    erlangshen_base_config,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForPreTraining,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

_CONFIGS = {"tiny": bert_tiny_config, "base": erlangshen_base_config,
            "1.3b": erlangshen_1b3_config}


class ErLangShenBert(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = MegatronBertForPreTraining(_CONFIGS[args.model_size]())

    def setup(self, stage=None):
        if self.global_rank == 0:
            total = sum(p.numel() for p in self.model.parameters())
            print(f"Total params: {total / 1e6:.1f}M")

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        if batch_idx % 100 == 0:
            acc = metrics_mlm_acc(out.prediction_logits, batch["labels"])
            self.log("train_mlm_acc", acc)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("val_loss", out.loss, sync_dist=True)

    def configure_optimizers(self):
        return configure_optimizers(self)

    def on_load_checkpoint(self, checkpoint) -> None:
        # exact data resume (ref pretrain_erlangshen.py:192-197)
        if "global_samples" in checkpoint and self.trainer is not None:
            self.trainer.global_samples = checkpoint["global_samples"]


def synthetic_corpus(n=512):
    import random
    rng = random.Random(0)
    sents = ["今天天气真好。", "我们一起去公园散步吧！", "他正在学习新的技术。",
             "这本书的内容非常有趣。", "人工智能正在改变世界。"]
    return [{"text": "".join(rng.sample(sents, 3))} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny", choices=list(_CONFIGS))
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

    collator = MlmSopCollator(tokenizer, max_seq_length=128)
    datasets = None
    if not args.train_file and not args.datasets_name:
        datasets = {"train": synthetic_corpus()}
    dm = UniversalDataModule(tokenizer, collator, args, datasets=datasets)
    module = ErLangShenBert(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm,
                ckpt_path=getattr(args, "load_ckpt_path", None))


if __name__ == "__main__":
    main()
