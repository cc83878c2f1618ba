"""Erlangshen-DeBERTa-v2 MLM pretraining app.

Behavioral parity: reference examples/pretrain_erlangshen_deberta_v2/
pretrain_deberta.py — wwm MLM collator over jsonl corpus,
DebertaV2ForMaskedLM, UniversalDataModule, exact-resume checkpointing.

Run:
  torchrun --standalone --nproc-per-node N pretrain_deberta.py \
    --strategy zero2 --max_steps 10000 --train_file corpus.jsonl
With no --train_file a synthetic corpus is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import MlmSopCollator
from fengshen_amd.metric.metric import metrics_mlm_acc
from fengshen_amd.models.deberta_v2.modeling_deberta_v2 import (
    DebertaV2Config,
    DebertaV2ForMaskedLM,
    deberta_tiny_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class ErlangshenDeberta(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = deberta_tiny_config() if args.model_size == "tiny" \
            else DebertaV2Config()
        self.model = DebertaV2ForMaskedLM(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(batch["input_ids"], batch["attention_mask"],
                         batch.get("token_type_ids"), labels=batch["labels"])
        self.log("train_loss", out.loss)
        if batch_idx % 100 == 0:
            self.log("train_mlm_acc",
                     metrics_mlm_acc(out.logits, batch["labels"]))
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(batch["input_ids"], batch["attention_mask"],
                         batch.get("token_type_ids"), labels=batch["labels"])
        self.log("val_loss", out.loss, sync_dist=True)

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_corpus(n=256):
    import random
    rng = random.Random(0)
    sents = ["今天天气真好。", "我们一起去公园散步吧！", "他正在学习新的技术。",
             "这本书的内容非常有趣。", "人工智能正在改变世界。"]
    return [{"text": "".join(rng.sample(sents, 3))} for _ in range(n)]


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

    collator = MlmSopCollator(tokenizer, max_seq_length=128)
    datasets = None
    if not args.train_file and not args.datasets_name:
        datasets = {"train": synthetic_corpus()}
    dm = UniversalDataModule(tokenizer, collator, args, datasets=datasets)
    module = ErlangshenDeberta(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm,
                ckpt_path=getattr(args, "load_ckpt_path", None))


if __name__ == "__main__":
    main()
