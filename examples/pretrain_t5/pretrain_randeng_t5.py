"""Randeng-T5 span-corruption pretraining (reference examples/pretrain_t5)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import T5SpanCollator
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.t5.modeling_t5 import (
    T5ForConditionalGeneration,
    randeng_t5_77m_config,
    randeng_t5_784m_config,
    t5_tiny_config,
)
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

_CONFIGS = {"tiny": t5_tiny_config, "77m": randeng_t5_77m_config,
            "784m": randeng_t5_784m_config}


class RandengT5(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = T5ForConditionalGeneration(_CONFIGS[args.model_size]())

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_corpus(n=256):
    import random
    rng = random.Random(0)
    base = "深度学习模型需要大量的数据进行训练。框架的设计决定了训练的效率。"
    return [{"text": base * rng.randint(2, 5)} for _ in range(n)]


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
        from fengshen_amd.tokenizer import SimpleCharTokenizer
        tokenizer = SimpleCharTokenizer()
    collator = T5SpanCollator(tokenizer, max_seq_length=128)
    datasets = {"train": synthetic_corpus()} \
        if not args.train_file and not args.datasets_name else None
    dm = UniversalDataModule(tokenizer, collator, args, datasets=datasets)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(RandengT5(args), datamodule=dm)


if __name__ == "__main__":
    main()
