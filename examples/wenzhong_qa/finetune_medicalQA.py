"""Wenzhong-GPT2 QA finetuning (reference examples/wenzhong_qa: GPT2 causal
finetune on medical QA)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import CausalCollator
from fengshen_amd.models.gpt2.configuration_gpt2 import (
    gpt2_tiny_config,
    wenzhong_gpt2_3b5_config,
)
from fengshen_amd.models.gpt2.modeling_gpt2 import GPT2LMHeadModel
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

_CONFIGS = {"tiny": gpt2_tiny_config, "3.5b": wenzhong_gpt2_3b5_config}


class WenzhongQA(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        if args.model_path:
            self.model = GPT2LMHeadModel.from_pretrained(args.model_path)
        else:
            self.model = GPT2LMHeadModel(_CONFIGS[args.model_size]())
        if args.activation_checkpointing:
            self.model.gradient_checkpointing_enable()

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("val_loss", out.loss, sync_dist=True)

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_qa(n=256):
    return [{"text": f"Question:症状{i}怎么办?Answer:建议及时就医咨询专业医生。"}
            for i in range(n)]


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
    collator = CausalCollator(tokenizer, max_seq_length=256)
    datasets = {"train": synthetic_qa()} \
        if not args.train_file and not args.datasets_name else None
    dm = UniversalDataModule(tokenizer, collator, args, datasets=datasets)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(WenzhongQA(args), datamodule=dm)


if __name__ == "__main__":
    main()
