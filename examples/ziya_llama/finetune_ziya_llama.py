"""Ziya-LLaMA SFT app (TP × ZeRO capable).

Behavioral parity: reference examples/ziya_llama/finetune_ziya_llama.py
:35-230 — "<human>:...\n<bot>:" prompt-masked SFT collator, per-TP-rank
checkpoint loading (part_{rank}), DeepSpeedStrategy(tensor_model_parallel_
size) replaced by --strategy zero2/zero3 + --tensor_model_parallel_size.

Run: torchrun --standalone --nproc-per-node 8 finetune_ziya_llama.py \
  --model_size 13b --strategy zero3 --tensor_model_parallel_size 2 \
  --train_file sft.jsonl --max_seq_length 2048
With no --train_file a synthetic SFT set is generated (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))  # repo root (for direct runs)


import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.collators import SftCollator
from fengshen_amd.models.llama.configuration_llama import (
    llama_tiny_config,
    ziya_llama_13b_config,
)
from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import LearningRateMonitor, ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

_CONFIGS = {"tiny": llama_tiny_config, "13b": ziya_llama_13b_config}


class ZiyaLlama(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.args = args
        self.model = None

    def setup(self, stage=None):
        if self.model is not None:
            return
        args = self.args
        if args.model_path:
            # per-TP-rank shards: {path}/part_{tp_rank}
            # (ref finetune_ziya_llama.py:103-105)
            from fengshen_amd.parallel import groups
            tp_rank = groups.get_tensor_model_parallel_rank()
            path = args.model_path
            part = os.path.join(path, f"part_{tp_rank}")
            if os.path.isdir(part):
                path = part
            self.model = LlamaForCausalLM.from_pretrained(path)
        else:
            self.model = LlamaForCausalLM(_CONFIGS[args.model_size]())
        if args.activation_checkpointing:
            self.model.gradient_checkpointing_enable()

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss, sync_dist=True)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("val_loss", out.loss, sync_dist=True)

    def predict_step(self, batch, batch_idx):
        return self.model.generate(batch["input_ids"], max_new_tokens=64)

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_sft(n=256):
    qa = [("介绍一下你自己", "我是一个由封神榜框架训练的语言模型。"),
          ("1加1等于几", "1加1等于2。"),
          ("写一句诗", "床前明月光，疑是地上霜。")]
    return [{"query": qa[i % 3][0], "answer": qa[i % 3][1]} for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny", choices=list(_CONFIGS))
    parser.add_argument("--tokenizer", default=None)
    parser.add_argument("--max_seq_length", type=int, default=256)
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

    collator = SftCollator(tokenizer, max_seq_length=args.max_seq_length)
    datasets = None
    if not args.train_file and not args.datasets_name:
        datasets = {"train": synthetic_sft()}
    dm = UniversalDataModule(tokenizer, collator, args, datasets=datasets)
    module = ZiyaLlama(args)
    trainer = Trainer.from_argparse_args(
        args, callbacks=[LearningRateMonitor(), ThroughputMonitor(),
                         UniversalCheckpoint(args)])
    trainer.fit(module, datamodule=dm,
                ckpt_path=getattr(args, "load_ckpt_path", None))


if __name__ == "__main__":
    main()
