"""UBERT multi-task span-extraction finetune (reference examples/ubert)."""
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
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.ubert.modeling_ubert import UbertModel
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

ENTITY_TYPES = ["人名", "地名"]


class UbertCollator:
    def __init__(self, tokenizer, max_len=48):
        self.tokenizer = tokenizer
        self.max_len = max_len

    def __call__(self, samples):
        vocab = self.tokenizer.get_vocab()
        b = len(samples)
        nl = len(ENTITY_TYPES)
        ids = torch.zeros(b, nl, self.max_len, dtype=torch.long)
        span = torch.zeros(b, nl, self.max_len, self.max_len)
        mask = torch.zeros(b, nl, self.max_len, self.max_len)
        for bi, s in enumerate(samples):
            for li, lab in enumerate(ENTITY_TYPES):
                prompt = f"[CLS]{lab}[SEP]{s['text']}[SEP]"
                toks = [vocab.get(c, 4) for c in prompt][:self.max_len]
                ids[bi, li, :len(toks)] = torch.tensor(toks)
                off = len(lab) + 10  # [CLS]+label+[SEP] prefix chars
                mask[bi, li, :len(toks), :len(toks)] = 1
                for st, en, typ in s["entities"]:
                    if typ == li and off + en < self.max_len:
                        span[bi, li, off + st, off + en] = 1
        return {"input_ids": ids, "span_labels": span, "span_mask": mask}


class UbertTask(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = UbertModel(bert_tiny_config())

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_ie(n=64):
    # 李明(PER 0-1) 北京(LOC 4-5) in "李明住在北京"
    return [{"text": "李明住在北京",
             "entities": [(0, 1, 0), (4, 5, 1)]} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, UbertCollator(tokenizer), args,
                             datasets={"train": synthetic_ie()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(UbertTask(args), datamodule=dm)


if __name__ == "__main__":
    main()
