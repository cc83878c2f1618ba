"""ZEN1 token-level finetune (NER, ontonotes-style)
(reference examples/zen1_finetune/fengshen_token_level_ft_task.py)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.model_utils import (
    add_module_args, configure_optimizers)
from fengshen_amd.models.zen.modeling_zen import (
    ZenForTokenClassification, ZenNgramDict, zen_tiny_config)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

NER_LABELS = ["O", "B-PER", "I-PER", "B-LOC", "I-LOC"]


class ZenTokenCollator:
    def __init__(self, tokenizer, ngram_dict, max_len=32, max_ngram=8):
        self.tokenizer = tokenizer
        self.ngram_dict = ngram_dict
        self.max_len = max_len
        self.max_ngram = max_ngram

    def __call__(self, samples):
        ids_b, ng_b, pos_b, lab_b = [], [], [], []
        for s in samples:
            chars = list(s["chars"])[:self.max_len]
            labels = list(s["labels"])[:self.max_len]
            ids = [self.tokenizer.get_vocab().get(c, 4) for c in chars]
            matches = self.ngram_dict.match(chars)[:self.max_ngram]
            ng = [m[2] for m in matches]
            pos = torch.zeros(self.max_len, self.max_ngram)
            for j, (st, en, _idx) in enumerate(matches):
                pos[st:en, j] = 1.0
            padlen = self.max_len - len(ids)
            ids_b.append(ids + [0] * padlen)
            lab_b.append(labels + [-100] * padlen)
            ng_b.append(ng + [0] * (self.max_ngram - len(ng)))
            pos_b.append(pos)
        return {"input_ids": torch.tensor(ids_b),
                "ngram_ids": torch.tensor(ng_b),
                "ngram_position_matrix": torch.stack(pos_b),
                "labels": torch.tensor(lab_b)}


class Zen1TokenTask(FengshenModule):
    def __init__(self, args, ngram_list):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = zen_tiny_config(num_labels=len(NER_LABELS))
        self.model = ZenForTokenClassification(cfg)
        self.ngram_dict = ZenNgramDict(ngram_list)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_ner(n=64):
    # 李明 (PER) 在北京 (LOC)
    sample = {"chars": "李明在北京工作",
              "labels": [1, 2, 0, 3, 4, 0, 0]}
    return [dict(sample) for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    task = Zen1TokenTask(args, ["李明", "北京", "工作"])
    dm = UniversalDataModule(
        tokenizer, ZenTokenCollator(tokenizer, task.ngram_dict), args,
        datasets={"train": synthetic_ner()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(task, datamodule=dm)


if __name__ == "__main__":
    main()
