"""ZEN1 sequence-level finetune (tnews-style classification)
(reference examples/zen1_finetune/fengshen_sequence_level_ft_task.py).

ZEN1 = the dual-stream n-gram BERT without ZEN2's QA head; the same
ZenModel backbone serves both (ref models/zen1 vs zen2 share the
layer[i] / word_layers[i] fusion, modeling.py:416-450)."""
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
    ZenForSequenceClassification, ZenNgramDict, zen_tiny_config)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

TNEWS_LABELS = ["news_story", "news_culture", "news_sports", "news_tech"]


class ZenSeqCollator:
    def __init__(self, tokenizer, ngram_dict, max_len=64, max_ngram=16):
        self.tokenizer = tokenizer
        self.ngram_dict = ngram_dict
        self.max_len = max_len
        self.max_ngram = max_ngram

    def __call__(self, samples):
        ids_b, ng_b, pos_b, lab_b = [], [], [], []
        for s in samples:
            chars = list(s["sentence"])[:self.max_len]
            ids = [self.tokenizer.get_vocab().get(c, 4) for c in chars]
            matches = self.ngram_dict.match(chars)[:self.max_ngram]
            ng = [m[2] for m in matches]
            pos = torch.zeros(self.max_len, self.max_ngram)
            for j, (st, en, _idx) in enumerate(matches):
                pos[st:en, j] = 1.0
            ids_b.append(ids + [0] * (self.max_len - len(ids)))
            ng_b.append(ng + [0] * (self.max_ngram - len(ng)))
            pos_b.append(pos)
            lab_b.append(int(s["label"]))
        return {"input_ids": torch.tensor(ids_b),
                "ngram_ids": torch.tensor(ng_b),
                "ngram_position_matrix": torch.stack(pos_b),
                "labels": torch.tensor(lab_b)}


class Zen1SeqTask(FengshenModule):
    def __init__(self, args, ngram_list):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = zen_tiny_config(num_labels=len(TNEWS_LABELS))
        self.model = ZenForSequenceClassification(cfg)
        self.ngram_dict = ZenNgramDict(ngram_list)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        preds = out.logits.argmax(-1)
        acc = (preds == batch["labels"]).float().mean()
        self.log("train_acc", acc)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_tnews(n=64):
    sents = ["故事里的小村庄十分宁静", "博物馆今天展出新文物",
             "球队昨晚赢得了比赛", "新款芯片性能大幅提升"]
    return [{"sentence": sents[i % 4], "label": i % 4} for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    ngram_list = ["小村庄", "博物馆", "新文物", "比赛", "芯片", "性能"]
    task = Zen1SeqTask(args, ngram_list)
    dm = UniversalDataModule(
        tokenizer, ZenSeqCollator(tokenizer, task.ngram_dict), args,
        datasets={"train": synthetic_tnews()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(task, datamodule=dm)


if __name__ == "__main__":
    main()
