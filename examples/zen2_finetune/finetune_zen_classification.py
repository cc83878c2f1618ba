"""ZEN2 classification finetune (reference examples/zen2_finetune, 30+
scripts) — demonstrates the n-gram pipeline: NgramDict match -> position
matrix -> dual-stream encoder."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.models.zen.modeling_zen import (
    ZenForSequenceClassification,
    ZenNgramDict,
    zen_tiny_config,
)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class ZenCollator:
    def __init__(self, tokenizer, ngram_dict: ZenNgramDict, max_len=64,
                 max_ngram=16):
        self.tokenizer = tokenizer
        self.ngram_dict = ngram_dict
        self.max_len = max_len
        self.max_ngram = max_ngram

    def __call__(self, samples):
        pad = self.tokenizer.pad_token_id
        ids_b, ng_b, pos_b, lab_b = [], [], [], []
        for s in samples:
            chars = list(s["sentence"])[:self.max_len]
            ids = [self.tokenizer.get_vocab().get(c, 4) for c in chars]
            matches = self.ngram_dict.match(chars)[:self.max_ngram]
            ngram_ids = [g for g, _, _ in matches]
            pos = torch.zeros(self.max_len, self.max_ngram)
            for k, (_, st, ln) in enumerate(matches):
                pos[st:st + ln, k] = 1
            ids_b.append(ids)
            ng_b.append(ngram_ids)
            pos_b.append(pos)
            lab_b.append(int(s["label"]))
        L = max(len(x) for x in ids_b)
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids_b]),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (L - len(x)) for x in ids_b]),
            "ngram_ids": torch.tensor(
                [x + [0] * (self.max_ngram - len(x)) for x in ng_b]),
            "ngram_position_matrix": torch.stack(pos_b)[:, :L, :],
            "labels": torch.tensor(lab_b),
        }


class ZenCls(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        cfg = zen_tiny_config()
        cfg.num_labels = 2
        self.model = ZenForSequenceClassification(cfg)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    ngram_dict = ZenNgramDict(["天气", "今天", "非常好", "很差"])
    data = [{"sentence": "今天天气非常好", "label": 1},
            {"sentence": "今天天气很差", "label": 0}] * 32
    dm = UniversalDataModule(tokenizer, ZenCollator(tokenizer, ngram_dict),
                             args, datasets={"train": data})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(ZenCls(args), datamodule=dm)


if __name__ == "__main__":
    main()
