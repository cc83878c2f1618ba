"""Randeng-BART denoising pretrain (reference examples/pretrain_randeng_bart:
sentence permutation + whole-word text infilling collator)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse
import random

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.data.data_utils import ChineseSentenceSplitter
from fengshen_amd.models.bart.modeling_bart import (
    BartForConditionalGeneration,
    bart_tiny_config,
    randeng_bart_139m_config,
)
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint

_CONFIGS = {"tiny": bart_tiny_config, "139m": randeng_bart_139m_config}


class DenoiseCollator:
    """sentence permutation + span infilling with a single mask token
    (ref pretrain_bart.py:27-120)."""

    def __init__(self, tokenizer, max_len=128, mask_ratio=0.3, seed=1234):
        self.tokenizer = tokenizer
        self.max_len = max_len
        self.mask_ratio = mask_ratio
        self.splitter = ChineseSentenceSplitter()
        self.rng = random.Random(seed)

    def __call__(self, samples):
        srcs, tgts = [], []
        for s in samples:
            text = s["text"] if isinstance(s, dict) else s
            sents = self.splitter.tokenize(text) or [text]
            self.rng.shuffle(sents)  # sentence permutation
            corrupted = []
            for sent in sents:
                chars = list(sent)
                n_mask = max(1, int(len(chars) * self.mask_ratio))
                start = self.rng.randint(0, max(0, len(chars) - n_mask))
                chars[start:start + n_mask] = ["[MASK]"]
                corrupted.append("".join(chars))
            src_ids = []
            for c in corrupted:
                for ch in c.split("[MASK]"):
                    src_ids += self.tokenizer.encode(ch,
                                                     add_special_tokens=False)
                    src_ids.append(self.tokenizer.mask_token_id)
                src_ids.pop()
            tgt_ids = self.tokenizer.encode(text, add_special_tokens=False)
            srcs.append(src_ids[:self.max_len])
            tgts.append(tgt_ids[:self.max_len])
        pad = self.tokenizer.pad_token_id
        Ls = max(len(x) for x in srcs)
        Lt = max(len(x) for x in tgts)
        return {
            "input_ids": torch.tensor(
                [x + [pad] * (Ls - len(x)) for x in srcs]),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (Ls - len(x)) for x in srcs]),
            "labels": torch.tensor(
                [x + [-100] * (Lt - len(x)) for x in tgts]),
        }


class RandengBart(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = BartForConditionalGeneration(_CONFIGS[args.model_size]())

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_corpus(n=128):
    base = ["今天的天气非常好。", "我们决定一起去爬山。", "山顶的风景让人难忘。"]
    return [{"text": "".join(base)} for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_size", default="tiny", choices=list(_CONFIGS))
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, DenoiseCollator(tokenizer), args,
                             datasets={"train": synthetic_corpus()})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(RandengBart(args), datamodule=dm)


if __name__ == "__main__":
    main()
