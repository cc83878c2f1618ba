"""Seq2seq summarization finetune (BART / T5 / Pegasus)
(reference examples/summary/seq2seq_summary.py: AbstractCollator + ROUGE
validation with chinese_char_tokenize; torchmetrics replaced by the native
fengshen_amd.metric.rouge implementation)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse
import json

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.metric.rouge import RougeScore
from fengshen_amd.models.bart.modeling_bart import (
    BartForConditionalGeneration, bart_tiny_config)
from fengshen_amd.models.model_utils import (
    add_module_args, configure_optimizers)
from fengshen_amd.trainer.callbacks import ThroughputMonitor
from fengshen_amd.utils import chinese_char_tokenize
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class AbstractCollator:
    """Summary-task collation (ref task_datasets.py AbstractCollator):
    prompt + text -> encoder; summary -> labels."""

    def __init__(self, tokenizer, max_enc_length=128, max_dec_length=64,
                 prompt=""):
        self.tokenizer = tokenizer
        self.max_enc_length = max_enc_length
        self.max_dec_length = max_dec_length
        self.prompt = prompt

    def __call__(self, samples):
        source, labels, attn = [], [], []
        for s in samples:
            enc = self.tokenizer.encode_plus(
                self.prompt + s["text"], max_length=self.max_enc_length,
                padding="max_length", truncation=True)
            dec = self.tokenizer.encode_plus(
                s["summary"], max_length=self.max_dec_length,
                padding="max_length", truncation=True)
            source.append(enc["input_ids"])
            attn.append(enc["attention_mask"])
            lab = [(t if m else -100) for t, m in
                   zip(dec["input_ids"], dec["attention_mask"])]
            labels.append(lab)
        return {
            "input_ids": torch.tensor(source),
            "attention_mask": torch.tensor(attn),
            "labels": torch.tensor(labels),
            "text": [s["text"] for s in samples],
            "summary": [s["summary"] for s in samples],
        }


class FinetuneSummary(FengshenModule):
    @staticmethod
    def add_model_specific_args(parent_args):
        g = parent_args.add_argument_group("FinetuneSummary")
        g.add_argument("--rouge_keys", default="rougeL,rouge1,rouge2")
        g.add_argument("--output_save_path", default="./predict.json")
        return parent_args

    def __init__(self, args, tokenizer=None):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = BartForConditionalGeneration(bart_tiny_config())
        self.tokenizer = tokenizer
        self.rouge = RougeScore(tuple(args.rouge_keys.split(",")))

    def training_step(self, batch, batch_idx):
        out = self.model(input_ids=batch["input_ids"],
                         attention_mask=batch["attention_mask"],
                         labels=batch["labels"])
        self.log("train_loss", out.loss, sync_dist=True)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(input_ids=batch["input_ids"],
                         attention_mask=batch["attention_mask"],
                         labels=batch["labels"])
        gen = self.model.generate(batch["input_ids"], max_new_tokens=32)
        preds = [self.tokenizer.decode(g) for g in gen]
        self.rouge.update(
            [chinese_char_tokenize(p) for p in preds],
            [chinese_char_tokenize(t) for t in batch["summary"]])
        with open(self.hparams.output_save_path, "a") as f:
            for t, p in zip(batch["text"], preds):
                f.write(json.dumps({"text": t, "pred": p},
                                   ensure_ascii=False) + "\n")
        self.log("val_loss", out.loss, sync_dist=True)
        return out.loss

    def on_validation_end(self):
        scores = self.rouge.compute()
        for k, v in scores.items():
            if k.endswith("fmeasure"):
                print(f"[rouge] {k} = {v:.4f}")
        self.rouge.reset()

    def configure_optimizers(self):
        return configure_optimizers(self)


def synthetic_lcsts(n=64):
    base = [("今天股市大幅上涨创历史新高投资者信心增强", "股市大涨"),
            ("本市今日降雨带来清凉气温明显下降", "降雨降温")]
    return [{"text": base[i % 2][0], "summary": base[i % 2][1]}
            for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    FinetuneSummary.add_model_specific_args(parser)
    args = parser.parse_args()
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    data = synthetic_lcsts()
    dm = UniversalDataModule(
        tokenizer, AbstractCollator(tokenizer, prompt="摘要:"), args,
        datasets={"train": data, "validation": data[:8]})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[ThroughputMonitor(), UniversalCheckpoint(args)])
    trainer.fit(FinetuneSummary(args, tokenizer), datamodule=dm)


if __name__ == "__main__":
    main()
