"""Sequence tagging (NER) pipeline — reference pipelines/sequence_tagging.py."""
from __future__ import annotations

import argparse
from typing import List, Union

import torch

from fengshen_amd.metric.utils_ner import get_entities
from fengshen_amd.pipelines.base import BasePipeline


class _TagCollator:
    def __init__(self, tokenizer, label2id, max_length=256,
                 text_key="text", label_key="labels"):
        self.tokenizer = tokenizer
        self.label2id = label2id
        self.max_length = max_length
        self.text_key = text_key
        self.label_key = label_key

    def __call__(self, samples):
        ids, labels = [], []
        for s in samples:
            text = s[self.text_key] if isinstance(s, dict) else s
            chars = list(text)[:self.max_length - 2]
            enc = [self.tokenizer.cls_token_id] + [
                self.tokenizer.get_vocab().get(c, 4) for c in chars] + \
                [self.tokenizer.sep_token_id]
            ids.append(enc)
            if isinstance(s, dict) and self.label_key in s:
                lab = [self.label2id.get(x, 0) for x in s[self.label_key]]
                labels.append([-100] + lab[:self.max_length - 2] + [-100])
        L = max(len(x) for x in ids)
        pad = self.tokenizer.pad_token_id or 0
        batch = {
            "input_ids": torch.tensor(
                [x + [pad] * (L - len(x)) for x in ids], dtype=torch.long),
            "attention_mask": torch.tensor(
                [[1] * len(x) + [0] * (L - len(x)) for x in ids],
                dtype=torch.long),
        }
        if labels:
            batch["labels"] = torch.tensor(
                [x + [-100] * (L - len(x)) for x in labels], dtype=torch.long)
        return batch


class SequenceTaggingPipeline(BasePipeline):
    """NER with linear/CRF heads + BIO decoding."""

    task_name = "sequence_tagging"

    def __init__(self, args=None, model=None, tokenizer=None, id2label=None,
                 config=None, head: str = "linear"):
        super().__init__(args, model, tokenizer)
        self.id2label = id2label or {}
        self.label2id = {v: k for k, v in self.id2label.items()}
        if self.model is None:
            from fengshen_amd.models.tagging_models.bert_for_tagging import (
                BertCrf, BertLinear)
            cls = {"linear": BertLinear, "crf": BertCrf}[head]
            if isinstance(model, str):
                self.model = cls.from_pretrained(model,
                                                 num_labels=len(id2label))
            else:
                self.model = cls(config, num_labels=len(id2label))
        self.model.eval()

    @classmethod
    def add_pipeline_specific_args(cls, parser: argparse.ArgumentParser):
        parser = super().add_pipeline_specific_args(parser)
        g = parser.add_argument_group("sequence tagging")
        g.add_argument("--markup", type=str, default="bio")
        g.add_argument("--max_length", type=int, default=256)
        return parser

    def collator(self):
        return _TagCollator(self.tokenizer, self.label2id,
                            max_length=getattr(self.args, "max_length", 256))

    @torch.no_grad()
    def __call__(self, texts: Union[str, List[str]]):
        single = isinstance(texts, str)
        if single:
            texts = [texts]
        batch = self.collator()([{"text": t} for t in texts])
        dev = next(self.model.parameters()).device
        batch = {k: v.to(dev) for k, v in batch.items()}
        out = self.model(**batch, decode=hasattr(self.model, "crf"))
        results = []
        if out.predictions is not None:
            paths = out.predictions
        else:
            paths = out.logits.argmax(-1).tolist()
        for text, path in zip(texts, paths):
            tags = [self.id2label.get(int(t), "O")
                    for t in path[1:1 + len(text)]]
            ents = get_entities(tags, markup=getattr(self.args, "markup", "bio")
                                if self.args else "bio")
            results.append([
                {"entity": text[s:e + 1], "type": t, "start": s, "end": e}
                for t, s, e in ents])
        return results[0] if single else results


Pipeline = SequenceTaggingPipeline
