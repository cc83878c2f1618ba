"""Text classification pipeline.

Behavioral parity: reference pipelines/text_classification.py:134-231
(model-type registry :25-31, train :194-218, predict __call__ :220-231).
"""
from __future__ import annotations

import argparse
from typing import List, Union

import torch

from fengshen_amd.pipelines.base import BasePipeline


_MODEL_REGISTRY = {}


def _registry():
    if not _MODEL_REGISTRY:
        from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
            MegatronBertForSequenceClassification)
        from fengshen_amd.models.roformer.modeling_roformer import (
            RoFormerForSequenceClassification)
        _MODEL_REGISTRY.update({
            "megatron_bert": MegatronBertForSequenceClassification,
            "fengshen_megatron_bert": MegatronBertForSequenceClassification,
            "roformer": RoFormerForSequenceClassification,
            "fengshen_roformer": RoFormerForSequenceClassification,
        })
    return _MODEL_REGISTRY


class _ClsCollator:
    def __init__(self, tokenizer, max_length=512,
                 text_key="sentence", label_key="label"):
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.text_key = text_key
        self.label_key = label_key

    def __call__(self, samples):
        ids, masks, types, labels = [], [], [], []
        for s in samples:
            text = s[self.text_key] if isinstance(s, dict) else s
            enc = self.tokenizer.encode(text)[:self.max_length]
            ids.append(enc)
            if isinstance(s, dict) and self.label_key in s:
                labels.append(int(s[self.label_key]))
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
            batch["labels"] = torch.tensor(labels, dtype=torch.long)
        return batch


class TextClassificationPipeline(BasePipeline):
    """Sequence classification over Erlangshen-family backbones."""

    task_name = "text_classification"

    def __init__(self, args=None, model=None, tokenizer=None,
                 model_type: str = "megatron_bert", config=None):
        super().__init__(args, model, tokenizer)
        if self.model is None:
            cls = _registry()[model_type]
            if isinstance(model, str):
                self.model = cls.from_pretrained(model)
            else:
                assert config is not None, "pass model= path or config="
                self.model = cls(config)
        self.model.eval()

    @classmethod
    def add_pipeline_specific_args(cls, parser: argparse.ArgumentParser):
        parser = super().add_pipeline_specific_args(parser)
        g = parser.add_argument_group("text classification")
        g.add_argument("--texta_name", type=str, default="sentence")
        g.add_argument("--label_name", type=str, default="label")
        g.add_argument("--max_length", type=int, default=512)
        return parser

    def collator(self):
        return _ClsCollator(
            self.tokenizer,
            max_length=getattr(self.args, "max_length", 512),
            text_key=getattr(self.args, "texta_name", "sentence"),
            label_key=getattr(self.args, "label_name", "label"))

    @torch.no_grad()
    def __call__(self, texts: Union[str, List[str]]):
        single = isinstance(texts, str)
        if single:
            texts = [texts]
        batch = self.collator()([{"sentence": t} for t in texts])
        batch = {k: v.to(next(self.model.parameters()).device)
                 for k, v in batch.items()}
        out = self.model(**batch)
        probs = out.logits.float().softmax(-1)
        results = [{"label": int(p.argmax()), "score": float(p.max())}
                   for p in probs]
        return results[0] if single else results


Pipeline = TextClassificationPipeline
