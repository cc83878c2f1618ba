"""NER collators for the four tagging head families.

Behavioral parity: reference data/sequence_tagging_dataloader/
sequence_tagging_collator.py:9-206 — CollatorForLinear/Crf (BIO labels),
CollatorForSpan (per-position start/end entity-type vectors),
CollatorForBiaffine (dense [s, s] span-type matrix).  Samples carry
`text` (char sequence) and `entities` [(start, end, type_id)] (span/
biaffine) or `labels` (BIO strings, linear/crf).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List

import torch


def _encode_chars(tokenizer, text: str, max_length: int) -> List[int]:
    chars = list(text)[:max_length - 2]
    vocab = tokenizer.get_vocab()
    return [tokenizer.cls_token_id] + \
        [vocab.get(c, tokenizer.unk_token_id) for c in chars] + \
        [tokenizer.sep_token_id]


def _pad_batch(tokenizer, ids: List[List[int]]) -> Dict[str, torch.Tensor]:
    L = max(len(x) for x in ids)
    pad = tokenizer.pad_token_id or 0
    return {
        "input_ids": torch.tensor(
            [x + [pad] * (L - len(x)) for x in ids], dtype=torch.long),
        "attention_mask": torch.tensor(
            [[1] * len(x) + [0] * (L - len(x)) for x in ids],
            dtype=torch.long),
    }, L


@dataclass
class CollatorForLinear:
    """BIO token labels -> `labels` [b, s] (-100 on specials/pad)."""

    tokenizer: Any
    label2id: Dict[str, int]
    max_length: int = 256

    def __call__(self, samples):
        ids, labels = [], []
        for s in samples:
            ids.append(_encode_chars(self.tokenizer, s["text"],
                                     self.max_length))
            lab = [self.label2id.get(x, 0) for x in s.get("labels", [])]
            labels.append([-100] + lab[:self.max_length - 2] + [-100])
        batch, L = _pad_batch(self.tokenizer, ids)
        batch["labels"] = torch.tensor(
            [x + [-100] * (L - len(x)) for x in labels], dtype=torch.long)
        return batch


CollatorForCrf = CollatorForLinear  # same label format; head differs


@dataclass
class CollatorForSpan:
    """entities [(start, end, type)] -> start_positions/end_positions
    [b, s] with the entity type at span boundaries, 0 elsewhere."""

    tokenizer: Any
    max_length: int = 256

    def __call__(self, samples):
        ids, starts, ends = [], [], []
        for s in samples:
            enc = _encode_chars(self.tokenizer, s["text"], self.max_length)
            ids.append(enc)
            sp = [0] * len(enc)
            ep = [0] * len(enc)
            sp[0] = ep[0] = sp[-1] = ep[-1] = -100
            for (st, en, t) in s.get("entities", []):
                if st + 1 < len(enc) - 1 and en + 1 < len(enc) - 1:
                    sp[st + 1] = t   # +1 for [CLS]
                    ep[en + 1] = t
            starts.append(sp)
            ends.append(ep)
        batch, L = _pad_batch(self.tokenizer, ids)
        batch["start_positions"] = torch.tensor(
            [x + [-100] * (L - len(x)) for x in starts], dtype=torch.long)
        batch["end_positions"] = torch.tensor(
            [x + [-100] * (L - len(x)) for x in ends], dtype=torch.long)
        return batch


@dataclass
class CollatorForBiaffine:
    """entities -> dense span matrix `span_labels` [b, s, s]: type at
    (start, end), 0 on valid non-entity cells, -100 outside the text or
    below the diagonal."""

    tokenizer: Any
    max_length: int = 256

    def __call__(self, samples):
        ids = [_encode_chars(self.tokenizer, s["text"], self.max_length)
               for s in samples]
        batch, L = _pad_batch(self.tokenizer, ids)
        span = torch.full((len(samples), L, L), -100, dtype=torch.long)
        for bi, s in enumerate(samples):
            n = len(ids[bi])
            valid = torch.triu(torch.ones(n - 2, n - 2, dtype=torch.bool))
            span[bi, 1:n - 1, 1:n - 1][valid] = 0
            for (st, en, t) in s.get("entities", []):
                if st + 1 < n - 1 and en + 1 < n - 1 and st <= en:
                    span[bi, st + 1, en + 1] = t
        batch["span_labels"] = span
        return batch
