"""BERT pretraining dataset over mmap sentence corpora.

Behavioral parity: reference data/megatron_dataloader/bert_dataset.py:30
(BertDataset: samples mapping via helpers build_mapping, then per-sample
build_training_sample = A/B split + truncate + [CLS]/[SEP] + wwm masking,
per-sample seeded numpy RNG :76-77).
"""
from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch

from fengshen_amd.data.data_utils import (
    create_masked_lm_predictions,
    create_tokens_and_tokentypes,
    get_a_and_b_segments,
    truncate_segments,
)
from fengshen_amd.data.helpers_py import build_mapping
from fengshen_amd.data.indexed_dataset import MMapIndexedDataset


class BertMmapDataset(torch.utils.data.Dataset):
    """Each indexed-dataset item is one SENTENCE; doc boundaries come from
    the index's doc_idx.  Produces MLM+SOP training samples."""

    def __init__(self, indexed: MMapIndexedDataset, vocab_id_list,
                 vocab_id_to_token: Dict[int, str], *, cls_id: int,
                 sep_id: int, mask_id: int, pad_id: int,
                 max_seq_length: int = 512, masked_lm_prob: float = 0.15,
                 short_seq_prob: float = 0.1, num_epochs: int = 1,
                 max_num_samples: Optional[int] = None, seed: int = 1234,
                 data_prefix: Optional[str] = None):
        self.indexed = indexed
        self.vocab_id_list = vocab_id_list
        self.vocab_id_to_token = vocab_id_to_token
        self.cls_id, self.sep_id = cls_id, sep_id
        self.mask_id, self.pad_id = mask_id, pad_id
        self.max_seq_length = max_seq_length
        self.masked_lm_prob = masked_lm_prob
        self.seed = seed
        if max_num_samples is None:
            max_num_samples = 2 ** 62
        if data_prefix is not None:
            # persisted .npy indexmap so TB corpora build the mapping once
            from fengshen_amd.data.helpers_py import get_samples_mapping
            self.samples_mapping = get_samples_mapping(
                indexed, data_prefix, num_epochs, max_num_samples,
                max_seq_length - 3, short_seq_prob, seed, "bert")
        else:
            self.samples_mapping = build_mapping(
                indexed.doc_idx, indexed.sizes.astype(np.int32), num_epochs,
                max_num_samples, max_seq_length - 3, short_seq_prob, seed)

    def __len__(self):
        return len(self.samples_mapping)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        start, end, target_len = self.samples_mapping[idx]
        sents = [self.indexed.get(int(i)).tolist()
                 for i in range(int(start), int(end))]
        # per-sample seeded RNG (ref bert_dataset.py:76-77)
        rng = np.random.RandomState(seed=(self.seed + idx) % 2 ** 31)
        if len(sents) == 1:
            half = max(1, len(sents[0]) // 2)
            sents = [sents[0][:half], sents[0][half:] or [self.sep_id]]
        tokens_a, tokens_b, is_random = get_a_and_b_segments(sents, rng)
        truncate_segments(tokens_a, tokens_b, len(tokens_a), len(tokens_b),
                          int(target_len), rng)
        tokens, tokentypes = create_tokens_and_tokentypes(
            tokens_a, tokens_b, self.cls_id, self.sep_id)
        max_pred = int(self.max_seq_length * self.masked_lm_prob) + 1
        tokens, positions, labels_ = create_masked_lm_predictions(
            tokens, self.vocab_id_list, self.vocab_id_to_token,
            self.masked_lm_prob, self.cls_id, self.sep_id, self.mask_id,
            max_pred, rng)
        L = self.max_seq_length
        labels = [-100] * len(tokens)
        for p, l in zip(positions, labels_):
            labels[p] = l
        pad = L - len(tokens)
        return {
            "input_ids": torch.tensor(tokens[:L] + [self.pad_id] * max(pad, 0),
                                      dtype=torch.long),
            "attention_mask": torch.tensor(
                [1] * min(len(tokens), L) + [0] * max(pad, 0), dtype=torch.long),
            "token_type_ids": torch.tensor(
                tokentypes[:L] + [0] * max(pad, 0), dtype=torch.long),
            "labels": torch.tensor(labels[:L] + [-100] * max(pad, 0),
                                   dtype=torch.long),
            "next_sentence_label": torch.tensor(int(is_random)),
        }
