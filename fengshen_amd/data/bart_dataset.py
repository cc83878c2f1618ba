"""BART denoising dataset over mmap sentence corpora.

Behavioral parity: reference data/megatron_dataloader/bart_dataset.py:13-98
(BartDataset.build_training_sample): sentences joined with [SEP] after
sentence-final punctuation, sentence-order permutation
(permute_sentences :190), whole-word masking with span collapse +
10% random-token replacement (add_whole_word_mask :290), target =
un-noised tokens shifted by one, pad/-100 fill to max_seq_length.
Chinese word starts use the ##-continuation convention
(word_starts :218).  Per-sample numpy RNG seeded with (seed + idx) for
exact reproducibility.
"""
from __future__ import annotations

import math
from typing import Dict, Optional, Set

import numpy as np
import torch

from fengshen_amd.data.helpers_py import build_mapping
from fengshen_amd.data.indexed_dataset import MMapIndexedDataset


class BartMmapDataset(torch.utils.data.Dataset):
    """Each indexed-dataset item is one SENTENCE; doc boundaries from the
    index's doc_idx.  Produces (noised input_ids, denoising labels)."""

    def __init__(self, indexed: MMapIndexedDataset,
                 vocab_id_to_token: Dict[int, str], *, cls_id: int,
                 sep_id: int, mask_id: int, pad_id: int, vocab_size: int,
                 max_seq_length: int = 512, masked_lm_prob: float = 0.15,
                 permute_sentence_ratio: float = 1.0,
                 random_ratio: float = 0.1,
                 seg_token_ids: Optional[Set[int]] = None,
                 short_seq_prob: float = 0.1, num_epochs: int = 1,
                 max_num_samples: Optional[int] = None, seed: int = 1234,
                 data_prefix: Optional[str] = None):
        self.indexed = indexed
        self.vocab_id_to_token = vocab_id_to_token
        self.cls_id, self.sep_id = cls_id, sep_id
        self.mask_id, self.pad_id = mask_id, pad_id
        self.vocab_size = vocab_size
        self.max_seq_length = max_seq_length
        self.mask_ratio = masked_lm_prob
        self.permute_sentence_ratio = permute_sentence_ratio
        self.random_ratio = random_ratio
        self.seg_token_ids = seg_token_ids or set()
        self.seed = seed
        if max_num_samples is None:
            max_num_samples = 2 ** 62
        if data_prefix is not None:
            # persisted .npy indexmap so TB corpora build the mapping once
            from fengshen_amd.data.helpers_py import get_samples_mapping
            self.samples_mapping = get_samples_mapping(
                indexed, data_prefix, num_epochs, max_num_samples,
                max_seq_length - 3, short_seq_prob, seed, "bart")
        else:
            self.samples_mapping = build_mapping(
                indexed.doc_idx, indexed.sizes.astype(np.int32), num_epochs,
                max_num_samples, max_seq_length - 3, short_seq_prob, seed)

    def __len__(self):
        return len(self.samples_mapping)

    # ------------------------------------------------------------------
    def _word_starts(self, tokens: np.ndarray) -> np.ndarray:
        """1 where a new word starts (## continuation => 0); specials 0
        (ref word_starts :218-252)."""
        starts = np.zeros(len(tokens), dtype=bool)
        for i, t in enumerate(tokens):
            if t in (self.cls_id, self.sep_id, self.pad_id):
                continue
            piece = self.vocab_id_to_token.get(int(t), "")
            starts[i] = not piece.startswith("##")
        return starts

    def _permute_sentences(self, tokens: np.ndarray,
                           np_rng) -> np.ndarray:
        """Shuffle [SEP]-delimited sentences, keep leading [CLS]
        (ref permute_sentences :190-208)."""
        ends = np.nonzero(tokens == self.sep_id)[0]
        if len(ends) <= 1:
            return tokens
        spans = []
        start = 1  # skip [CLS]
        for e in ends:
            spans.append(tokens[start:e + 1])
            start = e + 1
        order = np_rng.permutation(len(spans))
        out = [tokens[:1]]
        out.extend(spans[i] for i in order)
        if start < len(tokens):
            out.append(tokens[start:])
        return np.concatenate(out)

    def _whole_word_mask(self, tokens: np.ndarray, p: float,
                         np_rng) -> np.ndarray:
        """Mask whole words: word start -> [MASK] (10% random token),
        continuations dropped (span collapse, replace_length=1; ref
        add_whole_word_mask :290-406)."""
        is_start = self._word_starts(tokens)
        word_starts = np.nonzero(is_start)[0]
        num_to_mask = int(math.ceil(len(word_starts) * p))
        if num_to_mask == 0 or len(word_starts) == 0:
            return tokens
        chosen = np_rng.permutation(len(word_starts))[:num_to_mask]
        mask_pos = word_starts[chosen]
        keep = np.ones(len(tokens), dtype=bool)
        out = tokens.copy()
        for pos in mask_pos:
            if np_rng.random() < self.random_ratio:
                out[pos] = np_rng.randint(0, self.vocab_size)
            else:
                out[pos] = self.mask_id
            # collapse the rest of the word
            j = pos + 1
            while j < len(tokens) and not is_start[j] and \
                    tokens[j] not in (self.sep_id, self.cls_id):
                keep[j] = False
                j += 1
        return out[keep]

    # ------------------------------------------------------------------
    def build_training_sample(self, sentences, np_rng) -> Dict[str, torch.Tensor]:
        tokens = [self.cls_id]
        for sent in sentences:
            for t in sent:
                tokens.append(int(t))
                if int(t) in self.seg_token_ids:
                    tokens.append(self.sep_id)
            if tokens[-1] != self.sep_id:
                tokens.append(self.sep_id)
        if len(tokens) > self.max_seq_length:
            tokens = tokens[:self.max_seq_length]
            tokens[-1] = self.sep_id
        tokens = np.asarray(tokens, dtype=np.int64)

        target = tokens[1:].copy()
        source = tokens
        if self.permute_sentence_ratio > 0:
            source = self._permute_sentences(source, np_rng)
        if self.mask_ratio > 0:
            # ref :140-142: doubled ratio when the decoder reconstructs
            source = self._whole_word_mask(source, self.mask_ratio * 2,
                                           np_rng)
        assert source[0] == self.cls_id and source[-1] == self.sep_id

        L = self.max_seq_length
        src = np.full(L, self.pad_id, dtype=np.int64)
        src[:len(source)] = source[:L]
        lab = np.full(L, -100, dtype=np.int64)
        lab[:len(target)] = target[:L]
        return {
            "input_ids": torch.from_numpy(src),
            "labels": torch.from_numpy(lab),
            "attention_mask": torch.from_numpy(
                (src != self.pad_id).astype(np.int64)),
        }

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        start, end, _ = self.samples_mapping[idx]
        sentences = [self.indexed[i] for i in range(start, end)]
        np_rng = np.random.RandomState(seed=(self.seed + idx) % 2 ** 32)
        return self.build_training_sample(sentences, np_rng)
