"""GPT pretraining dataset over mmap corpora + weighted blending.

Behavioral parity: reference data/megatron_dataloader/gpt_dataset-style
contiguous-token-stream sampling (helpers.cpp build_sample_idx) and
blendable_dataset.py:64 (weighted corpus mix).
"""
from __future__ import annotations

from typing import Optional, Sequence

import numpy as np
import torch

from fengshen_amd.data.helpers_py import build_blending_indices, build_sample_idx
from fengshen_amd.data.indexed_dataset import MMapIndexedDataset


class GPTDataset(torch.utils.data.Dataset):
    """seq_length+1 token windows over the shuffled concatenated corpus."""

    def __init__(self, indexed: MMapIndexedDataset, seq_length: int,
                 num_epochs: int = 1, seed: int = 1234,
                 documents: Optional[np.ndarray] = None):
        self.indexed = indexed
        self.seq_length = seq_length
        if documents is None:
            documents = np.arange(len(indexed.doc_idx) - 1, dtype=np.int32)
        rng = np.random.RandomState(seed)
        doc_idx = np.concatenate(
            [rng.permutation(documents) for _ in range(num_epochs)]
        ).astype(np.int32)
        sizes = indexed.sizes.astype(np.int32)
        tokens_per_epoch = int(sizes[documents].sum())
        self.sample_idx = build_sample_idx(
            sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
        self.doc_idx = doc_idx

    def __len__(self):
        return self.sample_idx.shape[0] - 1

    def __getitem__(self, idx: int):
        doc_f, off_f = self.sample_idx[idx]
        doc_l, off_l = self.sample_idx[idx + 1]
        if doc_f == doc_l:
            sample = self.indexed.get(int(self.doc_idx[doc_f]), offset=int(off_f),
                                      length=int(off_l) - int(off_f) + 1)
        else:
            parts = [self.indexed.get(int(self.doc_idx[doc_f]), offset=int(off_f))]
            for i in range(int(doc_f) + 1, int(doc_l)):
                parts.append(self.indexed.get(int(self.doc_idx[i])))
            parts.append(self.indexed.get(int(self.doc_idx[doc_l]),
                                          length=int(off_l) + 1))
            sample = np.concatenate(parts)
        ids = torch.from_numpy(sample.astype(np.int64))
        return {"input_ids": ids[:-1], "labels": ids[1:]}


class BlendableDataset(torch.utils.data.Dataset):
    """Weighted mix of datasets (reference blendable_dataset.py:64)."""

    def __init__(self, datasets: Sequence[torch.utils.data.Dataset],
                 weights: Sequence[float], size: Optional[int] = None):
        assert len(datasets) == len(weights) and datasets
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        self.datasets = list(datasets)
        self.size = size if size is not None else sum(len(d) for d in datasets)
        self.dataset_index, self.dataset_sample_index = \
            build_blending_indices(w, self.size)

    def __len__(self):
        return self.size

    def __getitem__(self, idx: int):
        d = int(self.dataset_index[idx])
        s = int(self.dataset_sample_index[idx]) % len(self.datasets[d])
        return self.datasets[d][s]
