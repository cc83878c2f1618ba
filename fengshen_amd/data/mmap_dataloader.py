"""Simple multi-tensor mmap dataset (the lighter alternative to the
megatron stack).

Behavioral parity: reference data/mmap_dataloader/mmap_index_dataset.py:7
(MMapIndexDataset: per-field .npy index + .bin memmap) + mmap_datamodule
and utils/convert_py_to_npy.py:20.
"""
from __future__ import annotations

from typing import Dict, List, Sequence

import numpy as np
import torch


class MMapIndexDataset(torch.utils.data.Dataset):
    """Each field f has {prefix}_{f}.bin (int32 tokens back to back) and
    {prefix}_{f}.npy ([N+1] int64 offsets)."""

    def __init__(self, prefix: str, fields: Sequence[str] = ("input_ids",)):
        self.fields = list(fields)
        self._offsets = {}
        self._bins = {}
        n = None
        for f in self.fields:
            self._offsets[f] = np.load(f"{prefix}_{f}.npy")
            self._bins[f] = np.memmap(f"{prefix}_{f}.bin", dtype=np.int32,
                                      mode="r")
            fn = len(self._offsets[f]) - 1
            assert n is None or n == fn, "field lengths differ"
            n = fn
        self._len = n or 0

    def __len__(self):
        return self._len

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        out = {}
        for f in self.fields:
            off = self._offsets[f]
            arr = self._bins[f][off[idx]:off[idx + 1]]
            out[f] = torch.from_numpy(arr.astype(np.int64))
        return out


def convert_py_to_npy(samples: List[Dict[str, Sequence[int]]], prefix: str,
                      fields: Sequence[str] = ("input_ids",)):
    """Write python token lists into the .bin/.npy pair
    (reference utils/convert_py_to_npy.py:20)."""
    for f in fields:
        offsets = [0]
        chunks = []
        for s in samples:
            arr = np.asarray(s[f], dtype=np.int32)
            chunks.append(arr)
            offsets.append(offsets[-1] + len(arr))
        np.concatenate(chunks).tofile(f"{prefix}_{f}.bin")
        np.save(f"{prefix}_{f}.npy", np.asarray(offsets, dtype=np.int64))
