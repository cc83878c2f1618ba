"""HuBERT audio SSL dataset.

Behavioral parity: reference data/hubert/hubert_dataset.py:127-218 (audio +
k-means pseudo-label pairs, crop to max_sample_size, label-rate alignment) —
without the fairseq Dictionary dependency: labels are plain int arrays.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np
import torch


class HubertDataset(torch.utils.data.Dataset):
    def __init__(self, waveforms: Sequence[np.ndarray],
                 labels: Sequence[Sequence[int]],
                 sample_rate: int = 16000, label_rate: float = 50.0,
                 max_sample_size: Optional[int] = None,
                 min_sample_size: int = 0, random_crop: bool = True,
                 seed: int = 1234):
        assert len(waveforms) == len(labels)
        keep = [i for i, w in enumerate(waveforms)
                if len(w) >= min_sample_size]
        self.waveforms = [np.asarray(waveforms[i], dtype=np.float32)
                          for i in keep]
        self.labels = [np.asarray(labels[i], dtype=np.int64) for i in keep]
        self.sample_rate = sample_rate
        self.label_rate = label_rate
        self.max_sample_size = max_sample_size
        self.random_crop = random_crop
        self._rng = np.random.RandomState(seed)

    def __len__(self):
        return len(self.waveforms)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        wav = self.waveforms[idx]
        lab = self.labels[idx]
        if self.max_sample_size and len(wav) > self.max_sample_size:
            start = self._rng.randint(0, len(wav) - self.max_sample_size + 1) \
                if self.random_crop else 0
            wav = wav[start:start + self.max_sample_size]
            # align labels to the cropped window (label_rate per second)
            l0 = int(start / self.sample_rate * self.label_rate)
            l1 = int((start + len(wav)) / self.sample_rate * self.label_rate)
            lab = lab[l0:max(l1, l0 + 1)]
        return {"source": torch.from_numpy(wav.copy()),
                "label": torch.from_numpy(lab.copy())}

    def collater(self, samples: List[Dict]) -> Dict[str, torch.Tensor]:
        max_w = max(len(s["source"]) for s in samples)
        max_l = max(len(s["label"]) for s in samples)
        src = torch.zeros(len(samples), max_w)
        pad_mask = torch.ones(len(samples), max_w, dtype=torch.bool)
        lab = torch.full((len(samples), max_l), -100, dtype=torch.long)
        for i, s in enumerate(samples):
            src[i, :len(s["source"])] = s["source"]
            pad_mask[i, :len(s["source"])] = False
            lab[i, :len(s["label"])] = s["label"]
        return {"source": src, "padding_mask": pad_mask, "labels": lab}
