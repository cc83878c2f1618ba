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


# ---------------------------------------------------------------------------
# Full-parity manifest dataset (ref data/hubert/hubert_dataset.py:39-360)
# with fairseq-Dictionary label semantics, multi-stream labels, offset
# streaming, pad/crop collation and frame-rate alignment — no fairseq import.
# ---------------------------------------------------------------------------
import itertools
import logging
import os
import sys
from typing import Any, Union

logger = logging.getLogger(__name__)


class LabelDictionary:
    """fairseq-Dictionary semantics for k-means label streams: fixed
    special slots (<s>=0 <pad>=1 </s>=2 <unk>=3) then symbols in add
    order; encode_line maps space-separated symbols to indices."""

    def __init__(self, symbols: Optional[Sequence[str]] = None):
        self.symbols: List[str] = ["<s>", "<pad>", "</s>", "<unk>"]
        self.indices = {s: i for i, s in enumerate(self.symbols)}
        for s in symbols or []:
            self.add_symbol(str(s))

    def __len__(self):
        return len(self.symbols)

    def add_symbol(self, sym: str) -> int:
        if sym not in self.indices:
            self.indices[sym] = len(self.symbols)
            self.symbols.append(sym)
        return self.indices[sym]

    def bos(self):
        return 0

    def pad(self):
        return 1

    def eos(self):
        return 2

    def unk(self):
        return 3

    def index(self, sym: str) -> int:
        return self.indices.get(sym, self.unk())

    def string(self, ids) -> str:
        return " ".join(self.symbols[int(i)] for i in ids)

    def encode_line(self, line: str, append_eos: bool = False,
                    add_if_not_exist: bool = False) -> torch.Tensor:
        ids = []
        for sym in line.split():
            if add_if_not_exist:
                ids.append(self.add_symbol(sym))
            else:
                ids.append(self.index(sym))
        if append_eos:
            ids.append(self.eos())
        return torch.tensor(ids, dtype=torch.long)

    @classmethod
    def load(cls, path: str) -> "LabelDictionary":
        """Load a fairseq dict.*.txt ("symbol count" per line)."""
        d = cls()
        with open(path) as f:
            for line in f:
                parts = line.rstrip().split(" ")
                if parts and parts[0]:
                    d.add_symbol(parts[0])
        return d


def load_audio(manifest_path: str, max_keep: Optional[int],
               min_keep: Optional[int]):
    """Parse a fairseq-style TSV manifest (root line, name\tsize rows)
    filtering by sample count (ref :39-64)."""
    n_long, n_short = 0, 0
    names, inds, sizes = [], [], []
    ind = -1
    with open(manifest_path) as f:
        root = f.readline().strip()
        for ind, line in enumerate(f):
            items = line.strip().split("\t")
            assert len(items) == 2, line
            sz = int(items[1])
            if min_keep is not None and sz < min_keep:
                n_short += 1
            elif max_keep is not None and sz > max_keep:
                n_long += 1
            else:
                names.append(items[0])
                inds.append(ind)
                sizes.append(sz)
    tot = ind + 1
    logger.info("max_keep=%s, min_keep=%s, loaded %d, skipped %d short "
                "and %d long", max_keep, min_keep, len(names), n_short,
                n_long)
    return root, names, inds, tot, sizes


def load_label(label_path: str, inds, tot: int):
    with open(label_path) as f:
        labels = [line.rstrip() for line in f]
        assert len(labels) == tot, (len(labels), tot)
        return [labels[i] for i in inds]


def load_label_offset(label_path: str, inds, tot: int):
    """Byte offsets per label line for streaming reads (ref :77-86)."""
    with open(label_path) as f:
        code_lengths = [len(line.encode("utf-8")) for line in f]
    assert len(code_lengths) == tot, (len(code_lengths), tot)
    offsets = list(itertools.accumulate([0] + code_lengths))
    return [(offsets[i], offsets[i + 1]) for i in inds]


def verify_label_lengths(audio_sizes, audio_rate, label_path, label_rate,
                         inds, tot, tol=0.1):
    """Warn when audio and frame-label durations diverge (ref :88-124)."""
    if label_rate < 0:
        logger.info("%s is sequence label. skipped", label_path)
        return 0
    with open(label_path) as f:
        lengths = [len(line.rstrip().split()) for line in f]
    assert len(lengths) == tot
    lengths = [lengths[i] for i in inds]
    num_invalid = 0
    for i, ind in enumerate(inds):
        dur_a = audio_sizes[i] / audio_rate
        dur_l = lengths[i] / label_rate
        if abs(dur_a - dur_l) > tol:
            logger.warning(
                "audio/label duration differ (|%s - %s| > %s) line %d of %s",
                dur_a, dur_l, tol, ind + 1, label_path)
            num_invalid += 1
    if num_invalid:
        logger.warning("total %d (audio, label) pairs with mismatched "
                       "lengths", num_invalid)
    return num_invalid


def _default_audio_loader(path: str):
    """npy waveform loader (no soundfile in the image); returns
    (wav ndarray, sample_rate or None)."""
    return np.load(path), None


class HubertManifestDataset(torch.utils.data.Dataset):
    """Reference-parity HuBERT dataset (ref HubertDataset :127-360):
    manifest + label files, Dictionary label processors, multi-stream
    labels with per-stream rates (-1 = sequence labels), pad/crop audio
    collation with frame-aligned label windows."""

    def __init__(self, manifest_path: str, sample_rate: float,
                 label_paths: List[str],
                 label_rates: Union[List[float], float],
                 pad_list: List[int],
                 label_processors: Optional[List[Any]] = None,
                 max_keep_sample_size: Optional[int] = None,
                 min_keep_sample_size: Optional[int] = None,
                 max_sample_size: Optional[int] = None,
                 shuffle: bool = True, pad_audio: bool = False,
                 normalize: bool = False, store_labels: bool = True,
                 random_crop: bool = False, single_target: bool = False,
                 audio_loader=None):
        (self.audio_root, self.audio_names, inds, tot,
         self.sizes) = load_audio(manifest_path, max_keep_sample_size,
                                  min_keep_sample_size)
        self.sample_rate = sample_rate
        self.shuffle = shuffle
        self.random_crop = random_crop
        self.audio_loader = audio_loader or _default_audio_loader

        self.num_labels = len(label_paths)
        self.pad_list = pad_list
        self.label_processors = label_processors
        self.single_target = single_target
        self.label_rates = ([label_rates] * len(label_paths)
                            if isinstance(label_rates, (int, float))
                            else list(label_rates))
        self.store_labels = store_labels
        if store_labels:
            self.label_list = [load_label(p, inds, tot)
                               for p in label_paths]
        else:
            self.label_paths = label_paths
            self.label_offsets_list = [load_label_offset(p, inds, tot)
                                       for p in label_paths]
        assert (label_processors is None
                or len(label_processors) == self.num_labels)
        for label_path, label_rate in zip(label_paths, self.label_rates):
            verify_label_lengths(self.sizes, sample_rate, label_path,
                                 label_rate, inds, tot)

        self.max_sample_size = (max_sample_size if max_sample_size
                                is not None else sys.maxsize)
        self.pad_audio = pad_audio
        self.normalize = normalize

    # -- item access ----------------------------------------------------
    def get_audio(self, index):
        wav_path = os.path.join(self.audio_root, self.audio_names[index])
        wav, cur_sr = self.audio_loader(wav_path)
        wav = torch.as_tensor(np.asarray(wav), dtype=torch.float32)
        return self.postprocess(wav, cur_sr or self.sample_rate)

    def get_label(self, index, label_idx):
        if self.store_labels:
            label = self.label_list[label_idx][index]
        else:
            with open(self.label_paths[label_idx]) as f:
                s, e = self.label_offsets_list[label_idx][index]
                f.seek(s)
                label = f.read(e - s)
        if self.label_processors is not None:
            label = self.label_processors[label_idx](label)
        return label

    def get_labels(self, index):
        return [self.get_label(index, i) for i in range(self.num_labels)]

    def __getitem__(self, index):
        return {"id": index, "source": self.get_audio(index),
                "label_list": self.get_labels(index)}

    def __len__(self):
        return len(self.sizes)

    # -- collation ------------------------------------------------------
    def crop_to_max_size(self, wav, target_size):
        size = len(wav)
        diff = size - target_size
        if diff <= 0:
            return wav, 0
        start, end = 0, target_size
        if self.random_crop:
            start = np.random.randint(0, diff + 1)
            end = size - diff + start
        return wav[start:end], start

    def collater(self, samples):
        samples = [s for s in samples if s["source"] is not None]
        if not samples:
            return {}
        audios = [s["source"] for s in samples]
        audio_sizes = [len(a) for a in audios]
        if self.pad_audio:
            audio_size = min(max(audio_sizes), self.max_sample_size)
        else:
            audio_size = min(min(audio_sizes), self.max_sample_size)
        collated_audios, padding_mask, audio_starts = self.collater_audio(
            audios, audio_size)
        targets_by_label = [[s["label_list"][i] for s in samples]
                            for i in range(self.num_labels)]
        targets_list, lengths_list, ntokens_list = self.collater_label(
            targets_by_label, audio_size, audio_starts)
        batch = {
            "id": torch.LongTensor([s["id"] for s in samples]),
            "net_input": {"source": collated_audios,
                          "padding_mask": padding_mask},
        }
        if self.single_target:
            batch["target_lengths"] = lengths_list[0]
            batch["ntokens"] = ntokens_list[0]
            batch["target"] = targets_list[0]
        else:
            batch["target_lengths_list"] = lengths_list
            batch["ntokens_list"] = ntokens_list
            batch["target_list"] = targets_list
        return batch

    def collater_audio(self, audios, audio_size):
        collated = audios[0].new_zeros(len(audios), audio_size)
        padding_mask = torch.zeros(collated.shape, dtype=torch.bool)
        audio_starts = [0] * len(audios)
        for i, audio in enumerate(audios):
            diff = len(audio) - audio_size
            if diff == 0:
                collated[i] = audio
            elif diff < 0:
                assert self.pad_audio
                collated[i] = torch.cat(
                    [audio, audio.new_full((-diff,), 0.0)])
                padding_mask[i, diff:] = True
            else:
                collated[i], audio_starts[i] = self.crop_to_max_size(
                    audio, audio_size)
        return collated, padding_mask, audio_starts

    @staticmethod
    def _collate_tokens(targets, pad_idx):
        n = max(len(t) for t in targets)
        out = targets[0].new_full((len(targets), n), pad_idx)
        for i, t in enumerate(targets):
            out[i, :len(t)] = t
        return out

    def collater_frm_label(self, targets, audio_size, audio_starts,
                           label_rate, pad):
        """Frame labels: window to the cropped audio via label_rate /
        sample_rate alignment (ref :294-310)."""
        assert label_rate > 0
        s2f = label_rate / self.sample_rate
        frm_starts = [int(round(s * s2f)) for s in audio_starts]
        frm_size = int(round(audio_size * s2f))
        if not self.pad_audio:
            rem_size = [len(t) - s for t, s in zip(targets, frm_starts)]
            frm_size = min(frm_size, *rem_size)
        targets = [t[s:s + frm_size] for t, s in zip(targets, frm_starts)]
        lengths = torch.LongTensor([len(t) for t in targets])
        ntokens = int(lengths.sum())
        return self._collate_tokens(targets, pad), lengths, ntokens

    def collater_seq_label(self, targets, pad):
        lengths = torch.LongTensor([len(t) for t in targets])
        ntokens = int(lengths.sum())
        return self._collate_tokens(targets, pad), lengths, ntokens

    def collater_label(self, targets_by_label, audio_size, audio_starts):
        targets_list, lengths_list, ntokens_list = [], [], []
        for targets, label_rate, pad in zip(
                targets_by_label, self.label_rates, self.pad_list):
            if label_rate == -1.0:
                t, ln, nt = self.collater_seq_label(targets, pad)
            else:
                t, ln, nt = self.collater_frm_label(
                    targets, audio_size, audio_starts, label_rate, pad)
            targets_list.append(t)
            lengths_list.append(ln)
            ntokens_list.append(nt)
        return targets_list, lengths_list, ntokens_list

    # -- sizing / ordering ----------------------------------------------
    def num_tokens(self, index):
        return self.size(index)

    def size(self, index):
        if self.pad_audio:
            return self.sizes[index]
        return min(self.sizes[index], self.max_sample_size)

    def ordered_indices(self):
        """Shuffle then stable size-descending order (ref :341-348)."""
        if self.shuffle:
            order = [np.random.permutation(len(self))]
        else:
            order = [np.arange(len(self))]
        order.append(self.sizes)
        return np.lexsort(order)[::-1]

    def postprocess(self, wav, cur_sample_rate):
        if wav.dim() == 2:
            wav = wav.mean(-1)
        assert wav.dim() == 1, wav.dim()
        if cur_sample_rate != self.sample_rate:
            raise ValueError(
                f"sr {cur_sample_rate} != target {self.sample_rate}")
        if self.normalize:
            with torch.no_grad():
                wav = torch.nn.functional.layer_norm(wav, wav.shape)
        return wav
