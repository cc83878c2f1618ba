"""Python oracles / fallbacks for the C++ index builders (data/csrc/helpers.cpp).

The C++ extension is the fast path for TB-scale corpora; these are the
reference implementations used when the extension isn't built and as the
equivalence oracle in tests.
"""
from __future__ import annotations

import logging
import os

import numpy as np

logger = logging.getLogger(__name__)

_EXT = None
_TRIED = False


def _ext():
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            from fengshen_amd.data import _helpers  # type: ignore
            _EXT = _helpers
        except ImportError:
            logger.info("data _helpers extension not built; using python "
                        "fallbacks (fine for tests, slow for TB corpora)")
    return _EXT


def py_build_sample_idx(sizes: np.ndarray, doc_idx: np.ndarray,
                        seq_length: int, num_epochs: int,
                        tokens_per_epoch: int) -> np.ndarray:
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    sample_idx = np.zeros((num_samples + 1, 2), dtype=np.int32)
    di, off = 0, 0
    sample_idx[0] = (di, off)
    for i in range(1, num_samples + 1):
        remaining = seq_length + 1
        while remaining != 0:
            doc_id = doc_idx[di]
            doc_len = sizes[doc_id] - off
            remaining -= doc_len
            if remaining <= 0:
                off += remaining + doc_len - 1
                remaining = 0
            else:
                di += 1
                off = 0
        sample_idx[i] = (di, off)
    return sample_idx


def build_sample_idx(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    ext = _ext()
    if ext is not None:
        return ext.build_sample_idx(
            np.asarray(sizes, dtype=np.int32),
            np.asarray(doc_idx, dtype=np.int32),
            seq_length, num_epochs, tokens_per_epoch)
    return py_build_sample_idx(np.asarray(sizes), np.asarray(doc_idx),
                               seq_length, num_epochs, tokens_per_epoch)


def build_mapping(docs, sizes, num_epochs, max_num_samples, max_seq_length,
                  short_seq_prob, seed):
    ext = _ext()
    if ext is not None:
        return ext.build_mapping(
            np.asarray(docs, dtype=np.int64),
            np.asarray(sizes, dtype=np.int32), num_epochs,
            max_num_samples, max_seq_length, short_seq_prob, seed)
    raise NotImplementedError(
        "BERT sample mapping needs the _helpers extension "
        "(python -m fengshen_amd.ops.build builds it)")


def py_build_blending_indices(weights: np.ndarray, size: int):
    n = len(weights)
    dataset_index = np.zeros(size, dtype=np.int8)
    dataset_sample_index = np.zeros(size, dtype=np.int64)
    current = np.zeros(n, dtype=np.int64)
    for i in range(size):
        errors = weights * (i + 1) - current
        pick = int(np.argmax(errors))
        dataset_index[i] = pick
        dataset_sample_index[i] = current[pick]
        current[pick] += 1
    return dataset_index, dataset_sample_index


def build_blending_indices(weights, size: int):
    ext = _ext()
    w = np.asarray(weights, dtype=np.float64)
    if ext is not None:
        return ext.build_blending_indices(w, size)
    return py_build_blending_indices(w, size)


def get_samples_mapping(indexed, data_prefix, num_epochs, max_num_samples,
                        max_seq_length, short_seq_prob, seed, name):
    """Cached samples mapping: build once, mmap-load thereafter
    (reference dataset_utils.py:731-788 — '180GB loads in seconds').

    The mapping is persisted next to the corpus as
    {data_prefix}_{name}_{epochs}ep_{max_seq}msl_{seed}s_indexmap.npy.
    """
    cache = (f"{data_prefix}_{name}_{num_epochs}ep_{max_seq_length}msl_"
             f"{seed}s_indexmap.npy")
    if os.path.exists(cache):
        return np.load(cache, mmap_mode="r")
    mapping = build_mapping(
        indexed.doc_idx, np.asarray(indexed.sizes, dtype=np.int32),
        num_epochs, max_num_samples, max_seq_length, short_seq_prob, seed)
    # write atomically so concurrent DP ranks never read a partial file
    tmp = f"{cache}.tmp{os.getpid()}.npy"
    with open(tmp, "wb") as f:
        np.save(f, mapping)
    os.replace(tmp, cache)
    return np.load(cache, mmap_mode="r")
