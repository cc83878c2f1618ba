"""Memory-mapped token dataset (.bin + .idx), TB-scale pretraining path.

Behavioral parity: reference data/megatron_dataloader/indexed_dataset.py
(MMapIndexedDataset :344, builders, uint16-when-vocab<65500 rule :24-27).
Format-compatible with the megatron/fairseq MMIDIDX layout so existing
preprocessed corpora load directly:

  idx: magic b'MMIDIDX\\x00\\x00' | version u64 | dtype_code u8 |
       n_sequences u64 | n_documents u64 | sizes i32[n_seq] |
       pointers i64[n_seq] | doc_idx i64[n_docs+1]
  bin: raw token arrays back to back
"""
from __future__ import annotations

import os
import shutil
import struct
from typing import List, Optional

import numpy as np
import torch

_INDEX_HEADER = b"MMIDIDX\x00\x00"

_DTYPES = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float64,
    7: np.float32,
    8: np.uint16,
}
_DTYPE_CODES = {np.dtype(v): k for k, v in _DTYPES.items()}


def best_fitting_dtype(vocab_size: Optional[int]) -> np.dtype:
    """uint16 when vocab < 65500 (reference indexed_dataset.py:24-27)."""
    if vocab_size is not None and vocab_size < 65500:
        return np.dtype(np.uint16)
    return np.dtype(np.int32)


def data_file_path(prefix: str) -> str:
    return prefix + ".bin"


def index_file_path(prefix: str) -> str:
    return prefix + ".idx"


class MMapIndexedDataset(torch.utils.data.Dataset):
    def __init__(self, path: str, skip_warmup: bool = True):
        self._path = path
        with open(index_file_path(path), "rb") as f:
            magic = f.read(9)
            assert magic == _INDEX_HEADER, f"bad index magic in {path}"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            (dtype_code,) = struct.unpack("<B", f.read(1))
            self._dtype = np.dtype(_DTYPES[dtype_code])
            (self._len,) = struct.unpack("<Q", f.read(8))
            (self._doc_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()
        self._index_buffer = np.memmap(index_file_path(path), mode="r",
                                       order="C")
        self._sizes = np.frombuffer(self._index_buffer, dtype=np.int32,
                                    count=self._len, offset=offset)
        ptr_off = offset + self._sizes.nbytes
        self._pointers = np.frombuffer(self._index_buffer, dtype=np.int64,
                                       count=self._len, offset=ptr_off)
        doc_off = ptr_off + self._pointers.nbytes
        self._doc_idx = np.frombuffer(self._index_buffer, dtype=np.int64,
                                      count=self._doc_count, offset=doc_off)
        self._bin_buffer = np.memmap(data_file_path(path), mode="r", order="C")

    def __len__(self) -> int:
        return self._len

    @property
    def sizes(self) -> np.ndarray:
        return self._sizes

    @property
    def doc_idx(self) -> np.ndarray:
        return self._doc_idx

    @property
    def dtype(self):
        return self._dtype

    def size(self, idx: int) -> int:
        return int(self._sizes[idx])

    def get(self, idx: int, offset: int = 0, length: Optional[int] = None):
        ptr = int(self._pointers[idx]) + offset * self._dtype.itemsize
        if length is None:
            length = int(self._sizes[idx]) - offset
        return np.frombuffer(self._bin_buffer, dtype=self._dtype,
                             count=length, offset=ptr)

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            start, stop, step = idx.indices(len(self))
            assert step == 1
            return [self.get(i) for i in range(start, stop)]
        return self.get(idx)

    @staticmethod
    def exists(path: str) -> bool:
        return (os.path.exists(index_file_path(path))
                and os.path.exists(data_file_path(path)))


class MMapIndexedDatasetBuilder:
    def __init__(self, out_file: str, dtype=np.int32):
        self._data_file = open(out_file, "wb")
        self._dtype = np.dtype(dtype)
        self._sizes: List[int] = []
        self._doc_idx: List[int] = [0]

    def add_item(self, tensor_or_array):
        arr = np.asarray(tensor_or_array, dtype=self._dtype)
        self._data_file.write(arr.tobytes(order="C"))
        self._sizes.append(arr.size)

    def end_document(self):
        self._doc_idx.append(len(self._sizes))

    def merge_file_(self, another_prefix: str):
        index = MMapIndexedDataset(another_prefix)
        assert index.dtype == self._dtype
        doc_offset = len(self._sizes)
        for size in index.sizes:
            self._sizes.append(int(size))
        self._doc_idx.extend([doc_offset + int(d) for d in index.doc_idx[1:]])
        with open(data_file_path(another_prefix), "rb") as f:
            shutil.copyfileobj(f, self._data_file)

    def finalize(self, index_file: str):
        self._data_file.close()
        sizes = np.array(self._sizes, dtype=np.int32)
        pointers = np.zeros(len(sizes), dtype=np.int64)
        np.cumsum(sizes[:-1] * self._dtype.itemsize, out=pointers[1:])
        doc_idx = np.array(self._doc_idx, dtype=np.int64)
        with open(index_file, "wb") as f:
            f.write(_INDEX_HEADER)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", _DTYPE_CODES[self._dtype]))
            f.write(struct.pack("<Q", len(sizes)))
            f.write(struct.pack("<Q", len(doc_idx)))
            f.write(sizes.tobytes(order="C"))
            f.write(pointers.tobytes(order="C"))
            f.write(doc_idx.tobytes(order="C"))


def make_builder(out_file: str, vocab_size: Optional[int] = None):
    return MMapIndexedDatasetBuilder(out_file,
                                     dtype=best_fitting_dtype(vocab_size))


def make_dataset(path: str, skip_warmup: bool = True) -> MMapIndexedDataset:
    assert MMapIndexedDataset.exists(path), f"no .bin/.idx at {path}"
    return MMapIndexedDataset(path, skip_warmup)


# ----------------------------------------------------------------------
# Legacy (non-mmap) fairseq format: TNTIDX header, explicit offset arrays
# (reference indexed_dataset.py:130-216 IndexedDataset /
#  :217 IndexedCachedDataset).  Kept for loading old preprocessed corpora.

_LEGACY_MAGIC = b"TNTIDX\x00\x00"


class IndexedDataset(torch.utils.data.Dataset):
    """Reader for the legacy TNTIDX .bin/.idx pair (file reads, no mmap)."""

    def __init__(self, path: str):
        self.path = path
        self.data_file = None
        with open(index_file_path(path), "rb") as f:
            magic = f.read(8)
            assert magic == _LEGACY_MAGIC, "not a legacy TNTIDX index"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            code, self.element_size = struct.unpack("<QQ", f.read(16))
            self.dtype = np.dtype(_DTYPES[code])
            self._len, s = struct.unpack("<QQ", f.read(16))
            (doc_count,) = struct.unpack("<Q", f.read(8))
            self.dim_offsets = np.fromfile(f, dtype=np.int64,
                                           count=self._len + 1)
            self.data_offsets = np.fromfile(f, dtype=np.int64,
                                            count=self._len + 1)
            self.sizes = np.fromfile(f, dtype=np.int64, count=s)
            self.doc_idx = np.fromfile(f, dtype=np.int64, count=doc_count)

    def __len__(self):
        return self._len

    def _read(self, idx: int) -> np.ndarray:
        if self.data_file is None:
            self.data_file = open(data_file_path(self.path), "rb",
                                  buffering=0)
        size = int(self.data_offsets[idx + 1] - self.data_offsets[idx])
        a = np.empty(size, dtype=self.dtype)
        self.data_file.seek(int(self.data_offsets[idx]) * self.element_size)
        self.data_file.readinto(a)
        return a

    def __getitem__(self, idx: int) -> np.ndarray:
        if idx < 0 or idx >= self._len:
            raise IndexError("index out of range")
        return self._read(idx)

    def __del__(self):
        if self.data_file:
            self.data_file.close()

    @staticmethod
    def exists(path: str) -> bool:
        if not os.path.exists(index_file_path(path)):
            return False
        with open(index_file_path(path), "rb") as f:
            return f.read(8) == _LEGACY_MAGIC


class IndexedCachedDataset(IndexedDataset):
    """Legacy reader + in-RAM prefetch cache (reference :217-270)."""

    def __init__(self, path: str):
        super().__init__(path)
        self.cache = {}

    def prefetch(self, indices):
        for i in sorted(set(indices)):
            self.cache[i] = super()._read(i)

    def __getitem__(self, idx: int) -> np.ndarray:
        if idx in self.cache:
            return self.cache[idx]
        return super().__getitem__(idx)


class IndexedDatasetBuilder:
    """Writer for the legacy TNTIDX format (reference :271-343)."""

    def __init__(self, out_file: str, dtype=np.int32):
        self.out_file = open(out_file, "wb")
        self.dtype = np.dtype(dtype)
        self.element_size = self.dtype.itemsize
        self.data_offsets = [0]
        self.dim_offsets = [0]
        self.sizes: List[int] = []
        self.doc_idx = [0]

    def add_item(self, tensor_or_array):
        a = np.asarray(tensor_or_array, dtype=self.dtype)
        self.out_file.write(a.tobytes(order="C"))
        self.data_offsets.append(self.data_offsets[-1] + a.size)
        self.sizes.append(a.size)
        self.dim_offsets.append(self.dim_offsets[-1] + 1)

    def end_document(self):
        self.doc_idx.append(len(self.sizes))

    def finalize(self, index_file: str):
        self.out_file.close()
        with open(index_file, "wb") as f:
            f.write(_LEGACY_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<QQ", _DTYPE_CODES[self.dtype],
                                self.element_size))
            f.write(struct.pack("<QQ", len(self.data_offsets) - 1,
                                len(self.sizes)))
            f.write(struct.pack("<Q", len(self.doc_idx)))
            np.asarray(self.dim_offsets, dtype=np.int64).tofile(f)
            np.asarray(self.data_offsets, dtype=np.int64).tofile(f)
            np.asarray(self.sizes, dtype=np.int64).tofile(f)
            np.asarray(self.doc_idx, dtype=np.int64).tofile(f)


def infer_dataset_impl(path: str) -> Optional[str]:
    """'mmap' | 'cached' | None by magic (reference :33-46)."""
    if not os.path.exists(index_file_path(path)):
        return None
    with open(index_file_path(path), "rb") as f:
        magic = f.read(9)
    if magic[:8] == _LEGACY_MAGIC:
        return "cached"
    if magic == _INDEX_HEADER:
        return "mmap"
    return None
