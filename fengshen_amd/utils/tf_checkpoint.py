"""TF checkpoint importer without TensorFlow.

Behavioral parity: reference utils/convert_tf_checkpoint_to_pytorch.py
(which calls tf.train.load_variable).  TensorFlow is not in this image, so
this module reads the TF TensorBundle format natively:

- ``<ckpt>.index``: a LevelDB-format SSTable (written by TF with
  compression OFF for bundle indexes) whose keys are tensor names and
  whose values are BundleEntryProto protobufs (dtype, shape, shard,
  offset, size);
- ``<ckpt>.data-00000-of-NNNNN``: raw little-endian tensor bytes at the
  recorded offsets.

Implements just enough of each format: SSTable footer/block/prefix-key
decoding and the protobuf wire-format fields used by BundleEntryProto /
TensorShapeProto.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, Iterator, List, Tuple

import numpy as np

_TABLE_MAGIC = 0xDB4775248B80FB57

# TF DataType enum -> numpy dtype (the ones that appear in checkpoints)
_TF_DTYPES = {
    1: np.float32, 2: np.float64, 3: np.int32, 4: np.uint8, 5: np.int16,
    6: np.int8, 9: np.int64, 10: np.bool_, 14: np.uint16, 17: np.uint16,
    19: np.float16, 22: np.uint32, 23: np.uint64,
}
_TF_BFLOAT16 = 14


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _block_entries(block: bytes) -> Iterator[Tuple[bytes, bytes]]:
    """Decode a LevelDB table block (prefix-compressed keys)."""
    if len(block) < 4:
        return
    (num_restarts,) = struct.unpack_from("<I", block, len(block) - 4)
    data_end = len(block) - 4 - 4 * num_restarts
    pos = 0
    key = b""
    while pos < data_end:
        shared, pos = _read_varint(block, pos)
        unshared, pos = _read_varint(block, pos)
        vlen, pos = _read_varint(block, pos)
        key = key[:shared] + block[pos:pos + unshared]
        pos += unshared
        value = block[pos:pos + vlen]
        pos += vlen
        yield key, value


def _read_block(data: bytes, offset: int, size: int) -> bytes:
    """Block payload + 1-byte compression type + 4-byte crc."""
    ctype = data[offset + size]
    if ctype != 0:
        raise ValueError(
            f"compressed table block (type {ctype}) unsupported — TF "
            "bundle indexes are written uncompressed")
    return data[offset:offset + size]


def _parse_sstable(data: bytes) -> Dict[bytes, bytes]:
    (magic,) = struct.unpack_from("<Q", data, len(data) - 8)
    if magic != _TABLE_MAGIC:
        raise ValueError("not a LevelDB-format table (bad magic)")
    footer = data[len(data) - 48:len(data) - 8]
    # metaindex handle then index handle, varint-encoded
    _mi_off, p = _read_varint(footer, 0)
    _mi_sz, p = _read_varint(footer, p)
    idx_off, p = _read_varint(footer, p)
    idx_sz, p = _read_varint(footer, p)
    entries: Dict[bytes, bytes] = {}
    index_block = _read_block(data, idx_off, idx_sz)
    for _k, handle in _block_entries(index_block):
        boff, hp = _read_varint(handle, 0)
        bsz, hp = _read_varint(handle, hp)
        for key, value in _block_entries(_read_block(data, boff, bsz)):
            entries[key] = value
    return entries


def _parse_shape(buf: bytes) -> List[int]:
    """TensorShapeProto: repeated Dim dim = 2 { int64 size = 1; }."""
    dims: List[int] = []
    pos = 0
    while pos < len(buf):
        tag, pos = _read_varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if field == 2 and wire == 2:  # dim
            ln, pos = _read_varint(buf, pos)
            dim_msg = buf[pos:pos + ln]
            pos += ln
            dpos = 0
            size = 1
            while dpos < len(dim_msg):
                dtag, dpos = _read_varint(dim_msg, dpos)
                if dtag >> 3 == 1 and dtag & 7 == 0:  # size
                    size, dpos = _read_varint(dim_msg, dpos)
                elif dtag & 7 == 2:
                    ln2, dpos = _read_varint(dim_msg, dpos)
                    dpos += ln2
                else:
                    _, dpos = _read_varint(dim_msg, dpos)
            dims.append(size)
        elif wire == 2:
            ln, pos = _read_varint(buf, pos)
            pos += ln
        elif wire == 0:
            _, pos = _read_varint(buf, pos)
        else:
            raise ValueError(f"unexpected wire type {wire} in shape")
    return dims


def _parse_bundle_entry(buf: bytes) -> dict:
    """BundleEntryProto: dtype=1, shape=2, shard_id=3, offset=4, size=5."""
    out = {"dtype": 0, "shape": [], "shard_id": 0, "offset": 0, "size": 0}
    pos = 0
    while pos < len(buf):
        tag, pos = _read_varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            val, pos = _read_varint(buf, pos)
            if field == 1:
                out["dtype"] = val
            elif field == 3:
                out["shard_id"] = val
            elif field == 4:
                out["offset"] = val
            elif field == 5:
                out["size"] = val
        elif wire == 2:
            ln, pos = _read_varint(buf, pos)
            if field == 2:
                out["shape"] = _parse_shape(buf[pos:pos + ln])
            pos += ln
        elif wire == 5:
            pos += 4
        elif wire == 1:
            pos += 8
        else:
            raise ValueError(f"unexpected wire type {wire}")
    return out


class TFCheckpointReader:
    """Minimal tf.train.load_checkpoint equivalent."""

    def __init__(self, prefix: str):
        self.prefix = prefix
        with open(prefix + ".index", "rb") as f:
            raw = _parse_sstable(f.read())
        self.entries: Dict[str, dict] = {}
        num_shards = 1
        for key, value in raw.items():
            if key == b"":
                continue  # BundleHeaderProto (num_shards etc.)
            self.entries[key.decode("utf-8")] = _parse_bundle_entry(value)
        self._shards: Dict[int, np.memmap] = {}
        self.num_shards = num_shards

    def _shard_path(self, shard_id: int) -> str:
        import glob
        pat = f"{self.prefix}.data-{shard_id:05d}-of-*"
        matches = glob.glob(pat)
        if not matches:
            raise FileNotFoundError(pat)
        return matches[0]

    def variable_names(self) -> List[str]:
        return sorted(self.entries)

    def get_variable_shape(self, name: str) -> List[int]:
        return list(self.entries[name]["shape"])

    def load_variable(self, name: str) -> np.ndarray:
        e = self.entries[name]
        path = self._shard_path(e["shard_id"])
        if e["shard_id"] not in self._shards:
            self._shards[e["shard_id"]] = np.memmap(path, dtype=np.uint8,
                                                    mode="r")
        raw = bytes(self._shards[e["shard_id"]][
            e["offset"]:e["offset"] + e["size"]])
        if e["dtype"] == _TF_BFLOAT16:
            u16 = np.frombuffer(raw, dtype=np.uint16)
            arr = (u16.astype(np.uint32) << 16).view(np.float32)
        else:
            np_dtype = _TF_DTYPES.get(e["dtype"])
            if np_dtype is None:
                raise ValueError(f"unsupported TF dtype {e['dtype']}")
            arr = np.frombuffer(raw, dtype=np_dtype)
        return arr.reshape(e["shape"]).copy()


# ---------------------------------------------------------------------------
# BERT name mapping (ref convert_tf_checkpoint_to_pytorch.py semantics)
# ---------------------------------------------------------------------------
def convert_tf_bert_to_state_dict(prefix: str) -> Dict[str, "np.ndarray"]:
    """Google-BERT TF names -> HF/fengshen BERT names with kernel
    transposition (TF dense kernels are [in, out])."""
    reader = TFCheckpointReader(prefix)
    out: Dict[str, np.ndarray] = {}
    for name in reader.variable_names():
        if any(tok in name for tok in
               ("adam_m", "adam_v", "global_step", "beta1_power",
                "beta2_power")):
            continue
        arr = reader.load_variable(name)
        pt = (name
              .replace("bert/", "bert.")
              .replace("encoder/layer_", "encoder.layer.")
              .replace("embeddings/", "embeddings.")
              .replace("attention/self/", "attention.self.")
              .replace("attention/output/", "attention.output.")
              .replace("intermediate/", "intermediate.")
              .replace("output/", "output.")
              .replace("pooler/", "pooler.")
              .replace("cls/predictions/", "cls.predictions.")
              .replace("cls/seq_relationship/", "cls.seq_relationship.")
              .replace("transform/", "transform.")
              .replace("/", "."))
        if pt.endswith(".kernel"):
            pt = pt[:-len("kernel")] + "weight"
            arr = arr.T
        elif pt.endswith(".gamma"):
            pt = pt[:-len("gamma")] + "weight"
        elif pt.endswith(".beta"):
            pt = pt[:-len("beta")] + "bias"
        elif pt.endswith(".output_bias"):
            pt = pt[:-len("output_bias")] + "bias"
        elif pt.endswith(".output_weights"):
            pt = pt[:-len("output_weights")] + "weight"
        if pt.endswith("word_embeddings") or pt.endswith(
                "position_embeddings") or pt.endswith(
                "token_type_embeddings"):
            pt = pt + ".weight"
        out[pt] = arr
    return out
