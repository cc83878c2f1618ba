"""Experiment loggers (ref: WandbLogger in examples/ziya_llama/
finetune_ziya_llama.py:218, TensorBoard default elsewhere in PL).

The image has neither the tensorboard nor the wandb package, so:
- TensorBoardLogger writes REAL TensorBoard event files natively: the
  TFRecord framing (u64 length + masked CRC32C, payload, masked CRC32C)
  around hand-encoded `Event`/`Summary` protobuf wire-format messages —
  readable by any stock TensorBoard.
- WandbLogger uses the real wandb package when importable, else an
  offline run directory (config.json + history.jsonl) in wandb's layout.
"""
from __future__ import annotations

import json
import os
import socket
import struct
import time
from typing import Dict, Optional

__all__ = ["Logger", "TensorBoardLogger", "WandbLogger"]


# ---------------------------------------------------------------------------
# CRC32C (Castagnoli), table-based — required by the TFRecord framing
# ---------------------------------------------------------------------------
_CRC_TABLE = []


def _crc_table():
    global _CRC_TABLE
    if _CRC_TABLE:
        return _CRC_TABLE
    poly = 0x82F63B78
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        _CRC_TABLE.append(c)
    return _CRC_TABLE


def _crc32c(data: bytes) -> int:
    table = _crc_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = table[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# minimal protobuf wire-format encoders (Event / Summary messages)
# ---------------------------------------------------------------------------
def _varint(n: int) -> bytes:
    out = b""
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _pb_double(field: int, v: float) -> bytes:
    return _tag(field, 1) + struct.pack("<d", v)


def _pb_float(field: int, v: float) -> bytes:
    return _tag(field, 5) + struct.pack("<f", v)


def _pb_int64(field: int, v: int) -> bytes:
    return _tag(field, 0) + _varint(v & 0xFFFFFFFFFFFFFFFF)


def _pb_bytes(field: int, v: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(v)) + v


def _pb_string(field: int, v: str) -> bytes:
    return _pb_bytes(field, v.encode("utf-8"))


def _event(wall_time: float, step: int, *, file_version: Optional[str] = None,
           tag: Optional[str] = None,
           simple_value: Optional[float] = None) -> bytes:
    """Event{wall_time=1, step=2, file_version=3, summary=5};
    Summary{value=1: Value{tag=1, simple_value=2}}."""
    body = _pb_double(1, wall_time)
    if step:
        body += _pb_int64(2, step)
    if file_version is not None:
        body += _pb_string(3, file_version)
    if tag is not None:
        value = _pb_string(1, tag) + _pb_float(2, float(simple_value))
        body += _pb_bytes(5, _pb_bytes(1, value))
    return body


class Logger:
    """Base logger interface."""

    def log_hyperparams(self, params: dict):
        pass

    def log_metrics(self, metrics: Dict[str, float], step: int):
        raise NotImplementedError

    def finalize(self):
        pass


class TensorBoardLogger(Logger):
    """Native TensorBoard event-file writer (no tensorboard package)."""

    def __init__(self, save_dir: str, name: str = "default",
                 version: Optional[str] = None):
        self.save_dir = save_dir
        ver = version if version is not None else f"version_{os.getpid()}"
        self.log_dir = os.path.join(save_dir, name, str(ver))
        os.makedirs(self.log_dir, exist_ok=True)
        fname = (f"events.out.tfevents.{int(time.time())}."
                 f"{socket.gethostname()}")
        self._path = os.path.join(self.log_dir, fname)
        self._f = open(self._path, "ab")
        self._write(_event(time.time(), 0, file_version="brain.Event:2"))

    def _write(self, payload: bytes):
        header = struct.pack("<Q", len(payload))
        self._f.write(header)
        self._f.write(struct.pack("<I", _masked_crc(header)))
        self._f.write(payload)
        self._f.write(struct.pack("<I", _masked_crc(payload)))
        self._f.flush()

    def log_hyperparams(self, params: dict):
        with open(os.path.join(self.log_dir, "hparams.json"), "w") as f:
            json.dump({k: v for k, v in params.items()
                       if isinstance(v, (int, float, str, bool, type(None)))},
                      f, indent=2, default=str)

    def log_metrics(self, metrics: Dict[str, float], step: int):
        now = time.time()
        for tag, value in metrics.items():
            try:
                v = float(value)
            except (TypeError, ValueError):
                continue
            self._write(_event(now, step, tag=tag, simple_value=v))

    def finalize(self):
        try:
            self._f.close()
        except Exception:
            pass


class WandbLogger(Logger):
    """wandb when installed; offline wandb-layout run dir otherwise."""

    def __init__(self, project: str = "fengshen_amd",
                 name: Optional[str] = None, save_dir: str = "./wandb",
                 config: Optional[dict] = None):
        self.project = project
        self.name = name or f"run-{int(time.time())}"
        self._wandb = None
        try:
            import wandb  # noqa: F401 — optional, absent in this image
            self._wandb = wandb
            self._run = wandb.init(project=project, name=name,
                                   config=config or {}, dir=save_dir)
        except Exception:
            self.run_dir = os.path.join(save_dir, self.name)
            os.makedirs(self.run_dir, exist_ok=True)
            self._hist = open(
                os.path.join(self.run_dir, "history.jsonl"), "a")
            if config:
                self.log_hyperparams(config)

    def log_hyperparams(self, params: dict):
        if self._wandb is not None:
            self._run.config.update(params, allow_val_change=True)
            return
        with open(os.path.join(self.run_dir, "config.json"), "w") as f:
            json.dump(params, f, indent=2, default=str)

    def log_metrics(self, metrics: Dict[str, float], step: int):
        if self._wandb is not None:
            self._wandb.log(metrics, step=step)
            return
        rec = {"_step": step, "_timestamp": time.time()}
        for k, v in metrics.items():
            try:
                rec[k] = float(v)
            except (TypeError, ValueError):
                pass
        self._hist.write(json.dumps(rec) + "\n")
        self._hist.flush()

    def finalize(self):
        if self._wandb is not None:
            self._run.finish()
        else:
            try:
                self._hist.close()
            except Exception:
                pass
