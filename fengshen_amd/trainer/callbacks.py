"""Trainer callbacks: base class, LR monitor, throughput monitor.

Parity: PL callbacks used by the reference examples
(LearningRateMonitor in pretrain_erlangshen.py:223; the checkpoint callback
lives in utils/universal_checkpoint.py to mirror the reference layout).
"""
from __future__ import annotations

import time

import torch


class Callback:
    """Hook interface: subclass and override any on_* method."""
    def on_fit_start(self, trainer, module):
        pass

    def on_fit_end(self, trainer, module):
        pass

    def on_train_epoch_start(self, trainer, module):
        pass

    def on_train_epoch_end(self, trainer, module):
        pass

    def on_train_batch_start(self, trainer, module, batch, batch_idx):
        pass

    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        pass

    def on_validation_start(self, trainer, module):
        pass

    def on_validation_end(self, trainer, module):
        pass

    def on_save_checkpoint(self, trainer, module, checkpoint: dict):
        pass

    def on_load_checkpoint(self, trainer, module, checkpoint: dict):
        pass


class LearningRateMonitor(Callback):
    """Logs the first param group's lr each step (ref: PL LearningRateMonitor)."""

    def __init__(self, logging_interval: str = "step"):
        self.logging_interval = logging_interval

    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        if trainer.optimizer is not None and trainer._did_step:
            lr = trainer.optimizer.param_groups[0]["lr"]
            trainer._log_metric("lr", lr)


class ThroughputMonitor(Callback):
    """Per-step samples/sec + tokens/sec + HBM usage (ref: report_memory probe,
    utils/utils.py:62-74, + DeepSpeed throughput timers)."""

    def __init__(self, warmup_steps: int = 2, log_memory_every: int = 100):
        self.warmup = warmup_steps
        self.log_memory_every = log_memory_every
        self._t0 = None
        self._samples = 0
        self._tokens = 0

    def on_train_batch_start(self, trainer, module, batch, batch_idx):
        if trainer.global_step == self.warmup and self._t0 is None:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            self._t0 = time.perf_counter()
            self._samples = 0
            self._tokens = 0

    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        if self._t0 is None:
            return
        bs, toks = _batch_size_tokens(batch)
        self._samples += bs * trainer.strategy.data_parallel_world_size
        self._tokens += toks * trainer.strategy.data_parallel_world_size
        dt = time.perf_counter() - self._t0
        if dt > 0:
            trainer._log_metric("samples_per_sec", self._samples / dt)
            if self._tokens:
                trainer._log_metric("tokens_per_sec", self._tokens / dt)
        if (self.log_memory_every and trainer.global_step > 0
                and trainer.global_step % self.log_memory_every == 0
                and trainer.global_rank == 0):
            from fengshen_amd.utils.utils import report_memory
            report_memory(f"step {trainer.global_step}")


def _batch_size_tokens(batch):
    if isinstance(batch, dict):
        for key in ("input_ids", "input_id", "pixel_values", "labels"):
            if key in batch and torch.is_tensor(batch[key]):
                t = batch[key]
                return t.shape[0], t.numel() if t.dim() >= 2 else 0
        for v in batch.values():
            if torch.is_tensor(v) and v.dim() >= 1:
                return v.shape[0], 0
    if torch.is_tensor(batch):
        return batch.shape[0], 0
    if isinstance(batch, (list, tuple)) and batch and torch.is_tensor(batch[0]):
        return batch[0].shape[0], 0
    return 0, 0
