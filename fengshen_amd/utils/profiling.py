"""Step profiler callback — torch.profiler window over chosen train steps.

Aux-subsystem parity (SURVEY.md §5 tracing): the reference only exposes
DeepSpeed's profile flag + a manual memory probe; here a first-class
callback captures a kernel-level trace (Chrome format, works with the ROCm
backend) plus the per-kernel table, writeable next to rocprofv3 output.
"""
from __future__ import annotations

import logging
import os

import torch

from fengshen_amd.trainer.callbacks import Callback

logger = logging.getLogger(__name__)


class StepProfiler(Callback):
    def __init__(self, start_step: int = 5, num_steps: int = 2,
                 out_dir: str = "./profile_out", export_trace: bool = True,
                 row_limit: int = 30):
        self.start_step = start_step
        self.num_steps = num_steps
        self.out_dir = out_dir
        self.export_trace = export_trace
        self.row_limit = row_limit
        self._prof = None
        self._done = False

    def on_train_batch_start(self, trainer, module, batch, batch_idx):
        if self._done or trainer.global_rank != 0:
            return
        if trainer.global_step == self.start_step and self._prof is None:
            acts = [torch.profiler.ProfilerActivity.CPU]
            if torch.cuda.is_available():
                acts.append(torch.profiler.ProfilerActivity.CUDA)
            self._prof = torch.profiler.profile(
                activities=acts, record_shapes=False, with_stack=False)
            self._prof.__enter__()

    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        if self._prof is None or self._done:
            return
        if trainer.global_step >= self.start_step + self.num_steps:
            self._prof.__exit__(None, None, None)
            os.makedirs(self.out_dir, exist_ok=True)
            table = self._prof.key_averages().table(
                sort_by="cuda_time_total" if torch.cuda.is_available()
                else "cpu_time_total", row_limit=self.row_limit)
            with open(os.path.join(self.out_dir, "step_profile.txt"), "w") as f:
                f.write(table)
            if self.export_trace:
                self._prof.export_chrome_trace(
                    os.path.join(self.out_dir, "step_trace.json"))
            logger.info("step profile written to %s", self.out_dir)
            print(table.splitlines()[0:5], flush=True)
            self._prof = None
            self._done = True


class RocTXMarker(Callback):
    """Per-step rocTX ranges (torch.cuda.nvtx maps to rocTX on ROCm).

    Makes steps visible in `rocprofv3 --marker-trace` timelines so kernel
    traces can be cut per step; no-op overhead when no profiler attached.
    """

    def __init__(self, prefix: str = "fengshen_step"):
        self.prefix = prefix
        self._open = False

    def on_train_batch_start(self, trainer, module, batch, batch_idx):
        torch.cuda.nvtx.range_push(f"{self.prefix}_{trainer.global_step}")
        self._open = True

    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        if self._open:
            torch.cuda.nvtx.range_pop()
            self._open = False
