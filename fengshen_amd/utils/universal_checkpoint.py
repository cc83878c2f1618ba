"""UniversalCheckpoint — step-monitored checkpoint callback.

Behavioral parity: reference fengshen/utils/universal_checkpoint.py:5-41
(PL ModelCheckpoint subclass; monitors `step`, every_n_train_steps,
save_top_k, save_weights_only, save_last).  Ours writes the trainer's
sharded checkpoint-dir layout (model_part_{tp}.pt + optim shards + meta).
"""
from __future__ import annotations

import argparse
import os
import shutil
from typing import List, Optional, Tuple

from fengshen_amd.trainer.callbacks import Callback


class UniversalCheckpoint(Callback):
    @staticmethod
    def add_argparse_args(parent_args: argparse.ArgumentParser):
        parser = parent_args.add_argument_group("universal checkpoint callback")
        parser.add_argument("--monitor", default="step", type=str)
        parser.add_argument("--mode", default="max", type=str)
        parser.add_argument("--save_ckpt_path", default="./ckpt/", type=str)
        parser.add_argument("--load_ckpt_path", default="./ckpt/", type=str)
        parser.add_argument("--filename", default="model-{step:02d}", type=str)
        parser.add_argument("--save_last", action="store_true", default=False)
        parser.add_argument("--save_top_k", default=10, type=int)
        parser.add_argument("--every_n_train_steps", default=None, type=int)
        parser.add_argument("--save_weights_only", action="store_true", default=False)
        parser.add_argument("--every_n_epochs", default=None, type=int)
        parser.add_argument("--save_on_train_epoch_end", action="store_true",
                            default=None)
        return parent_args

    def __init__(self, args):
        self.monitor = getattr(args, "monitor", "step")
        self.mode = getattr(args, "mode", "max")
        self.save_ckpt_path = getattr(args, "save_ckpt_path", "./ckpt/")
        self.load_ckpt_path = getattr(args, "load_ckpt_path", None)
        self.filename = getattr(args, "filename", "model-{step:02d}")
        self.save_last = getattr(args, "save_last", False)
        self.save_top_k = getattr(args, "save_top_k", 10)
        self.every_n_train_steps = getattr(args, "every_n_train_steps", None)
        self.save_weights_only = getattr(args, "save_weights_only", False)
        self.every_n_epochs = getattr(args, "every_n_epochs", None)
        self._saved: List[Tuple[float, str]] = []

    # ------------------------------------------------------------------
    def _ckpt_dir(self, trainer) -> str:
        name = self.filename.format(step=trainer.global_step,
                                    epoch=trainer.current_epoch)
        for key, val in trainer._metrics.items():
            name = name.replace("{" + key + ":.4f}", f"{val:.4f}")
        return os.path.join(self.save_ckpt_path, name + ".ckpt")

    def _monitored_value(self, trainer) -> Optional[float]:
        if self.monitor == "step":
            return float(trainer.global_step)
        if self.monitor == "epoch":
            return float(trainer.current_epoch)
        return trainer._metrics.get(self.monitor)

    def _save(self, trainer):
        value = self._monitored_value(trainer)
        if value is None:
            return
        path = self._ckpt_dir(trainer)
        trainer.save_checkpoint(path, weights_only=self.save_weights_only)
        self._saved.append((value, path))
        self._prune(trainer)
        if self.save_last and trainer.global_rank == 0:
            last = os.path.join(self.save_ckpt_path, "last.ckpt")
            if os.path.islink(last) or os.path.exists(last):
                if os.path.islink(last):
                    os.unlink(last)
                else:
                    shutil.rmtree(last, ignore_errors=True)
            try:
                os.symlink(os.path.abspath(path), last)
            except OSError:
                pass

    def _prune(self, trainer):
        if self.save_top_k is None or self.save_top_k < 0:
            return
        reverse = self.mode == "max"
        self._saved.sort(key=lambda t: t[0], reverse=reverse)
        while len(self._saved) > self.save_top_k:
            _, path = self._saved.pop()
            if trainer.global_rank == 0 and os.path.exists(path):
                shutil.rmtree(path, ignore_errors=True)

    # ------------------------------------------------------------------
    def on_train_batch_end(self, trainer, module, outputs, batch, batch_idx):
        if (self.every_n_train_steps and trainer._did_step
                and trainer.global_step > 0
                and trainer.global_step % self.every_n_train_steps == 0):
            self._save(trainer)

    def on_train_epoch_end(self, trainer, module):
        if self.every_n_epochs and (
                trainer.current_epoch + 1) % self.every_n_epochs == 0:
            self._save(trainer)

    def on_fit_end(self, trainer, module):
        if self.save_last:
            self._save(trainer)
