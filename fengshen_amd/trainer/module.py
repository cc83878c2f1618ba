"""FengshenModule — the Lightning-style module API.

Behavioral parity with the reference's use of pytorch_lightning.LightningModule
(every example in fengshen/examples/* subclasses it): users implement
``training_step`` / ``validation_step`` / ``configure_optimizers`` / ``setup``
and hand the module to :class:`fengshen_amd.trainer.Trainer`.  The loop itself
is ours (the reference outsources it to PL `Trainer.fit`).
"""
from __future__ import annotations

import logging
from typing import Any, Dict, Optional

import torch
import torch.nn as nn

logger = logging.getLogger(__name__)


class _HParams(dict):
    """Attribute-style access to hyperparameters (PL's hparams behavior)."""

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k, v):
        self[k] = v


class FengshenModule(nn.Module):
    """Base class for trainable modules.

    Overridable hooks (all optional except training_step):
      setup(stage)                     -- called once before fit/validate
      training_step(batch, batch_idx)  -> loss tensor or dict w/ 'loss'
      validation_step(batch, batch_idx)
      test_step(batch, batch_idx)
      predict_step(batch, batch_idx)
      configure_optimizers()           -> optimizer | (optimizer, scheduler_cfg)
      on_train_batch_start/end, on_validation_epoch_end, on_save_checkpoint,
      on_load_checkpoint, on_fit_start, on_fit_end
    """

    def __init__(self):
        super().__init__()
        self.trainer = None  # set by Trainer.fit
        self._hparams = _HParams()

    # -- hyperparameters ---------------------------------------------------
    def save_hyperparameters(self, args=None, ignore=()):
        if args is None:
            return
        src = vars(args) if not isinstance(args, dict) else args
        for k, v in src.items():
            if k not in ignore:
                self._hparams[k] = v

    @property
    def hparams(self) -> _HParams:
        return self._hparams

    # -- trainer-provided context -----------------------------------------
    @property
    def global_rank(self) -> int:
        return self.trainer.global_rank if self.trainer is not None else 0

    @property
    def local_rank(self) -> int:
        return self.trainer.local_rank if self.trainer is not None else 0

    @property
    def world_size(self) -> int:
        return self.trainer.world_size if self.trainer is not None else 1

    @property
    def global_step(self) -> int:
        return self.trainer.global_step if self.trainer is not None else 0

    @property
    def current_epoch(self) -> int:
        return self.trainer.current_epoch if self.trainer is not None else 0

    @property
    def device(self) -> torch.device:
        try:
            return next(self.parameters()).device
        except StopIteration:
            return torch.device("cpu")

    def log(self, name: str, value, prog_bar: bool = False, sync_dist: bool = False,
            on_step: bool = True, on_epoch: bool = False, **_kw):
        """Record a scalar metric.  sync_dist=True averages across ranks
        (reference: PL self.log(..., sync_dist=True), e.g.
        finetune_ziya_llama.py:147,152)."""
        if self.trainer is not None:
            self.trainer._log_metric(name, value, sync_dist=sync_dist, prog_bar=prog_bar)

    def log_dict(self, metrics: Dict[str, Any], **kw):
        for k, v in metrics.items():
            self.log(k, v, **kw)

    def print(self, *args, **kw):
        if self.global_rank == 0:
            print(*args, **kw)

    # -- hooks (default no-ops) --------------------------------------------
    def setup(self, stage: Optional[str] = None):
        pass

    def configure_optimizers(self):
        raise NotImplementedError

    def training_step(self, batch, batch_idx: int):
        raise NotImplementedError

    def validation_step(self, batch, batch_idx: int):
        pass

    def test_step(self, batch, batch_idx: int):
        pass

    def predict_step(self, batch, batch_idx: int):
        pass

    def on_fit_start(self):
        pass

    def on_fit_end(self):
        pass

    def on_train_batch_start(self, batch, batch_idx: int):
        pass

    def on_train_batch_end(self, outputs, batch, batch_idx: int):
        pass

    def on_train_epoch_start(self):
        pass

    def on_train_epoch_end(self):
        pass

    def on_validation_epoch_start(self):
        pass

    def on_validation_epoch_end(self):
        pass

    def on_save_checkpoint(self, checkpoint: Dict[str, Any]):
        pass

    def on_load_checkpoint(self, checkpoint: Dict[str, Any]):
        pass
