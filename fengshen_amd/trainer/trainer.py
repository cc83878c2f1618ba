"""The Trainer — owns the fit/validate/predict loops.

The reference delegates its loop to pytorch_lightning.Trainer.fit
(SURVEY.md §1 "the training loop lives in PL's Trainer.fit"); here the loop
is first-class.  Responsibilities: device placement, precision, gradient
accumulation, grad clipping, optimizer/scheduler stepping, validation
cadence, metric logging, checkpoint save/resume with exact data-order resume
(consumed_samples), callbacks.

Argparse parity: ``Trainer.add_argparse_args`` exposes the PL flag names the
reference examples use (max_epochs, max_steps, precision, strategy,
gradient_clip_val, accumulate_grad_batches, val_check_interval, ...).
"""
from __future__ import annotations

import argparse
import logging
import os
import random
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from fengshen_amd.trainer.callbacks import Callback
from fengshen_amd.trainer.strategies import Strategy, parse_strategy

logger = logging.getLogger(__name__)


def seed_everything(seed: int):
    """Seed python/numpy/torch (+CUDA) for reproducible runs."""
    random.seed(seed)
    np.random.seed(seed % (2 ** 32))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def _move_to_device(batch, device):
    if torch.is_tensor(batch):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, dict):
        return {k: _move_to_device(v, device) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        t = type(batch)
        return t(_move_to_device(v, device) for v in batch)
    return batch


class CSVLogger:
    """Rank-0 metrics.csv sink (header re-emitted when keys change)."""

    def __init__(self, root: str, rank: int):
        self.rank = rank
        self.path = os.path.join(root, "metrics.csv")
        self._keys: List[str] = []
        if rank == 0:
            os.makedirs(root, exist_ok=True)

    def log(self, step: int, metrics: Dict[str, float]):
        if self.rank != 0 or not metrics:
            return
        new_keys = [k for k in metrics if k not in self._keys]
        if new_keys:
            self._keys += new_keys
            with open(self.path, "a") as f:
                f.write("# step," + ",".join(self._keys) + "\n")
        with open(self.path, "a") as f:
            f.write(f"{step}," + ",".join(
                f"{metrics.get(k, '')}" for k in self._keys) + "\n")


class Trainer:
    """The training orchestrator (see module docstring for scope).

    Construct with keyword flags (or `Trainer.from_argparse_args`), then
    `fit(module, datamodule=... | train_dataloaders=...)`;
    `validate` / `predict` reuse the same strategy setup.
    """

    def __init__(
        self,
        max_steps: int = -1,
        max_epochs: int = -1,
        precision: str = "bf16",
        strategy: str = "auto",
        tensor_model_parallel_size: int = 1,
        pipe_model_parallel_size: int = 1,
        mpu_seed: int = 42,
        accumulate_grad_batches: int = 1,
        gradient_clip_val: float = 0.0,
        val_check_interval: Optional[float] = None,
        check_val_every_n_epoch: int = 1,
        limit_val_batches: Optional[int] = None,
        limit_train_batches: Optional[int] = None,
        log_every_n_steps: int = 10,
        callbacks: Optional[List[Callback]] = None,
        default_root_dir: str = "./fengshen_output",
        seed: int = 1234,
        num_sanity_val_steps: int = 0,
        activation_checkpointing: bool = False,
        ckpt_skip_interval: int = 0,
        zero_bucket_numel: int = 128 * 1024 * 1024,
        overlap_comm: bool = True,
        logger=None,
        **_unused,
    ):
        s = parse_strategy(strategy)
        self.strategy = Strategy(
            kind=s["kind"], stage=s["stage"],
            tensor_model_parallel_size=tensor_model_parallel_size,
            pipe_model_parallel_size=pipe_model_parallel_size,
            mpu_seed=mpu_seed, bucket_numel=zero_bucket_numel,
            overlap_comm=overlap_comm,
            cpu_offload=s.get("cpu_offload", False))
        self.max_steps = max_steps
        self.max_epochs = max_epochs
        self.precision = precision
        self.accumulate_grad_batches = max(1, accumulate_grad_batches)
        self.gradient_clip_val = gradient_clip_val
        self.val_check_interval = val_check_interval
        self.check_val_every_n_epoch = check_val_every_n_epoch
        self.limit_val_batches = limit_val_batches
        self.limit_train_batches = limit_train_batches
        self.log_every_n_steps = log_every_n_steps
        self.callbacks: List[Callback] = list(callbacks or [])
        self.default_root_dir = default_root_dir
        self.seed = seed
        self.seed_offset = 0
        self.num_sanity_val_steps = num_sanity_val_steps
        self.activation_checkpointing = activation_checkpointing
        self.ckpt_skip_interval = ckpt_skip_interval

        # loop state
        self.global_step = 0
        self.global_samples = 0
        self.current_epoch = 0
        self.should_stop = False
        self.optimizer = None
        self.scheduler_cfg = None
        self.module = None
        self.datamodule = None
        self._metrics: Dict[str, float] = {}
        self._did_step = False
        self._csv = None
        # experiment loggers (TensorBoardLogger / WandbLogger / custom);
        # CSV always written in addition (ref: PL logger= semantics)
        if logger is None:
            self.loggers = []
        elif isinstance(logger, (list, tuple)):
            self.loggers = list(logger)
        else:
            self.loggers = [logger]

    # ------------------------------------------------------------------
    @property
    def global_rank(self) -> int:
        return int(os.environ.get("RANK", "0"))

    @property
    def local_rank(self) -> int:
        return int(os.environ.get("LOCAL_RANK", self.global_rank % max(
            torch.cuda.device_count(), 1) if torch.cuda.is_available() else 0))

    @property
    def world_size(self) -> int:
        return int(os.environ.get("WORLD_SIZE", "1"))

    @property
    def device(self) -> torch.device:
        if torch.cuda.is_available():
            return torch.device("cuda", self.local_rank)
        return torch.device("cpu")

    @property
    def consumed_samples(self) -> int:
        return self.global_samples

    @property
    def estimated_stepping_batches(self) -> int:
        """PL-compat: total optimizer steps for the run (used by schedulers)."""
        if self.max_steps and self.max_steps > 0:
            return self.max_steps
        if self.datamodule is not None and self.max_epochs > 0:
            try:
                loader = self.datamodule.train_dataloader()
                steps_per_epoch = len(loader) // self.accumulate_grad_batches
                return steps_per_epoch * self.max_epochs
            except TypeError:
                pass
        return 1000000

    # ------------------------------------------------------------------
    @staticmethod
    def add_argparse_args(parent_parser: argparse.ArgumentParser):
        p = parent_parser.add_argument_group("Trainer")
        p.add_argument("--max_steps", type=int, default=-1)
        p.add_argument("--max_epochs", type=int, default=-1)
        p.add_argument("--precision", type=str, default="bf16")
        p.add_argument("--strategy", type=str, default="auto")
        p.add_argument("--tensor_model_parallel_size", type=int, default=1)
        p.add_argument("--pipe_model_parallel_size", type=int, default=1)
        p.add_argument("--mpu_seed", type=int, default=42)
        p.add_argument("--accumulate_grad_batches", type=int, default=1)
        p.add_argument("--gradient_clip_val", type=float, default=0.0)
        p.add_argument("--val_check_interval", type=float, default=None)
        p.add_argument("--check_val_every_n_epoch", type=int, default=1)
        p.add_argument("--limit_val_batches", type=int, default=None)
        p.add_argument("--limit_train_batches", type=int, default=None)
        p.add_argument("--log_every_n_steps", type=int, default=10)
        p.add_argument("--default_root_dir", type=str, default="./fengshen_output")
        p.add_argument("--seed", type=int, default=1234)
        p.add_argument("--num_sanity_val_steps", type=int, default=0)
        p.add_argument("--activation_checkpointing", action="store_true", default=False)
        p.add_argument("--ckpt_skip_interval", type=int, default=0,
                       help="selective act-ckpt: every k-th layer keeps "
                            "activations (0 = checkpoint all layers)")
        p.add_argument("--zero_bucket_numel", type=int, default=128 * 1024 * 1024)
        return parent_parser

    @classmethod
    def from_argparse_args(cls, args, callbacks=None, **kwargs):
        known = {k: v for k, v in vars(args).items()}
        known.update(kwargs)
        return cls(callbacks=callbacks, **known)

    # ------------------------------------------------------------------
    def _log_metric(self, name: str, value, sync_dist: bool = False, prog_bar=False):
        if torch.is_tensor(value):
            if sync_dist:
                value = self.strategy.reduce_metric(value.float())
            value = value.detach().float().item() if value.numel() == 1 else None
        if value is not None:
            self._metrics[name] = value

    def _flush_logs(self):
        if self.global_rank == 0 and self._metrics:
            if self._csv is not None:
                self._csv.log(self.global_step, self._metrics)
            for lg in self.loggers:
                lg.log_metrics(self._metrics, self.global_step)
            msg = " | ".join(
                f"{k} {v:.6g}" if isinstance(v, float) else f"{k} {v}"
                for k, v in self._metrics.items())
            logger.info("step %d | %s", self.global_step, msg)
            print(f"[step {self.global_step}] {msg}", flush=True)

    def _call(self, hook: str, *args):
        for cb in self.callbacks:
            getattr(cb, hook)(self, self.module, *args)

    # ------------------------------------------------------------------
    def fit(self, model, datamodule=None, train_dataloaders=None,
            val_dataloaders=None, ckpt_path: Optional[str] = None):
        self.strategy.setup_environment(self)
        seed_everything(self.seed)
        self._csv = CSVLogger(self.default_root_dir, self.global_rank)
        if self.global_rank == 0:
            hp = getattr(model, "hparams", None)
            if hp:
                for lg in self.loggers:
                    lg.log_hyperparams(dict(hp) if not isinstance(hp, dict)
                                       else hp)
        self.module = model
        model.trainer = self
        self.datamodule = datamodule
        if datamodule is not None:
            datamodule.trainer = self

        # restore loop counters BEFORE dataloaders so samplers resume exactly
        resume_state = None
        if ckpt_path is not None and os.path.exists(ckpt_path):
            resume_state = self._peek_checkpoint(ckpt_path)
            self.global_step = resume_state["global_step"]
            self.global_samples = resume_state["global_samples"]
            self.current_epoch = resume_state.get("epoch", 0)
        elif ckpt_path is not None:
            logger.warning("ckpt_path %s not found; training from scratch "
                           "(reference behavior: universal_checkpoint.py:38-41)",
                           ckpt_path)
            ckpt_path = None

        if datamodule is not None and hasattr(datamodule, "setup"):
            datamodule.setup("fit")
        model.setup("fit")

        if self.activation_checkpointing:
            # enable on every submodule that supports it (apps usually hold
            # the HF model at module.model / module.bert / ...)
            for m in model.modules():
                fn = getattr(m, "gradient_checkpointing_enable", None)
                if fn is not None and m is not model:
                    try:
                        fn(skip_interval=self.ckpt_skip_interval)
                    except TypeError:
                        fn()

        dev = self.device
        wrapped = self.strategy.setup_model(model, dev, self.precision)
        self.module = wrapped
        opt_out = wrapped.configure_optimizers()
        optimizer, scheduler_cfg = self._parse_optim_output(opt_out)
        self.optimizer, self.scheduler_cfg = self.strategy.setup_optimizers(
            wrapped, optimizer, scheduler_cfg)

        if ckpt_path is not None:
            self._load_checkpoint(ckpt_path)

        train_loader = train_dataloaders
        if train_loader is None and datamodule is not None:
            train_loader = datamodule.train_dataloader()
        val_loader = val_dataloaders
        if val_loader is None and datamodule is not None and hasattr(
                datamodule, "val_dataloader"):
            try:
                val_loader = datamodule.val_dataloader()
            except Exception:
                val_loader = None

        wrapped.on_fit_start()
        self._call("on_fit_start")

        if self.num_sanity_val_steps and val_loader is not None:
            self._run_validation(val_loader, limit=self.num_sanity_val_steps)

        try:
            self._fit_loop(wrapped, train_loader, val_loader)
        finally:
            wrapped.on_fit_end()
            self._call("on_fit_end")

    def _parse_optim_output(self, out):
        scheduler_cfg = None
        if isinstance(out, torch.optim.Optimizer):
            return out, None
        if isinstance(out, dict):
            optimizer = out["optimizer"]
            lrs = out.get("lr_scheduler")
            if lrs is not None:
                scheduler_cfg = lrs if isinstance(lrs, dict) else {
                    "scheduler": lrs, "interval": "step"}
                scheduler_cfg.setdefault("interval", "step")
            return optimizer, scheduler_cfg
        if isinstance(out, (list, tuple)) and len(out) == 2:
            opts, scheds = out
            optimizer = opts[0] if isinstance(opts, (list, tuple)) else opts
            sch = scheds[0] if isinstance(scheds, (list, tuple)) else scheds
            if isinstance(sch, dict):
                scheduler_cfg = sch
                scheduler_cfg.setdefault("interval", "step")
            elif sch is not None:
                scheduler_cfg = {"scheduler": sch, "interval": "step"}
            return optimizer, scheduler_cfg
        raise ValueError(f"cannot parse configure_optimizers output: {type(out)}")

    # ------------------------------------------------------------------
    def _fit_loop(self, model, train_loader, val_loader):
        dev = self.device
        accum = self.accumulate_grad_batches
        steps_per_epoch = None
        try:
            steps_per_epoch = len(train_loader)
        except TypeError:
            pass

        max_epochs = self.max_epochs if self.max_epochs > 0 else (
            10 ** 8 if self.max_steps > 0 else 1)

        while self.current_epoch < max_epochs and not self.should_stop:
            model.train()
            model.on_train_epoch_start()
            self._call("on_train_epoch_start")
            # Prefer batch_sampler: UniversalDataModule builds DataLoader with
            # batch_sampler=, leaving .sampler a default SequentialSampler, so
            # checking only .sampler would never reseed the epoch shuffle.
            for samp in (getattr(train_loader, "batch_sampler", None),
                         getattr(train_loader, "sampler", None)):
                if samp is not None and hasattr(samp, "set_epoch"):
                    samp.set_epoch(self.current_epoch)
                    break

            micro_idx = 0
            for batch_idx, batch in enumerate(train_loader):
                if (self.limit_train_batches is not None
                        and batch_idx >= self.limit_train_batches):
                    break
                # guard BEFORE the step so resuming at global_step ==
                # max_steps trains zero further steps
                if self.max_steps > 0 and self.global_step >= self.max_steps:
                    self.should_stop = True
                    break
                batch = _move_to_device(batch, dev)
                model.on_train_batch_start(batch, batch_idx)
                self._call("on_train_batch_start", batch, batch_idx)

                is_sync = (micro_idx + 1) % accum == 0
                self.strategy.set_sync(is_sync)
                out = model.training_step(batch, batch_idx)
                loss = out["loss"] if isinstance(out, dict) else out
                if accum > 1:
                    loss = loss / accum
                self.strategy.backward(loss)
                self._count_samples(batch)
                micro_idx += 1
                self._did_step = False

                if is_sync:
                    self.strategy.pre_step()
                    self._clip_and_step(model)
                    self._did_step = True
                    self.global_step += 1
                    if (self.scheduler_cfg is not None
                            and self.scheduler_cfg.get("interval", "step") == "step"):
                        self.scheduler_cfg["scheduler"].step()

                model.on_train_batch_end(out, batch, batch_idx)
                self._call("on_train_batch_end", out, batch, batch_idx)

                if self._did_step:
                    if self.global_step % self.log_every_n_steps == 0:
                        self._flush_logs()
                    if self._should_check_val(steps_per_epoch) and val_loader is not None:
                        self._run_validation(val_loader)
                        model.train()
                    if self.max_steps > 0 and self.global_step >= self.max_steps:
                        self.should_stop = True
                        break

            if accum > 1 and micro_idx % accum != 0 and not self._did_step:
                # Trailing micro-batches (epoch length not divisible by accum)
                # accumulated gradients but never stepped; flush a final
                # optimizer step so they don't leak into the next epoch.
                self.strategy.set_sync(True)
                self.strategy.pre_step()
                self._clip_and_step(model)
                self._did_step = True
                self.global_step += 1
                if (self.scheduler_cfg is not None
                        and self.scheduler_cfg.get("interval", "step") == "step"):
                    self.scheduler_cfg["scheduler"].step()

            model.on_train_epoch_end()
            self._call("on_train_epoch_end")
            if (self.scheduler_cfg is not None
                    and self.scheduler_cfg.get("interval") == "epoch"):
                self.scheduler_cfg["scheduler"].step()
            if (val_loader is not None and self.check_val_every_n_epoch > 0
                    and (self.current_epoch + 1) % self.check_val_every_n_epoch == 0
                    and self.val_check_interval is None):
                self._run_validation(val_loader)
                model.train()
            self.current_epoch += 1
        self._flush_logs()
        for lg in self.loggers:
            lg.finalize()

    def _clip_and_step(self, model):
        from fengshen_amd.parallel.zero import ZeroOptimizer
        from fengshen_amd.parallel.zero3 import Zero3Engine

        if isinstance(self.optimizer, (ZeroOptimizer, Zero3Engine)):
            self.optimizer.clip_grad = self.gradient_clip_val or 0.0
            self.optimizer.step()
            if getattr(self.optimizer, "_last_grad_norm", None) is not None:
                self._log_metric("grad_norm", self.optimizer._last_grad_norm)
        else:
            norm = self.strategy.clip_gradients(
                self.optimizer, model, self.gradient_clip_val)
            if norm is not None:
                self._log_metric("grad_norm", norm)
            self.optimizer.step()
        self.optimizer.zero_grad()

    def _count_samples(self, batch):
        from fengshen_amd.trainer.callbacks import _batch_size_tokens
        bs, _ = _batch_size_tokens(batch)
        self.global_samples += bs * self.strategy.data_parallel_world_size

    def _should_check_val(self, steps_per_epoch) -> bool:
        v = self.val_check_interval
        if v is None:
            return False
        if isinstance(v, float) and v <= 1.0 and steps_per_epoch:
            interval = max(1, int(v * steps_per_epoch / self.accumulate_grad_batches))
        else:
            interval = max(1, int(v))
        return self.global_step > 0 and self.global_step % interval == 0

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _run_validation(self, val_loader, limit: Optional[int] = None):
        model = self.module
        model.eval()
        model.on_validation_epoch_start()
        self._call("on_validation_start")
        limit = limit if limit is not None else self.limit_val_batches
        for i, batch in enumerate(val_loader):
            if limit is not None and i >= limit:
                break
            batch = _move_to_device(batch, self.device)
            model.validation_step(batch, i)
        model.on_validation_epoch_end()
        self._call("on_validation_end")
        self._flush_logs()

    def validate(self, model, datamodule=None, dataloaders=None):
        self.strategy.setup_environment(self)
        self.module = model
        model.trainer = self
        if datamodule is not None:
            datamodule.trainer = self
            datamodule.setup("validate")
            dataloaders = datamodule.val_dataloader()
        model.setup("validate")
        wrapped = self.strategy.setup_model(model, self.device, self.precision)
        self.module = wrapped
        self._run_validation(dataloaders)
        return dict(self._metrics)

    @torch.no_grad()
    def predict(self, model, dataloaders=None, datamodule=None):
        if dataloaders is None and datamodule is not None:
            datamodule.setup("predict")
            for name in ("predict_dataloader", "test_dataloader",
                         "val_dataloader"):
                fn = getattr(datamodule, name, None)
                if fn is not None:
                    try:
                        dataloaders = fn()
                    except Exception:
                        dataloaders = None
                    if dataloaders is not None:
                        break
        if dataloaders is None:
            raise ValueError("predict needs dataloaders or a datamodule")
        self.strategy.setup_environment(self)
        self.module = model
        model.trainer = self
        model.setup("predict")
        wrapped = self.strategy.setup_model(model, self.device, self.precision)
        wrapped.eval()
        outs = []
        for i, batch in enumerate(dataloaders):
            batch = _move_to_device(batch, self.device)
            outs.append(wrapped.predict_step(batch, i))
        return outs

    # ------------------------------------------------------------------
    # checkpointing
    # ------------------------------------------------------------------
    def _ckpt_paths(self, ckpt_dir: str):
        from fengshen_amd.parallel import groups as pg
        tp = pg.get_tensor_model_parallel_rank()
        dp = pg.get_data_parallel_rank()
        return {
            "meta": os.path.join(ckpt_dir, "meta.pt"),
            "model": os.path.join(ckpt_dir, f"model_part_{tp}.pt"),
            "optim": os.path.join(ckpt_dir, f"optim_dp{dp}_tp{tp}.pt"),
        }

    def save_checkpoint(self, ckpt_dir: str, weights_only: bool = False):
        from fengshen_amd.parallel import groups as pg
        os.makedirs(ckpt_dir, exist_ok=True)
        paths = self._ckpt_paths(ckpt_dir)
        checkpoint: Dict[str, Any] = {
            "global_step": self.global_step,
            "global_samples": self.global_samples,
            "epoch": self.current_epoch,
            "fengshen_amd_version": 1,
        }
        self.module.on_save_checkpoint(checkpoint)
        self._call("on_save_checkpoint", checkpoint)
        # model shard: saved once per TP rank by DP rank 0; ZeRO-3 params
        # must be materialized for state_dict
        from fengshen_amd.parallel.zero3 import Zero3Engine
        if isinstance(self.optimizer, Zero3Engine):
            with self.optimizer.gathered_params():
                if pg.get_data_parallel_rank() == 0:
                    sd = {k: v.detach().clone()
                          for k, v in self.module.state_dict().items()}
                    torch.save(sd, paths["model"])
        elif pg.get_data_parallel_rank() == 0:
            torch.save(self.module.state_dict(), paths["model"])
        if not weights_only and self.optimizer is not None:
            payload = {"optimizer": self.optimizer.state_dict()}
            if self.scheduler_cfg is not None:
                payload["scheduler"] = self.scheduler_cfg["scheduler"].state_dict()
            torch.save(payload, paths["optim"])
        if self.global_rank == 0:
            checkpoint["rng"] = {
                "torch": torch.get_rng_state(),
                "numpy": np.random.get_state(),
                "python": random.getstate(),
            }
            torch.save(checkpoint, paths["meta"])
        self.strategy.barrier()

    def _peek_checkpoint(self, ckpt_dir: str) -> dict:
        meta = torch.load(os.path.join(ckpt_dir, "meta.pt"),
                          map_location="cpu", weights_only=False)
        return meta

    def _load_checkpoint(self, ckpt_dir: str):
        paths = self._ckpt_paths(ckpt_dir)
        meta = self._peek_checkpoint(ckpt_dir)
        state = torch.load(paths["model"], map_location="cpu", weights_only=False)
        from fengshen_amd.parallel.zero3 import Zero3Engine
        if isinstance(self.optimizer, Zero3Engine):
            # params are sharded stubs: materialize, load, re-shard
            with self.optimizer.gathered_params():
                self.module.load_state_dict(state)
                self.optimizer.refresh_shards_from_params()
        else:
            self.module.load_state_dict(state)
        # re-sync flat buffers in ZeRO (params were re-assigned by load? no:
        # load_state_dict copies INTO the flat views, so buffers are current)
        if os.path.exists(paths["optim"]) and self.optimizer is not None:
            payload = torch.load(paths["optim"], map_location="cpu", weights_only=False)
            try:
                self.optimizer.load_state_dict(payload["optimizer"])
            except Exception as e:  # optimizer topology changed
                logger.warning("optimizer state not restored: %s", e)
            if self.scheduler_cfg is not None and "scheduler" in payload:
                sch = self.scheduler_cfg["scheduler"]
                sch.load_state_dict(payload["scheduler"])
                # refresh param_groups' lr immediately (torch schedulers only
                # apply the restored lr on the NEXT .step())
                try:
                    for g, lr in zip(self.optimizer.param_groups, sch.get_last_lr()):
                        g["lr"] = lr
                except Exception:
                    pass
        self.module.on_load_checkpoint(meta)
        self._call("on_load_checkpoint", meta)
        logger.info("resumed from %s at step %d (samples %d)", ckpt_dir,
                    self.global_step, self.global_samples)
